"""Concurrency / robustness stress tests (CPU, TCP fabric)."""

import socket
import struct
import threading
import uuid

import pytest
import torch

import infinistore_amd as ifs
from conftest import make_client


def test_many_threads_one_server(cpu_server):
    """8 threads × separate connections hammer mixed ops concurrently."""
    errors = []

    def worker(tid):
        try:
            conn = make_client(cpu_server)
            src = torch.full((2048,), float(tid))
            dst = torch.zeros(2048)
            conn.register_mr(src)
            conn.register_mr(dst)
            pre = f"t{tid}-{uuid.uuid4().hex}"
            for it in range(5):
                keys = [f"{pre}-{it}-{i}" for i in range(4)]
                blocks = conn.allocate_rdma(keys, 512 * 4)
                conn.rdma_write_cache(src, [0, 512, 1024, 1536], 512, blocks)
                conn.sync()
                conn.read_cache(dst, list(zip(keys, [0, 512, 1024, 1536])), 512)
                conn.sync()
                assert torch.equal(src, dst)
                assert conn.get_match_last_index(keys) == 3
                conn.delete_keys(keys[2:])
                assert not conn.check_exist(keys[3])
            conn.close()
        except Exception as e:  # pragma: no cover
            errors.append(f"t{tid}: {e}")

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors


def test_async_ops_interleaved(cpu_server):
    """Async writes from the worker thread interleaved with sync queries on
    the caller thread (the vLLM usage pattern)."""
    import asyncio

    async def run():
        cfg = ifs.ClientConfig(
            host_addr="127.0.0.1", service_port=cpu_server,
            connection_type=ifs.TYPE_RDMA, link_type="TCP",
        )
        conn = ifs.InfinityConnection(cfg)
        await conn.connect_async()
        try:
            src = torch.arange(8192, dtype=torch.float32)
            conn.register_mr(src)
            pre = uuid.uuid4().hex
            futs = []
            for it in range(8):
                keys = [f"{pre}-{it}-{i}" for i in range(4)]
                blocks = await conn.allocate_rdma_async(keys, 512 * 4)
                futs.append(conn.rdma_write_cache_async(
                    src, [0, 512, 1024, 1536], 512, blocks))
                conn.check_exist(f"{pre}-0-0")  # interleaved sync query
            await asyncio.gather(*futs)
            conn.sync()
            assert conn.check_exist(f"{pre}-7-3")
        finally:
            conn.close()

    asyncio.run(run())


def test_bad_magic_closes_connection(cpu_server):
    s = socket.create_connection(("127.0.0.1", cpu_server), timeout=5)
    s.sendall(struct.pack("<IcI", 0x12345678, b"S", 0))
    # server should close on bad magic
    s.settimeout(5)
    assert s.recv(4) == b""
    s.close()


def test_oversized_body_rejected(cpu_server):
    s = socket.create_connection(("127.0.0.1", cpu_server), timeout=5)
    s.sendall(struct.pack("<IcI", 0xDEADBEEF, b"M", 1 << 30))
    s.settimeout(5)
    assert s.recv(4) == b""
    s.close()


def test_partial_header_then_disconnect(cpu_server):
    # half a header, then hang up — the server must survive.
    s = socket.create_connection(("127.0.0.1", cpu_server), timeout=5)
    s.sendall(b"\xef\xbe\xad")
    s.close()
    # server still serves afterwards
    conn = make_client(cpu_server)
    assert conn.check_exist("still-alive-" + uuid.uuid4().hex) is False
    conn.close()


def test_garbage_flatbuffer_body(cpu_server):
    s = socket.create_connection(("127.0.0.1", cpu_server), timeout=5)
    body = b"\xff" * 64
    s.sendall(struct.pack("<IcI", 0xDEADBEEF, b"M", len(body)) + body)
    s.settimeout(5)
    # reference query framing: FINISH + i32 value (-1 = no match / bad body)
    code, val = struct.unpack("<ii", _recv_exact(s, 8))
    assert code == 200 and val == -1  # error value, but connection alive
    s.sendall(struct.pack("<IcI", 0xDEADBEEF, b"S", 0))
    code, remain = struct.unpack("<ii", _recv_exact(s, 8))
    assert code == 200 and remain == 0
    s.close()


def _recv_exact(s, n):
    out = b""
    while len(out) < n:
        chunk = s.recv(n - len(out))
        assert chunk
        out += chunk
    return out


def test_kv_index_rehash_churn(ports):
    """Force the kv index through growth rehashes and tombstone compaction:
    tiny initial capacity + insert/delete churn, verifying contents
    throughout."""
    import os
    import subprocess
    import sys
    import time as _t

    # The knob is read at server start; run in a subprocess with it set.
    code = r"""
import os, torch, uuid
import infinistore_amd as ifs
port = int(os.environ["PORT"])
ifs.register_server(ifs.ServerConfig(service_port=port, manage_port=port+1,
                                     prealloc_size=1, minimal_allocate_size=16,
                                     cpu_only=True))
cfg = ifs.ClientConfig(host_addr="127.0.0.1", service_port=port,
                       connection_type=ifs.TYPE_RDMA, link_type="TCP")
conn = ifs.InfinityConnection(cfg)
conn.connect()
src = torch.arange(4096, dtype=torch.float32)
dst = torch.zeros(4096, dtype=torch.float32)
conn.register_mr(src); conn.register_mr(dst)
live = {}
import random
rng = random.Random(3)
for gen in range(40):
    keys = [f"g{gen}-{i}-{uuid.uuid4().hex[:6]}" for i in range(64)]
    blocks = conn.allocate_rdma(keys, 1024)
    conn.rdma_write_cache(src, [ (i % 16) * 256 for i in range(64)], 256, blocks)
    conn.sync()
    for i, k in enumerate(keys):
        live[k] = (i % 16) * 256
    # delete a random half of an old generation
    victims = rng.sample(list(live.keys()), min(40, len(live)//2))
    conn.delete_keys(victims)
    for v in victims: live.pop(v)
    # verify a random sample reads back right
    sample = rng.sample(list(live.items()), min(8, len(live)))
    for k, off in sample:
        conn.read_cache(dst, [(k, 0)], 256)
        conn.sync()
        assert torch.equal(dst[:256], src[off:off+256]), k
assert ifs.get_kvmap_len() == len(live)
conn.delete_keys(list(live.keys()))
assert ifs.get_kvmap_len() == 0
conn.close(); ifs.unregister_server()
print("CHURN_OK")
"""
    service_port, _ = ports
    env = dict(os.environ)
    env["IFS_KV_INITIAL"] = "256"
    env["PORT"] = str(service_port)
    r = subprocess.run([sys.executable, "-c", code], env=env, capture_output=True,
                       text=True, timeout=300, cwd=os.path.dirname(os.path.dirname(
                           os.path.abspath(__file__))))
    assert r.returncode == 0 and "CHURN_OK" in r.stdout, r.stdout + r.stderr
