"""Fuzz/robustness tests: random bytes against the wire parsers and random
op sequences against a live server — nothing may crash the process."""

import random
import socket
import struct
import uuid

import torch

import infinistore_amd as ifs
from infinistore_amd import _native as n
from conftest import make_client


def test_parser_fuzz_random_bytes():
    rng = random.Random(1234)
    parsers = [
        n._dbg_parse_local_meta,
        n._dbg_parse_remote_meta,
        n._dbg_parse_alloc_resp,
        n._dbg_parse_match_req,
    ]
    for trial in range(300):
        size = rng.choice([0, 1, 4, 8, 16, 64, 256, 1024])
        data = bytes(rng.getrandbits(8) for _ in range(size))
        for p in parsers:
            try:
                p(data)
            except RuntimeError:
                pass  # parse failure is the expected outcome


def test_parser_fuzz_mutated_valid():
    """Bit-flip valid buffers; parsers must fail cleanly or produce garbage
    without crashing."""
    rng = random.Random(99)
    base = n._dbg_build_remote_meta(["abc", "defg"], 4096, 3, [1, 2], 65)
    for trial in range(300):
        b = bytearray(base)
        for _ in range(rng.randint(1, 6)):
            b[rng.randrange(len(b))] ^= 1 << rng.randrange(8)
        try:
            n._dbg_parse_remote_meta(bytes(b))
        except (RuntimeError, UnicodeDecodeError):
            # parse failure, or mutated key bytes that are not valid UTF-8
            # (rejected at the Python boundary)
            pass


def test_server_random_op_stream(cpu_server):
    """Random valid-ish framed requests with garbage bodies interleaved with
    real traffic; the server must keep serving."""
    rng = random.Random(7)
    for trial in range(30):
        s = socket.create_connection(("127.0.0.1", cpu_server), timeout=5)
        op = rng.choice(b"RWSEDATCMPGXwr??")
        body = bytes(rng.getrandbits(8) for _ in range(rng.randint(0, 200)))
        s.sendall(struct.pack("<IcI", 0xDEADBEEF, bytes([op]), len(body)) + body)
        s.settimeout(2)
        try:
            s.recv(64)
        except socket.timeout:
            pass
        s.close()
    # real traffic still works afterwards
    conn = make_client(cpu_server)
    src = torch.zeros(256)
    conn.register_mr(src)
    key = f"post-fuzz-{uuid.uuid4()}"
    blocks = conn.allocate_rdma([key], 1024)
    conn.rdma_write_cache(src, [0], 256, blocks)
    conn.sync()
    assert conn.check_exist(key)
    conn.close()


def test_match_index_scales(cpu_server):
    """get_match_last_index must answer via O(log n) lookups even with a
    large stored set and long probe chains."""
    import time

    conn = make_client(cpu_server)
    try:
        src = torch.zeros(256)
        conn.register_mr(src)
        pre = uuid.uuid4().hex
        # store a 2000-page chain (committed)
        keys = [f"{pre}-{i}" for i in range(2000)]
        blocks = conn.allocate_rdma(keys, 1024)
        offs = [0] * 0
        for start in range(0, 2000, 500):
            conn.rdma_write_cache(
                src, [0] * 500, 256, blocks[start : start + 500]
            )
        conn.sync()
        probe = keys + [f"missing-{i}" for i in range(2000)]
        t0 = time.perf_counter()
        idx = conn.get_match_last_index(probe)
        dt = time.perf_counter() - t0
        assert idx == 1999
        assert dt < 0.5  # one RTT + O(log n) lookups, not O(n) scans
    finally:
        conn.close()
