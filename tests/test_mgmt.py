"""Management-plane tests: FastAPI endpoints (purge, kvmap_len, stats,
compact, selftest) exercised with the in-process server + TestClient."""

import json

import pytest
import torch

import infinistore_amd as ifs
from conftest import make_client

fastapi = pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402


@pytest.fixture
def app_client(cpu_server):
    from infinistore_amd import server as srv

    cfg = ifs.ServerConfig(service_port=cpu_server, manage_port=1, log_level="warning")
    app = srv.make_app(cfg)
    with TestClient(app) as tc:
        yield tc, cpu_server


def test_kvmap_len_and_purge(app_client):
    tc, port = app_client
    conn = make_client(port)
    try:
        src = torch.zeros(1024)
        conn.register_mr(src)
        blocks = conn.allocate_rdma(["mgmt-a", "mgmt-b"], 2048)
        conn.rdma_write_cache(src, [0, 512], 512, blocks)
        conn.sync()
        assert tc.get("/kvmap_len").json()["len"] == 2
        stats = tc.get("/stats").json()
        assert stats["kv_len"] == 2 and stats["puts"] >= 1
        assert "op_us" in stats
        r = tc.post("/purge").json()
        assert r["count"] == 2
        assert tc.get("/kvmap_len").json()["len"] == 0
    finally:
        conn.close()


def test_compact_endpoint(app_client):
    tc, port = app_client
    r = tc.post("/compact").json()
    assert "moved_blocks" in r


def test_selftest_endpoint(app_client):
    tc, port = app_client
    r = tc.post(f"/selftest/{port}").json()
    assert r["status"] == "ok"


def test_metrics_endpoint(app_client):
    tc, cpu_server = app_client
    conn = make_client(cpu_server)
    try:
        src = torch.zeros(256)
        conn.register_mr(src)
        blocks = conn.allocate_rdma(["metrics-key"], 1024)
        conn.rdma_write_cache(src, [0], 256, blocks)
        conn.sync()
    finally:
        conn.close()
    text = tc.get("/metrics").text
    assert "infinistore_kv_len 1" in text
    assert 'infinistore_op_count{op="allocate"}' in text
    assert "# TYPE infinistore_bytes_in_total gauge" in text


def test_stats_json_parses():
    s = ifs.get_server_stats()
    json.loads(s)


def test_snapshot_restore(cpu_server, tmp_path):
    """Warm-restart persistence: snapshot committed pages, purge, restore —
    data and metadata round-trip; live keys win over the snapshot."""
    conn = make_client(cpu_server)
    src = torch.arange(8192, dtype=torch.float32)
    dst = torch.zeros_like(src)
    conn.register_mr(src)
    conn.register_mr(dst)
    keys = [f"snap-{i}" for i in range(4)]
    blocks = conn.allocate_rdma(keys, 2048 * 4)
    for i, b in enumerate(blocks):
        conn.rdma_write_cache(src, [i * 2048], 2048, [b])
    conn.sync()

    snap = str(tmp_path / "pool.snap")
    n, nbytes = ifs.snapshot_pool(snap)
    assert n == 4 and nbytes == 4 * 2048 * 4

    assert ifs.purge_kv_map() >= 4
    assert not conn.check_exist("snap-0")

    rn, rbytes = ifs.restore_pool(snap)
    assert rn == 4 and rbytes == nbytes
    conn.read_cache(dst, [(k, i * 2048) for i, k in enumerate(keys)], 2048)
    conn.sync()
    assert torch.equal(src, dst)

    # restore into a live index: existing keys win, duplicates are skipped
    rn2, _ = ifs.restore_pool(snap)
    assert rn2 == 0
    conn.close()


def test_snapshot_http_and_cli_restore(tmp_path):
    """The full warm-restart story across processes: snapshot over HTTP,
    then a FRESH server process started with --restore-from serves the
    same keys."""
    import subprocess
    import sys
    import time

    from conftest import free_port

    repo = __import__("os").path.dirname(__import__("os").path.dirname(
        __import__("os").path.abspath(__file__)))
    snap = str(tmp_path / "cli.snap")

    # phase 1: in-process server, write keys, snapshot via the HTTP app
    port1 = free_port()
    cfg = ifs.ServerConfig(service_port=port1, manage_port=free_port(),
                           prealloc_size=1, minimal_allocate_size=16,
                           cpu_only=True)
    ifs.register_server(cfg)
    try:
        from infinistore_amd import server as srv
        with TestClient(srv.make_app(cfg)) as tc:
            conn = make_client(port1)
            src = torch.arange(2048, dtype=torch.float32)
            conn.register_mr(src)
            blocks = conn.allocate_rdma(["cli-a", "cli-b"], 4096)
            conn.rdma_write_cache(src, [0], 1024, [blocks[0]])
            conn.rdma_write_cache(src, [1024], 1024, [blocks[1]])
            conn.sync()
            conn.close()
            r = tc.post(f"/snapshot?path={snap}").json()
            assert r["entries"] == 2
    finally:
        ifs.unregister_server()

    # phase 2: fresh server PROCESS restores the snapshot at startup
    port2 = free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "infinistore_amd.server",
         "--service-port", str(port2), "--manage-port", str(free_port()),
         "--prealloc-size", "1", "--minimal-allocate-size", "16",
         "--cpu-only", "--no-manage", "--restore-from", snap],
        cwd=repo,
    )
    try:
        t0 = time.time()
        conn = None
        while time.time() - t0 < 60:
            try:
                conn = make_client(port2)
                break
            except Exception:
                time.sleep(0.3)
        assert conn is not None
        dst = torch.zeros(2048, dtype=torch.float32)
        conn.register_mr(dst)
        conn.read_cache(dst, [("cli-a", 0), ("cli-b", 1024)], 1024)
        conn.sync()
        src = torch.arange(2048, dtype=torch.float32)
        assert torch.equal(src, dst)
        conn.close()
    finally:
        proc.terminate()
        proc.wait(timeout=20)
