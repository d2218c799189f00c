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
