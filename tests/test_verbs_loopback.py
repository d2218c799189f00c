"""ibverbs fabric exercised at RUNTIME against the in-process loopback
provider (tests/mock_verbs/mock_verbs.cpp). This environment has no
rdma-core and no NIC, so the production verbs code is otherwise dead weight
at test time; under `scripts/san_build.py mockverbs` the real code paths
run: driver bring-up, RC QP INIT→RTR→RTS with conn-info exchanged in the
'E' op, allocate over IBV_WR_SEND into pre-posted 4 MB recv buffers,
payload writes as RDMA_WRITE chains (WrFlow ≤32 WR / ≤4096 outstanding),
and server-pushed reads finishing in RDMA_WRITE_WITH_IMM.

These tests are SKIPPED on the normal build (negotiation falls back to the
TCP fabric there); san_build's mockverbs mode runs them for real.
"""

import uuid

import pytest
import torch

import infinistore_amd as ifs


def verbs_client(port):
    cfg = ifs.ClientConfig(
        host_addr="127.0.0.1",
        service_port=port,
        connection_type=ifs.TYPE_RDMA,
        link_type="IB",  # "TCP" skips the verbs negotiation entirely
    )
    conn = ifs.InfinityConnection(cfg)
    conn.connect()
    if not conn.conn.using_verbs():
        conn.close()
        pytest.skip("verbs fabric not active (normal build: no mock provider)")
    return conn


def test_verbs_negotiation(cpu_server):
    conn = verbs_client(cpu_server)
    assert conn.rdma_connected
    conn.close()


def test_verbs_roundtrip(cpu_server):
    conn = verbs_client(cpu_server)
    try:
        src = torch.randn(65536, dtype=torch.float32)
        dst = torch.zeros_like(src)
        conn.register_mr(src)
        conn.register_mr(dst)
        run = uuid.uuid4().hex[:8]
        page = 4096  # elements
        keys = [f"v-{run}-{i}" for i in range(16)]
        offsets = [i * page for i in range(16)]
        blocks = conn.allocate_rdma(keys, page * 4)
        assert len(blocks) == 16
        conn.rdma_write_cache(src, offsets, page, blocks)
        conn.sync()
        conn.read_cache(dst, list(zip(keys, offsets)), page)
        conn.sync()
        assert torch.equal(src, dst)
    finally:
        conn.close()


def test_verbs_dedup_and_queries(cpu_server):
    conn = verbs_client(cpu_server)
    try:
        src = torch.arange(4096, dtype=torch.float32)
        conn.register_mr(src)
        key = f"vd-{uuid.uuid4().hex[:8]}"
        b1 = conn.allocate_rdma([key], 4096 * 4)
        conn.rdma_write_cache(src, [0], 4096, b1)
        conn.sync()
        # second allocate of the same key -> FAKE block (first write wins)
        b2 = conn.allocate_rdma([key], 4096 * 4)
        assert tuple(b2[0])[1] == 0
        assert conn.check_exist(key)
        assert conn.get_match_last_index([key]) == 0
    finally:
        conn.close()


def test_verbs_many_blocks_wr_chaining(cpu_server):
    """>32 blocks per request exercises the ≤32-WR chain splitting and the
    outstanding-WR accounting in WrFlow on both ends."""
    conn = verbs_client(cpu_server)
    try:
        n, page = 300, 1024  # elements
        src = torch.randn(n * page, dtype=torch.float32)
        dst = torch.zeros_like(src)
        conn.register_mr(src)
        conn.register_mr(dst)
        run = uuid.uuid4().hex[:8]
        keys = [f"vc-{run}-{i}" for i in range(n)]
        offsets = [i * page for i in range(n)]
        blocks = conn.allocate_rdma(keys, page * 4)
        conn.rdma_write_cache(src, offsets, page, blocks)
        conn.sync()
        conn.read_cache(dst, list(zip(keys, offsets)), page)
        conn.sync()
        assert torch.equal(src, dst)
    finally:
        conn.close()


def test_verbs_async_api(cpu_server):
    """allocate_rdma_async / rdma_write_cache_async / read_cache_async over
    the verbs data plane (client CQ thread + callback marshalling)."""
    import asyncio

    conn = verbs_client(cpu_server)
    try:
        src = torch.randn(8192, dtype=torch.float32)
        dst = torch.zeros_like(src)
        conn.register_mr(src)
        conn.register_mr(dst)
        run = uuid.uuid4().hex[:8]
        keys = [f"va-{run}-{i}" for i in range(8)]
        offsets = [i * 1024 for i in range(8)]

        async def go():
            blocks = await conn.allocate_rdma_async(keys, 1024 * 4)
            await conn.rdma_write_cache_async(src, offsets, 1024, blocks)
            conn.sync()
            await conn.read_cache_async(dst, list(zip(keys, offsets)), 1024)
            conn.sync()

        asyncio.run(go())
        assert torch.equal(src, dst)
    finally:
        conn.close()


def test_verbs_missing_key(cpu_server):
    conn = verbs_client(cpu_server)
    try:
        dst = torch.zeros(1024, dtype=torch.float32)
        conn.register_mr(dst)
        with pytest.raises(Exception):
            conn.read_cache(dst, [(f"vm-{uuid.uuid4().hex}", 0)], 1024)
            conn.sync()
    finally:
        conn.close()
