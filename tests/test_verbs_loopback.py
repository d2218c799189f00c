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


def test_verbs_bad_rkey_rejected(cpu_server):
    """The mock provider validates rkeys against registered MRs (like a real
    NIC): a write posted with a doctored rkey must fail loudly, not silently
    succeed. Round 1's mock skipped this check, which let the server return
    placeholder rkeys in allocate responses without any test noticing."""
    conn = verbs_client(cpu_server)
    try:
        src = torch.randn(4096, dtype=torch.float32)
        conn.register_mr(src)
        key = f"vk-{uuid.uuid4().hex[:8]}"
        blocks = conn.allocate_rdma([key], 4096 * 4)
        assert len(blocks) == 1
        rkey, addr = tuple(blocks[0])[0], tuple(blocks[0])[1]
        assert rkey != 0 and addr != 0
        doctored = [(rkey + 7, addr)]
        with pytest.raises(Exception):
            conn.rdma_write_cache(src, [0], 4096, doctored)
            conn.sync()
    finally:
        conn.close()


def test_verbs_extended_arena_registered(ports):
    """Arenas added by pool auto-extension must be MR-registered while the
    verbs fabric is live: allocate past the first arena's capacity, then
    write+read a block that can only live in the extended arena. Round 1
    registered arenas only once at the first verbs handshake, so this failed
    with 'pool arena not registered'."""
    import time

    import infinistore_amd as ifs

    service_port, manage_port = ports
    cfg = ifs.ServerConfig(
        service_port=service_port,
        manage_port=manage_port,
        prealloc_size=1,       # GB
        extend_size=1,         # GB
        auto_increase=True,
        minimal_allocate_size=1024,  # KB -> 1 MB blocks, 1024 per arena
        cpu_only=True,
    )
    ifs.register_server(cfg)
    try:
        conn = verbs_client(service_port)
        try:
            page_elems = 256 * 1024  # 1 MB of float32
            run = uuid.uuid4().hex[:8]
            total = 0
            target = 1088  # > 1024 => the tail must come from arena 2
            batch = 64
            last_keys = None
            deadline = time.time() + 60
            while total < target and time.time() < deadline:
                keys = [f"ve-{run}-{total + i}" for i in range(batch)]
                try:
                    blocks = conn.allocate_rdma(keys, page_elems * 4)
                except Exception:
                    time.sleep(0.3)  # extension still in flight; retry
                    continue
                assert len(blocks) == batch
                total += batch
                last_keys = (keys, blocks)
            assert total >= target, f"only {total} blocks allocated"
            keys, blocks = last_keys
            src = torch.randn(page_elems, dtype=torch.float32)
            dst = torch.zeros_like(src)
            conn.register_mr(src)
            conn.register_mr(dst)
            conn.rdma_write_cache(src, [0], page_elems, [tuple(blocks[0])])
            conn.sync()
            conn.read_cache(dst, [(keys[0], 0)], page_elems)
            conn.sync()
            assert torch.equal(src, dst)
        finally:
            conn.close()
    finally:
        ifs.unregister_server()


def test_verbs_missing_key(cpu_server):
    """The server answers a read of a missing key with an error IMM; the
    client must fail FAST (not wait out its 10 s CQ timeout)."""
    import time

    conn = verbs_client(cpu_server)
    try:
        dst = torch.zeros(1024, dtype=torch.float32)
        conn.register_mr(dst)
        t0 = time.time()
        with pytest.raises(Exception):
            conn.read_cache(dst, [(f"vm-{uuid.uuid4().hex}", 0)], 1024)
            conn.sync()
        assert time.time() - t0 < 5.0, "missing-key read should error via IMM, not timeout"
    finally:
        conn.close()
