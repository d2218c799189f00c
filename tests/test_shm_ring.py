"""Shared-memory ring transport: framing/wraparound unit tests (CPU) via the
_dbg_shmring_echo harness, plus the CPU-server handshake behavior.

The full client→ring→server→completion-thread path needs a GPU (the packed
ops are local-GPU ops) and is covered by tests/test_gpu.py::test_shm_transport.
"""

import random

from infinistore_amd import _native

from conftest import make_client


def _echo(msgs, cap=256, drain_every=1):
    out = _native._dbg_shmring_echo([bytes(m) for m in msgs], cap, drain_every)
    return [bytes(o) for o in out]


def test_roundtrip_simple():
    msgs = [b"hello", b"", b"x" * 40, b"yz"]
    assert _echo(msgs) == msgs


def test_wraparound():
    # Records of varying size forced around a tiny ring many times.
    random.seed(7)
    msgs = [bytes(random.getrandbits(8) for _ in range(random.randint(0, 100)))
            for _ in range(500)]
    assert _echo(msgs, cap=256, drain_every=1) == msgs


def test_backpressure_batches():
    # Producer runs ahead until the ring fills, then drains in bursts.
    random.seed(11)
    msgs = [bytes([i % 256]) * (i % 60) for i in range(300)]
    for cap in (128, 512, 4096):
        for de in (3, 7, 50):
            assert _echo(msgs, cap=cap, drain_every=de) == msgs


def test_exact_capacity_records():
    # Records that exactly exhaust the ring (no room for a wrap marker).
    cap = 128
    body = cap - 24  # one record == cap
    msgs = [bytes([i]) * body for i in range(10)]
    assert _echo(msgs, cap=cap, drain_every=1) == msgs


def test_cpu_server_accepts_handshake(cpu_server):
    # On a CPU-only server the handshake itself is transport-level and
    # succeeds; the packed ops then fail with SYSTEM_ERROR through the ring
    # exactly as they do through the socket. The Python client only attempts
    # setup for LOCAL_GPU connections, so a default (RDMA-type) client
    # reports the ring inactive.
    conn = make_client(cpu_server)
    assert not conn.conn.shm_active()
    conn.close()
