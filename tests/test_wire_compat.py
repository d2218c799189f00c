"""Byte-level wire compatibility with the reference protocol.

A hand-rolled raw-socket client speaks the reference's exact TCP framing at
this server — 9-byte header {u32 magic 0xdeadbeef, char op, u32 body_size},
raw-key check_exist, FINISH + 4-byte-int responses for the query ops
(reference: /root/reference/src/libinfinistore.cpp:632-724,
/root/reference/src/infinistore.cpp:1055-1108) — with the flatbuffer bodies
built by an independent, spec-derived Python builder (NOT this repo's
wire.h), standing in for the reference's flatc-generated code. If these
transcripts round-trip, a reference client's bytes would too.
"""

import socket
import struct

from conftest import make_client

MAGIC = 0xDEADBEEF
FINISH = 200


# --- independent minimal flatbuffers builder (spec-derived) -----------------
def fb_build_keys_table(keys):
    """Build a flatbuffers table with field 0 = [string] (the layout of
    GetMatchLastIndexRequest, reference get_match_last_index.fbs)."""
    buf = bytearray()

    def pad_to(align):
        while len(buf) % align:
            buf.append(0)

    # root offset placeholder
    buf += b"\0\0\0\0"
    # vtable: vt_len=6, table_len=8, field0 slot=4
    pad_to(2)
    vt_pos = len(buf)
    buf += struct.pack("<HHH", 6, 8, 4)
    # table
    pad_to(4)
    t_pos = len(buf)
    buf += struct.pack("<i", t_pos - vt_pos)  # soffset back to vtable
    f0_pos = len(buf)
    buf += b"\0\0\0\0"  # field 0: u32 forward offset to the vector (patched)
    # vector of string offsets
    pad_to(4)
    vec_pos = len(buf)
    buf += struct.pack("<I", len(keys))
    elem_pos = len(buf)
    buf += b"\0\0\0\0" * len(keys)  # patched below
    # strings
    str_pos = []
    for k in keys:
        pad_to(4)
        str_pos.append(len(buf))
        kb = k.encode()
        buf += struct.pack("<I", len(kb)) + kb + b"\0"
    # patch forward offsets (flatbuffers uoffsets are target - location)
    struct.pack_into("<I", buf, 0, t_pos)
    struct.pack_into("<I", buf, f0_pos, vec_pos - f0_pos)
    for i, sp in enumerate(str_pos):
        struct.pack_into("<I", buf, elem_pos + 4 * i, sp - (elem_pos + 4 * i))
    return bytes(buf)


class RefFramedClient:
    """Raw TCP client using the reference's framing only."""

    def __init__(self, port):
        self.sock = socket.create_connection(("127.0.0.1", port), timeout=30)

    def send(self, op, body=b""):
        self.sock.sendall(struct.pack("<IcI", MAGIC, op, len(body)) + body)

    def recv_exact(self, n):
        out = b""
        while len(out) < n:
            chunk = self.sock.recv(n - len(out))
            assert chunk, "server closed the socket"
            out += chunk
        return out

    def status_plus_int(self):
        """Reference query response: 4-byte status then 4-byte value."""
        code, val = struct.unpack("<ii", self.recv_exact(8))
        return code, val

    def close(self):
        self.sock.close()


def put_keys(port, keys, page_elems=256):
    """Store committed keys via the normal client (TCP fabric)."""
    import torch

    conn = make_client(port)
    src = torch.randn(page_elems * len(keys), dtype=torch.float32)
    conn.register_mr(src)
    blocks = conn.allocate_rdma(keys, page_elems * 4)
    conn.rdma_write_cache(src, [i * page_elems for i in range(len(keys))],
                          page_elems, blocks)
    conn.sync()
    conn.close()


def test_ref_framed_check_exist(cpu_server):
    c = RefFramedClient(cpu_server)
    try:
        # missing key: FINISH + 1 (reference check_key: 0 = exists)
        c.send(b"C", b"definitely-missing")
        assert c.status_plus_int() == (FINISH, 1)
        put_keys(cpu_server, ["wirecompat-k1"])
        c.send(b"C", b"wirecompat-k1")
        assert c.status_plus_int() == (FINISH, 0)
    finally:
        c.close()


def test_ref_framed_sync(cpu_server):
    c = RefFramedClient(cpu_server)
    try:
        # nothing in flight on this conn: FINISH + remain 0
        c.send(b"S")
        assert c.status_plus_int() == (FINISH, 0)
    finally:
        c.close()


def test_ref_framed_match_last_index(cpu_server):
    keys = [f"wc-m-{i}" for i in range(4)]
    put_keys(cpu_server, keys)
    c = RefFramedClient(cpu_server)
    try:
        # reference scenario (test_infinistore.py:258-275): stored prefix of
        # 4, query a longer chain -> last matching index = 3
        body = fb_build_keys_table(keys + ["wc-m-absent-1", "wc-m-absent-2"])
        c.send(b"M", body)
        assert c.status_plus_int() == (FINISH, 3)
        # no match at all -> -1
        body = fb_build_keys_table(["wc-nope-a", "wc-nope-b"])
        c.send(b"M", body)
        assert c.status_plus_int() == (FINISH, -1)
    finally:
        c.close()


def test_ref_framed_interleaved_on_one_conn(cpu_server):
    """Several reference-framed ops back-to-back on one connection — the
    2-state header/body machine must re-arm correctly between ops."""
    put_keys(cpu_server, ["wc-seq-1"])
    c = RefFramedClient(cpu_server)
    try:
        for _ in range(3):
            c.send(b"C", b"wc-seq-1")
            assert c.status_plus_int() == (FINISH, 0)
            c.send(b"S")
            assert c.status_plus_int() == (FINISH, 0)
            c.send(b"M", fb_build_keys_table(["wc-seq-1", "wc-seq-absent"]))
            assert c.status_plus_int() == (FINISH, 0)
    finally:
        c.close()
