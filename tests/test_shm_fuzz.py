"""Hostile-client fuzzing of the shared-memory ring transport (CPU server).

The ring is client-created shared memory: a buggy or malicious client can
write anything into it. The server poller must survive garbage record
headers, bogus lengths, truncated bodies and unknown ops — replying
INVALID_REQ where it can and never crashing or wedging the connection's
socket side. The segment is built by hand here (shm_open is just a file
in /dev/shm) so no GPU/local-path client is needed.
"""

import mmap
import os
import random
import socket
import struct
import uuid

from conftest import make_client

MAGIC = 0x49465352494E4731  # shm_ring.h kMagic
REQ_CAP = 1 << 16
RESP_CAP = 1 << 14
RING_HDR = 24  # head(8) + tail(8) + cap(4) + pad(4)
CTRL = 32
REC_HDR = 24

WIRE_MAGIC = 0xDEADBEEF


class RawShmConn:
    """Minimal hand-rolled client: UDS control socket + handcrafted segment."""

    def __init__(self, port, req_cap=REQ_CAP, resp_cap=RESP_CAP, req_off_skew=0,
                 expect_code=200):
        self.req_cap = req_cap
        self.resp_cap = resp_cap
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(30)
        self.sock.connect(f"/tmp/infinistore-amd-{port}.sock")
        name = f"/ifs-fuzz-{os.getpid()}-{uuid.uuid4().hex[:8]}"
        path = "/dev/shm" + name
        total = CTRL + RING_HDR + req_cap + RING_HDR + resp_cap + req_off_skew
        with open(path, "wb") as f:
            f.write(b"\0" * total)
        self.f = open(path, "r+b")
        self.mm = mmap.mmap(self.f.fileno(), total)
        req_off = CTRL + req_off_skew
        resp_off = req_off + RING_HDR + req_cap
        struct.pack_into("<QIIIIII", self.mm, 0, MAGIC, 1, req_off, req_cap,
                         resp_off, resp_cap, 0)
        struct.pack_into("<I", self.mm, req_off + 16, req_cap)
        struct.pack_into("<I", self.mm, resp_off + 16, resp_cap)
        self.req_data = req_off + RING_HDR
        self.req_head_off = req_off
        self.resp_off = resp_off
        # handshake
        self._send(b"h", name.encode())
        code = struct.unpack("<i", self._recv(4))[0]
        assert code == expect_code, code
        os.unlink(path)

    def _send(self, op, body):
        self.sock.sendall(struct.pack("<IcI", WIRE_MAGIC, op, len(body)) + body)

    def _recv(self, n):
        out = b""
        while len(out) < n:
            chunk = self.sock.recv(n - len(out))
            assert chunk, "server closed the socket"
            out += chunk
        return out

    def push_raw(self, raw):
        """Append raw bytes as-is at the ring head and publish."""
        head = struct.unpack_from("<Q", self.mm, self.req_head_off)[0]
        pos = head % self.req_cap
        assert pos + len(raw) <= self.req_cap  # keep the fuzz simple: no wrap
        self.mm[self.req_data + pos : self.req_data + pos + len(raw)] = raw
        struct.pack_into("<Q", self.mm, self.req_head_off, head + len(raw))

    def set_cursors(self, head, tail):
        """Hostile: teleport head and tail (tail is normally server-owned)."""
        struct.pack_into("<Q", self.mm, self.req_head_off, head)
        struct.pack_into("<Q", self.mm, self.req_head_off + 8, tail)

    def write_at(self, pos, raw):
        """Write raw bytes at ring offset pos without touching cursors."""
        self.mm[self.req_data + pos : self.req_data + pos + len(raw)] = raw

    def publish(self, adv):
        head = struct.unpack_from("<Q", self.mm, self.req_head_off)[0]
        struct.pack_into("<Q", self.mm, self.req_head_off, head + adv)

    def push_record(self, op, body, seq, rec_len=None, body_len=None):
        body_len = len(body) if body_len is None else body_len
        need = (REC_HDR + len(body) + 7) & ~7
        rec_len = need if rec_len is None else rec_len
        hdr = struct.pack("<IB3sIIQ", rec_len, op, b"\0\0\0", body_len, 0, seq)
        self.push_raw(hdr + body + b"\0" * (need - REC_HDR - len(body)))

    def pop_responses(self, timeout_s=10.0):
        """Collect (seq, status) pairs until the ring drains."""
        import time

        resp_off = self.resp_off
        out = []
        t0 = time.time()
        while time.time() - t0 < timeout_s:
            head = struct.unpack_from("<Q", self.mm, resp_off)[0]
            tail = struct.unpack_from("<Q", self.mm, resp_off + 8)[0]
            if head == tail:
                if out:
                    return out
                time.sleep(0.01)
                continue
            pos = tail % self.resp_cap
            ln, op = struct.unpack_from("<IB", self.mm, resp_off + RING_HDR + pos)
            seq = struct.unpack_from("<Q", self.mm, resp_off + RING_HDR + pos + 16)[0]
            status = struct.unpack_from("<i", self.mm, resp_off + RING_HDR + pos + 24)[0]
            out.append((seq, status))
            struct.pack_into("<Q", self.mm, resp_off + 8, tail + ln)
        return out

    def close(self):
        self.sock.close()
        self.mm.close()
        self.f.close()


def test_shm_garbage_records(cpu_server):
    c = RawShmConn(cpu_server)
    try:
        random.seed(5)
        # 1. unknown op -> INVALID_REQ response with the right seq
        c.push_record(ord("Z"), b"junk", seq=1)
        assert (1, 400) in c.pop_responses()
        # 2. body_len larger than the record -> INVALID_REQ
        c.push_record(ord("w"), b"tiny", seq=2, body_len=10_000)
        assert (2, 400) in c.pop_responses()
        # 3. well-formed record, garbage packed body -> INVALID_REQ (or a
        #    clean error), server stays up
        c.push_record(ord("w"), bytes(random.getrandbits(8) for _ in range(80)),
                      seq=3)
        got = c.pop_responses()
        assert got and all(s != 0 or q != 3 for q, s in got)
        # 4. random garbage ops keep getting individual error replies
        for i in range(20):
            c.push_record(random.randrange(1, 255), os.urandom(random.randrange(0, 64)),
                          seq=10 + i)
        assert len(c.pop_responses(20)) >= 1
        # 5. the socket side still works after all of that
        c._send(b"C", b"key")  # reference framing: raw key bytes
        code, exist = struct.unpack("<ii", c._recv(8))
        assert code == 200 and exist in (0, 1)
    finally:
        c.close()
    # 6. and the server accepts fresh clients
    conn = make_client(cpu_server)
    assert conn.check_exist("nope") in (0, False)
    conn.close()


def test_shm_straddling_record(cpu_server):
    """A record whose len runs past the ring's end must NOT be followed.

    The legit producer always wraps before the end, so a straddling header
    can only be hand-written; following it reads past the ring (with a large
    ring, past the mapped segment → SIGSEGV pre-fix). The server must detach
    the poller and keep the socket side of the conn alive.
    """
    # 32 MB ring: under the old code len≈cap from a near-end pos reads ~32 MB
    # past the segment end and crashes the server inside shm_poll_main.
    big = 32 << 20
    c = RawShmConn(cpu_server, req_cap=big, resp_cap=RESP_CAP)
    try:
        pos = big - 32  # 32 bytes before the end
        c.set_cursors(pos, pos)  # empty ring, head at the danger zone
        # header: len = cap-8 (8-aligned, <= cap, <= h-t after publish)
        ln = big - 8
        hdr = struct.pack("<IB3sIIQ", ln, ord("w"), b"\0\0\0", ln - REC_HDR, 0, 7)
        c.write_at(pos, hdr)
        c.publish(ln)
        # poller must detach: no response, no crash
        assert c.pop_responses(2.0) == []
        # the socket path of this very conn still works
        c._send(b"C", b"key")  # reference framing: raw key bytes
        code, exist = struct.unpack("<ii", c._recv(8))
        assert code == 200 and exist in (0, 1)
    finally:
        c.close()
    conn = make_client(cpu_server)
    assert conn.check_exist("nope") in (0, False)
    conn.close()


def test_shm_misaligned_record_len(cpu_server):
    """A record len that is not 8-byte aligned is provably corrupt: the
    producer rounds every record to 8 bytes. The poller detaches."""
    c = RawShmConn(cpu_server)
    try:
        hdr = struct.pack("<IB3sIIQ", 44, ord("w"), b"\0\0\0", 20, 0, 3)
        c.push_raw(hdr + b"\0" * 20)  # publishes 44 bytes... 44 % 8 != 0
        assert c.pop_responses(2.0) == []
        c._send(b"C", b"key")  # reference framing: raw key bytes
        code, exist = struct.unpack("<ii", c._recv(8))
        assert code == 200 and exist in (0, 1)
    finally:
        c.close()


def test_shm_misaligned_ring_offsets_rejected(cpu_server):
    """req_off/resp_off that are not 8-aligned would put the atomics at
    misaligned addresses; attach() must reject the segment (handshake 500)."""
    c = RawShmConn(cpu_server, req_off_skew=4, expect_code=500)
    c.close()
