"""Wire-format tests: roundtrips through the clean-room flatbuffers core and
independent binary-layout checks with a tiny pure-Python flatbuffers reader
(so a C++ builder bug cannot hide behind a matching C++ parser bug)."""

import struct

from infinistore_amd import _native as n


# --- minimal independent flatbuffers reader (spec-derived) ------------------
def fb_root(buf):
    (off,) = struct.unpack_from("<I", buf, 0)
    return off


def fb_field(buf, table, field_id):
    """Return absolute position of field value, or None."""
    (soff,) = struct.unpack_from("<i", buf, table)
    vtable = table - soff
    vt_len, _tbl_len = struct.unpack_from("<HH", buf, vtable)
    slot = 4 + field_id * 2
    if slot + 2 > vt_len:
        return None
    (fo,) = struct.unpack_from("<H", buf, vtable + slot)
    if fo == 0:
        return None
    return table + fo


def fb_indirect(buf, pos):
    (rel,) = struct.unpack_from("<I", buf, pos)
    return pos + rel


def fb_string(buf, pos):
    (ln,) = struct.unpack_from("<I", buf, pos)
    return buf[pos + 4 : pos + 4 + ln].decode()


def fb_vec(buf, pos):
    (ln,) = struct.unpack_from("<I", buf, pos)
    return ln, pos + 4


def test_remote_meta_roundtrip():
    keys = ["a", "bb", "ccc", "x" * 100]
    addrs = [1, 2, 3, 2**63 + 5]
    b = n._dbg_build_remote_meta(keys, 131072, 42, addrs, 68)
    out = n._dbg_parse_remote_meta(b)
    assert out == (keys, 131072, 42, addrs, 68)


def test_remote_meta_binary_layout():
    """Parse the C++-built buffer with the independent Python reader."""
    keys = ["k1", "k2"]
    addrs = [0xAABB, 0xCCDD]
    buf = n._dbg_build_remote_meta(keys, 4096, 9, addrs, 65)
    root = fb_root(buf)
    # field 0: keys ([string])
    kp = fb_indirect(buf, fb_field(buf, root, 0))
    ln, data = fb_vec(buf, kp)
    assert ln == 2
    got = [fb_string(buf, fb_indirect(buf, data + 4 * i)) for i in range(ln)]
    assert got == keys
    # field 1: block_size (int32 inline)
    (bs,) = struct.unpack_from("<i", buf, fb_field(buf, root, 1))
    assert bs == 4096
    # field 2: rkey (uint32)
    (rk,) = struct.unpack_from("<I", buf, fb_field(buf, root, 2))
    assert rk == 9
    # field 3: remote_addrs ([uint64])
    ap = fb_indirect(buf, fb_field(buf, root, 3))
    ln, data = fb_vec(buf, ap)
    assert ln == 2
    # element alignment: uint64 vector data must be 8-aligned
    assert data % 8 == 0
    got = list(struct.unpack_from("<2Q", buf, data))
    assert got == addrs
    # field 4: op (int8)
    (op,) = struct.unpack_from("<b", buf, fb_field(buf, root, 4))
    assert op == 65


def test_remote_meta_defaults_absent():
    """Scalar fields equal to their default must be omitted from the table."""
    buf = n._dbg_build_remote_meta([], 0, 0, [], 0)
    root = fb_root(buf)
    assert fb_field(buf, root, 1) is None  # block_size == default 0
    assert fb_field(buf, root, 2) is None
    assert fb_field(buf, root, 4) is None
    out = n._dbg_parse_remote_meta(buf)
    assert out == ([], 0, 0, [], 0)


def test_local_meta_roundtrip():
    ipc = bytes(range(64))
    blocks = [("key-%d" % i, i * 4096) for i in range(10)]
    b = n._dbg_build_local_meta(7, ipc, 65536, blocks, 512)
    dev, h, bs, blks, boff = n._dbg_parse_local_meta(b)
    assert dev == 7 and h == ipc and bs == 65536 and boff == 512
    assert [tuple(x) for x in blks] == blocks


def test_local_meta_block_subtables_layout():
    ipc = b"\x01" * 64
    buf = n._dbg_build_local_meta(1, ipc, 128, [("abc", 999)], 0)
    root = fb_root(buf)
    # field 1: ipc_handle [ubyte]
    hp = fb_indirect(buf, fb_field(buf, root, 1))
    ln, data = fb_vec(buf, hp)
    assert ln == 64 and buf[data : data + 64] == ipc
    # field 3: blocks [Block]
    bp = fb_indirect(buf, fb_field(buf, root, 3))
    ln, data = fb_vec(buf, bp)
    assert ln == 1
    bt = fb_indirect(buf, data)
    assert fb_string(buf, fb_indirect(buf, fb_field(buf, bt, 0))) == "abc"
    (off,) = struct.unpack_from("<Q", buf, fb_field(buf, bt, 1))
    assert off == 999


def test_alloc_resp_struct_vector_layout():
    blocks = [(3, 0x1122334455667788), (0, 0)]
    buf = n._dbg_build_alloc_resp(blocks)
    assert n._dbg_parse_alloc_resp(buf) == blocks
    root = fb_root(buf)
    vp = fb_indirect(buf, fb_field(buf, root, 0))
    ln, data = fb_vec(buf, vp)
    assert ln == 2
    # RemoteBlock struct: 16 bytes inline, {u32 rkey, pad, u64 addr}
    assert data % 8 == 0
    rkey, addr = struct.unpack_from("<I4xQ", buf, data)
    assert rkey == 3 and addr == 0x1122334455667788
    rkey2, addr2 = struct.unpack_from("<I4xQ", buf, data + 16)
    assert (rkey2, addr2) == (0, 0)


def test_match_request_roundtrip():
    keys = ["h%d" % i for i in range(100)]
    buf = n._dbg_build_match_req(keys)
    assert n._dbg_parse_match_req(buf) == keys


def test_empty_and_unicode_keys():
    keys = ["", "日本語キー", "k"]
    buf = n._dbg_build_match_req(keys)
    assert n._dbg_parse_match_req(buf) == keys
