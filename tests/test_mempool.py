"""Allocator unit tests (hierarchical bitmap pool)."""

from infinistore_amd import _native as n

MB = 1 << 20
KB = 1 << 10


def test_basic_alloc_free():
    p = n._TestPool(1 * MB, 4 * KB)
    assert p.total_blocks() == 256
    a = p.allocate(4 * KB)
    b = p.allocate(4 * KB)
    assert a != 0 and b != 0 and a != b
    assert p.used_blocks() == 2
    assert p.deallocate(a, 4 * KB)
    assert p.deallocate(b, 4 * KB)
    assert p.used_blocks() == 0


def test_multi_block_contiguous():
    p = n._TestPool(1 * MB, 4 * KB)
    a = p.allocate(64 * KB)  # 16 blocks contiguous
    assert a != 0
    assert p.used_blocks() == 16
    # must be block-aligned within the arena
    assert (a - p.base()) % (4 * KB) == 0
    assert p.deallocate(a, 64 * KB)
    assert p.used_blocks() == 0


def test_exhaustion_and_reuse():
    p = n._TestPool(64 * KB, 4 * KB)  # 16 blocks
    ptrs = [p.allocate(4 * KB) for _ in range(16)]
    assert all(x != 0 for x in ptrs)
    assert p.allocate(4 * KB) == 0  # full
    assert p.deallocate(ptrs[7], 4 * KB)
    c = p.allocate(4 * KB)
    assert c == ptrs[7]  # freed slot found again via summary
    assert p.allocate(4 * KB) == 0


def test_fragmented_run_search():
    p = n._TestPool(64 * KB, 4 * KB)  # 16 blocks
    ptrs = [p.allocate(4 * KB) for _ in range(16)]
    # free blocks 2,3 and 8,9,10 -> largest runs: 2 and 3
    for i in (2, 3, 8, 9, 10):
        assert p.deallocate(ptrs[i], 4 * KB)
    assert p.allocate(16 * KB) == 0  # needs 4 contiguous: none
    got = p.allocate(12 * KB)  # needs 3: fits at block 8
    assert got == ptrs[8]
    got2 = p.allocate(8 * KB)  # needs 2: fits at block 2
    assert got2 == ptrs[2]


def test_double_free_detected():
    p = n._TestPool(64 * KB, 4 * KB)
    a = p.allocate(8 * KB)
    assert p.deallocate(a, 8 * KB)
    assert not p.deallocate(a, 8 * KB)


def test_invalid_pointer_rejected():
    p = n._TestPool(64 * KB, 4 * KB)
    assert not p.deallocate(p.base() + 1, 4 * KB)  # unaligned
    assert not p.deallocate(p.base() + 10 * MB, 4 * KB)  # out of range


def test_cross_word_runs():
    # 128 blocks; allocate a 65-block run spanning a 64-bit bitmap word.
    p = n._TestPool(512 * KB, 4 * KB)
    assert p.total_blocks() == 128
    one = p.allocate(4 * KB)
    big = p.allocate(65 * 4 * KB)
    assert big != 0
    assert p.used_blocks() == 66
    assert p.deallocate(big, 65 * 4 * KB)
    assert p.deallocate(one, 4 * KB)


def test_many_allocs_stress():
    import random

    rng = random.Random(7)
    p = n._TestPool(4 * MB, 4 * KB)  # 1024 blocks
    live = {}
    for step in range(2000):
        if live and (rng.random() < 0.45 or p.used_blocks() > 900):
            ptr, size = live.popitem()[1]
            assert p.deallocate(ptr, size)
        else:
            nb = rng.choice([1, 1, 1, 2, 4])
            ptr = p.allocate(nb * 4 * KB)
            if ptr:
                live[len(live) + step * 10000] = (ptr, nb * 4 * KB)
    # all remaining valid; free them
    for ptr, size in live.values():
        assert p.deallocate(ptr, size)
    assert p.used_blocks() == 0


def test_single_block_churn_fast():
    """Steady-state KV churn: free/alloc single blocks at high rate on a
    fragmented pool. The O(1) free-list path must keep this cheap — the
    round-1 first-fit degraded to O(pool) per request under 64-client
    churn (25 ms/request measured)."""
    import time

    p = n._TestPool(64 * MB, 4 * KB)  # 16384 blocks
    # Fragment: fill completely, then free every other block.
    ptrs = [p.allocate(4 * KB) for _ in range(16384)]
    assert all(x != 0 for x in ptrs)
    for i in range(0, 16384, 2):
        assert p.deallocate(ptrs[i], 4 * KB)
    # Churn: 100k single-block alloc+free on the fragmented pool.
    t0 = time.time()
    live = []
    for i in range(100_000):
        a = p.allocate(4 * KB)
        assert a != 0
        live.append(a)
        if len(live) > 64:
            assert p.deallocate(live.pop(0), 4 * KB)
    for a in live:
        assert p.deallocate(a, 4 * KB)
    took = time.time() - t0
    assert took < 5.0, f"churn took {took:.1f}s - free-list path not engaged?"
    assert p.used_blocks() == 8192  # the odd-index half still allocated
    for i in range(1, 16384, 2):
        assert p.deallocate(ptrs[i], 4 * KB)
    assert p.used_blocks() == 0
