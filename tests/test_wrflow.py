"""WR flow-control unit tests (chain split, outstanding cap, overflow drain)."""

from infinistore_amd import _native as n


def test_chain_split_and_imm():
    sizes, peak, outstanding, parked_peak, parked_end = n._dbg_wrflow_sim(100, 32, 4096, 0)
    assert sizes == [32, 32, 32, 4]  # last chain carries the immediate
    assert outstanding == 0 and parked_end == 0
    assert peak <= 4096


def test_outstanding_cap_parks_chains():
    # 300 WRs, batch 32, cap 64: only 2 chains fit at once.
    sizes, peak, outstanding, parked_peak, parked_end = n._dbg_wrflow_sim(300, 32, 64, 0)
    assert sum(sizes) == 300
    assert peak <= 64
    assert parked_peak > 0          # overflow queue engaged
    assert parked_end == 0          # fully drained by completions
    assert outstanding == 0


def test_single_wr():
    sizes, peak, outstanding, parked_peak, parked_end = n._dbg_wrflow_sim(1, 32, 4096, 0)
    assert sizes == [1]
    assert outstanding == 0


def test_zero_wr_imm_only():
    sizes, peak, outstanding, parked_peak, parked_end = n._dbg_wrflow_sim(0, 32, 4096, 0)
    assert sizes == [0]  # bare-IMM chain still posted
