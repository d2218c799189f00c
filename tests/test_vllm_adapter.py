"""vLLM-shaped adapter (infinistore_amd/vllm_adapter.py) over the CPU
server: block-table mapping on both sides, prefix-hit fast path, partial
prefix save (skip_leading_pages) and eviction. The full model-level check
(decode logits from adapter-loaded KV == monolithic forward) lives in
tests/test_disaggregated.py, which now drives the demo through the adapter.
"""

import random
import uuid

import torch

from infinistore_amd.vllm_adapter import InfiniStoreKVAdapter

BT = 8          # tokens per block/page
ELEMS = 256     # elements per page per layer
LAYERS = 3


def make_adapter(port, tag=None):
    return InfiniStoreKVAdapter("127.0.0.1", port, tag or f"t-{uuid.uuid4().hex[:8]}",
                                LAYERS, BT, ELEMS, local=False)


def fill_caches(n_blocks, seed):
    g = torch.Generator().manual_seed(seed)
    return [torch.randn(n_blocks, ELEMS, generator=g) for _ in range(LAYERS)]


def test_block_table_mapping_roundtrip(cpu_server):
    """Prefill writes through one shuffled block table, decode reads through
    a different one; logical page contents must match exactly."""
    tag = f"t-{uuid.uuid4().hex[:8]}"
    tokens = list(range(100, 100 + 4 * BT))
    n_pages = 4

    pre = make_adapter(cpu_server, tag)
    try:
        pre_blocks = [5, 2, 7, 0]  # scattered physical placement
        caches = fill_caches(8, seed=1)
        for li in range(LAYERS):
            pre.save_kv_layer(li, caches[li], tokens, pre_blocks)
        pre.wait_for_save()
    finally:
        pre.close()

    dec = make_adapter(cpu_server, tag)
    try:
        assert dec.get_num_new_matched_tokens(tokens) == n_pages * BT
        dec_blocks = [1, 6, 3, 9]
        out = [torch.zeros(10, ELEMS) for _ in range(LAYERS)]
        assert dec.start_load_kv(out, tokens, dec_blocks) == n_pages
        for li in range(LAYERS):
            assert dec.wait_for_layer_load(li)
            for p in range(n_pages):
                assert torch.equal(out[li][dec_blocks[p]],
                                   caches[li][pre_blocks[p]]), (li, p)
    finally:
        dec.close()


def test_prefix_hit_partial(cpu_server):
    """Only a prefix is cached: matched tokens reflect the stored pages, and
    num_computed_tokens is subtracted."""
    tag = f"t-{uuid.uuid4().hex[:8]}"
    tokens = list(range(6 * BT))
    ad = make_adapter(cpu_server, tag)
    try:
        caches = fill_caches(6, seed=2)
        for li in range(LAYERS):
            # store only the first 2 pages: truncate the token list
            ad.save_kv_layer(li, caches[li], tokens[: 2 * BT], [0, 1])
        ad.wait_for_save()
        assert ad.get_num_new_matched_tokens(tokens) == 2 * BT
        assert ad.get_num_new_matched_tokens(tokens, num_computed_tokens=BT) == BT
        assert ad.get_num_new_matched_tokens(tokens, num_computed_tokens=3 * BT) == 0
        # a DIFFERENT prefix shares no pages (hash chain commits to prefix)
        other = [9999] + tokens[1:]
        assert ad.get_num_new_matched_tokens(other) == 0
    finally:
        ad.close()


def test_skip_leading_pages_and_dedup(cpu_server):
    """skip_leading_pages uploads only the new tail; the store's first-write-
    wins dedup keeps earlier pages intact."""
    tag = f"t-{uuid.uuid4().hex[:8]}"
    tokens = list(range(3 * BT))
    ad = make_adapter(cpu_server, tag)
    try:
        c1 = fill_caches(4, seed=3)
        for li in range(LAYERS):
            ad.save_kv_layer(li, c1[li], tokens[: 2 * BT], [0, 1])
        ad.wait_for_save()
        # second writer: full 3 pages but skips the 2 already cached
        c2 = fill_caches(4, seed=4)
        for li in range(LAYERS):
            ad.save_kv_layer(li, c2[li], tokens, [0, 1, 2],
                             skip_leading_pages=2)
        ad.wait_for_save()
        assert ad.get_num_new_matched_tokens(tokens) == 3 * BT
        out = [torch.zeros(4, ELEMS) for _ in range(LAYERS)]
        ad.start_load_kv(out, tokens, [0, 1, 2])
        for li in range(LAYERS):
            assert ad.wait_for_layer_load(li)
            assert torch.equal(out[li][0], c1[li][0])  # first write wins
            assert torch.equal(out[li][1], c1[li][1])
            assert torch.equal(out[li][2], c2[li][2])  # new tail from writer 2
    finally:
        ad.close()


def test_evict_request(cpu_server):
    tag = f"t-{uuid.uuid4().hex[:8]}"
    tokens = list(range(2 * BT))
    ad = make_adapter(cpu_server, tag)
    try:
        caches = fill_caches(2, seed=5)
        for li in range(LAYERS):
            ad.save_kv_layer(li, caches[li], tokens, [0, 1])
        ad.wait_for_save()
        assert ad.get_num_new_matched_tokens(tokens) == 2 * BT
        assert ad.evict_request(tokens) == 2 * LAYERS
        assert ad.get_num_new_matched_tokens(tokens) == 0
    finally:
        ad.close()


def test_load_into_short_block_table_raises(cpu_server):
    ad = make_adapter(cpu_server)
    try:
        tokens = list(range(4 * BT))
        out = [torch.zeros(2, ELEMS) for _ in range(LAYERS)]
        try:
            ad.start_load_kv(out, tokens, [0, 1])  # 4 pages, 2 entries
            raised = False
        except ValueError:
            raised = True
        assert raised
    finally:
        ad.close()
