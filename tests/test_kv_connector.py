"""PagedKVConnector (engine integration layer) end-to-end on the CPU fabric."""

import numpy as np
import torch

from infinistore_amd.kv_connector import PagedKVConnector, token_page_hashes


def test_hash_chain_properties():
    toks = list(range(100))
    a = token_page_hashes(toks, 16, "m1")
    assert len(a) == 6  # 96 tokens -> 6 full pages
    b = token_page_hashes(toks, 16, "m1")
    assert a == b  # deterministic
    c = token_page_hashes(toks, 16, "m2")
    assert a != c  # model tag matters
    # prefix property: a longer sequence shares the leading keys
    d = token_page_hashes(toks + [1, 2, 3] * 6, 16, "m1")
    assert d[:6] == a
    # divergence in an early page changes every later key
    toks2 = [999] + toks[1:]
    e = token_page_hashes(toks2, 16, "m1")
    assert e[0] != a[0] and e[5] != a[5]


def test_connector_roundtrip(cpu_server):
    n_layers = 4
    page_elems = 256
    pages = 8
    conn = PagedKVConnector("127.0.0.1", cpu_server, "llama-test", n_layers,
                            local=False)
    try:
        toks = list(range(pages * 16))
        keys = token_page_hashes(toks, 16, "llama-test")
        assert conn.cached_pages(keys) == 0
        offs = np.arange(pages, dtype=np.uint64) * page_elems
        layers = [torch.randn(pages * page_elems) for _ in range(n_layers)]
        for li, kv in enumerate(layers):
            conn.save_layer(li, kv, keys, offs, page_elems)
        conn.flush()
        assert conn.cached_pages(keys) == pages
        # decode side: load back each layer
        for li in range(n_layers):
            out = torch.zeros(pages * page_elems)
            assert conn.load_layer(li, out, keys, offs, page_elems)
            assert torch.equal(out, layers[li])
        # partial prefix: extend the sequence; only the stored prefix hits
        keys_ext = token_page_hashes(toks + list(range(64)), 16, "llama-test")
        assert conn.cached_pages(keys_ext) == pages
        # eviction drops everything
        assert conn.evict(keys) == pages * n_layers
        assert conn.cached_pages(keys) == 0
    finally:
        conn.close()
