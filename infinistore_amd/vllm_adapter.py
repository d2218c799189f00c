"""vLLM-shaped KV-cache connector adapter.

Presents the interface shape of vLLM v1's KV connector (scheduler-side
prefix-hit query + worker-side per-layer save/load hooks with async waits)
on top of :class:`PagedKVConnector`, so an engine that manages paged KV
caches through block tables can plug the store in without writing any
store-specific code. vLLM itself is NOT imported — the adapter is
duck-typed against the engine objects it needs (per-layer cache tensors +
block tables), which is also what makes it testable here.

Engine model assumed (vLLM-style paged attention):

* per layer, one physical cache tensor laid out block-major:
  ``cache[layer].view(num_blocks, -1)`` — physical block ``b`` of layer
  ``l`` occupies elements ``[b*page_elems, (b+1)*page_elems)``. The usual
  ``[num_blocks, 2, block_tokens, n_kv, head]`` layout satisfies this.
* a request's logical page ``p`` lives in physical block
  ``block_table[p]`` (an arbitrary permutation — the adapter maps through
  it on both save and load, so prefill and decode engines can use
  completely different physical placements).
* page keys are the token-prefix hash chain (``token_page_hashes``), so a
  page key commits to the whole prefix and `get_match_last_index` answers
  "how many leading pages are cached" in one round trip.

Reference behavior being packaged: the layer-by-layer streaming pattern of
/root/reference/infinistore/example/demo_prefill.py:58-90 and the prefix
lookup of /root/reference/src/infinistore.cpp:1092-1108.
"""

from typing import Dict, List, Optional, Sequence

import torch

from .kv_connector import PagedKVConnector, token_page_hashes


class InfiniStoreKVAdapter:
    """vLLM-v1-shaped connector over a PagedKVConnector.

    Scheduler side:
        ``get_num_new_matched_tokens(token_ids, num_computed_tokens)``
    Worker side (prefill):
        ``save_kv_layer(layer, kv_cache, token_ids, block_table)`` per
        layer, then ``wait_for_save()``.
    Worker side (decode):
        ``start_load_kv(kv_caches, token_ids, block_table)`` once, then
        ``wait_for_layer_load(layer)`` right before each layer's attention.

    Args:
        host, port: store address.
        model_tag: namespace (model + dtype + layout version).
        n_layers: engine layer count.
        block_tokens: tokens per engine block (= store page).
        page_elems: elements per page per layer (``2*block_tokens*n_kv*hd``
            for the usual K+V layout).
        local: same-host GPU path (IPC + HIP gather) vs the fabric path.
        quant: "fp8" to store pages compressed (bf16 engines only).
    """

    def __init__(self, host: str, port: int, model_tag: str, n_layers: int,
                 block_tokens: int, page_elems: int, local: bool = True,
                 quant: Optional[str] = None):
        self.block_tokens = block_tokens
        self.page_elems = page_elems
        self.n_layers = n_layers
        self._conn = PagedKVConnector(host, port, model_tag, n_layers,
                                      local=local, quant=quant)
        self._load_tickets: Dict[int, object] = {}
        self._load_fallback: Dict[int, tuple] = {}
        self._key_cache: Dict[tuple, List[str]] = {}

    def close(self):
        self._conn.close()

    # -- scheduler side -------------------------------------------------------
    def get_num_new_matched_tokens(self, token_ids: Sequence[int],
                                   num_computed_tokens: int = 0) -> int:
        """Prefix-hit fast path: how many tokens beyond
        ``num_computed_tokens`` can be served from the store (whole pages
        only). One `get_match_last_index` round trip."""
        keys = self._page_keys(token_ids)
        if not keys:
            return 0
        hit_pages = self._conn.cached_pages(keys)
        hit_tokens = hit_pages * self.block_tokens
        return max(0, hit_tokens - num_computed_tokens)

    # -- worker side: prefill save -------------------------------------------
    def save_kv_layer(self, layer: int, kv_cache: torch.Tensor,
                      token_ids: Sequence[int],
                      block_table: Sequence[int],
                      skip_leading_pages: int = 0) -> None:
        """Stream layer ``layer``'s full pages of this request into the
        store (async; overlaps later layers' compute). ``kv_cache`` is the
        engine's physical cache tensor for this layer (block-major);
        ``block_table[p]`` names the physical block of logical page ``p``.
        ``skip_leading_pages`` skips pages already known cached (prefix
        hits reported by the scheduler)."""
        keys = self._page_keys(token_ids)
        n_pages = len(keys)
        if n_pages > len(block_table):
            raise ValueError(f"block_table has {len(block_table)} entries for "
                             f"{n_pages} full pages")
        s = skip_leading_pages
        if s >= n_pages:
            return
        offsets = [int(block_table[p]) * self.page_elems
                   for p in range(s, n_pages)]
        self._conn.save_layer(layer, kv_cache.view(-1), keys[s:], offsets,
                              self.page_elems)

    def wait_for_save(self):
        """Barrier: all save_kv_layer uploads committed (readable by any
        client, including other hosts)."""
        self._conn.flush()

    # -- worker side: decode load ----------------------------------------------
    def start_load_kv(self, kv_caches: List[torch.Tensor],
                      token_ids: Sequence[int],
                      block_table: Sequence[int],
                      n_pages: Optional[int] = None) -> int:
        """Kick off the gather of every layer's cached pages straight into
        the engine's cache tensors (ticketed async reads on the local path;
        all layers' gathers are in flight at once). Returns the number of
        pages being loaded. Call ``wait_for_layer_load(l)`` before layer
        ``l``'s attention reads its pages."""
        keys = self._page_keys(token_ids)
        if n_pages is not None:
            keys = keys[:n_pages]
        if not keys:
            return 0
        if len(keys) > len(block_table):
            raise ValueError(f"block_table has {len(block_table)} entries for "
                             f"{len(keys)} pages")
        offsets = [int(block_table[p]) * self.page_elems
                   for p in range(len(keys))]
        self._load_tickets.clear()
        self._load_fallback.clear()
        for li in range(self.n_layers):
            tk = self._conn.load_layer_async(li, kv_caches[li].view(-1), keys,
                                             offsets, self.page_elems)
            if tk is not None:
                self._load_tickets[li] = tk
            else:  # fabric path (or ring unavailable): blocking load at wait
                self._load_fallback[li] = (kv_caches[li], keys, offsets)
        return len(keys)

    def wait_for_layer_load(self, layer: int) -> bool:
        """Block until layer ``layer``'s pages have landed. False = a page
        was missing or the read failed (caller recomputes that layer)."""
        tk = self._load_tickets.pop(layer, None)
        if tk is not None:
            return self._conn.wait_load(tk)
        fb = self._load_fallback.pop(layer, None)
        if fb is not None:
            kv, keys, offsets = fb
            return self._conn.load_layer(layer, kv.view(-1), keys, offsets,
                                         self.page_elems)
        return True  # nothing pending for this layer (e.g. 0 pages)

    # -- maintenance -----------------------------------------------------------
    def evict_request(self, token_ids: Sequence[int]) -> int:
        """Drop all layers of this sequence's pages from the store."""
        return self._conn.evict(self._page_keys(token_ids))

    # -- internals -------------------------------------------------------------
    def _page_keys(self, token_ids: Sequence[int]) -> List[str]:
        ck = tuple(token_ids)
        got = self._key_cache.get(ck)
        if got is not None:
            return got
        keys = token_page_hashes(list(token_ids), self.block_tokens,
                                 self._conn.model_tag)
        if len(self._key_cache) < 256:
            self._key_cache[ck] = keys
        return keys
