"""Paged-KV connector: the integration layer an inference engine (vLLM-style
disaggregated prefill/decode) uses to stream per-layer KV pages into the
store and pull prefix hits back out.

This is the pattern the reference documents through its examples
(/root/reference/infinistore/example/demo_prefill.py, docs/source/design.rst:
54-63) packaged as a reusable class: page keys are a hash chain over token
pages (so `get_match_last_index` answers "how many leading pages of this
sequence are cached"), writes stream layer-by-layer during prefill, reads
gather straight into the decode engine's KV tensors.

Framework-agnostic: anything that exposes per-layer KV page tensors can use
it (duck-typed; vLLM not required).
"""

import hashlib
from typing import List, Optional, Sequence

import torch

from . import lib


def token_page_hashes(token_ids: Sequence[int], page_size: int,
                      model_tag: str) -> List[str]:
    """Hash-chain page keys: key_i = H(model_tag, key_{i-1}, tokens_of_page_i).
    A page's key therefore commits to the whole prefix, which is what makes
    key-presence monotone (the property get_match_last_index relies on)."""
    keys = []
    prev = model_tag.encode()
    for start in range(0, len(token_ids) - len(token_ids) % page_size, page_size):
        page = token_ids[start : start + page_size]
        h = hashlib.blake2b(digest_size=16)
        h.update(prev)
        h.update(bytes(str(list(page)), "utf-8"))
        digest = h.hexdigest()
        keys.append(digest)
        prev = digest.encode()
    return keys


class PagedKVConnector:
    """Streams paged KV between an engine and an infinistore-amd server.

    Args:
        host, port: server address.
        model_tag: disambiguates models sharing one store.
        n_layers: layer count (layer index becomes part of the key).
        local: use the local-GPU IPC path (same host) vs the fabric path.
    """

    def __init__(self, host: str, port: int, model_tag: str, n_layers: int,
                 local: bool = True, quant: Optional[str] = None):
        """quant="fp8": store KV pages fp8-compressed (half HBM per page,
        ~3-bit mantissa; reads return bf16). Local path + bf16 KV only."""
        self.model_tag = model_tag
        self.n_layers = n_layers
        self.quant = quant
        cfg = lib.ClientConfig(
            host_addr=host,
            service_port=port,
            connection_type=lib.TYPE_LOCAL_GPU if local else lib.TYPE_RDMA,
            link_type="Ethernet" if not local else "Ethernet",
        )
        self.conn = lib.InfinityConnection(cfg)
        self.conn.connect()
        self.local = local
        self._registered = set()

    def close(self):
        self.conn.close()

    # -- lookup ---------------------------------------------------------------
    def cached_pages(self, page_keys: List[str]) -> int:
        """How many leading pages of this sequence are fully cached (all
        layers present). Probes layer keys of the LAST layer written per page
        (layers are written 0..n-1, so the last layer's presence implies the
        rest on the prefill side)."""
        probe = [self._key(self.n_layers - 1, k) for k in page_keys]
        try:
            return self.conn.get_match_last_index(probe) + 1
        except Exception:
            return 0

    # -- prefill side ---------------------------------------------------------
    def save_layer(self, layer: int, kv: torch.Tensor, page_keys: List[str],
                   page_offsets, page_elems: int):
        """Store this layer's pages of `kv` (offsets in elements). Call per
        layer as prefill produces them; uploads overlap later layers'
        compute (writes are async until sync())."""
        keys = [self._key(layer, k) for k in page_keys]
        if self.local:
            self.conn.write_pages(kv, keys, page_offsets, page_elems,
                                  quant=self.quant)
        else:
            self._ensure_mr(kv)
            es = kv.element_size()
            blocks = self.conn.allocate_rdma(keys, page_elems * es)
            self.conn.rdma_write_cache(kv, list(page_offsets), page_elems, blocks)

    def flush(self):
        self.conn.sync()

    # -- decode side ----------------------------------------------------------
    def load_layer(self, layer: int, kv_out: torch.Tensor, page_keys: List[str],
                   page_offsets, page_elems: int) -> bool:
        """Gather this layer's cached pages into the engine's KV tensor.
        Returns False if any page is missing (caller falls back to compute)."""
        keys = [self._key(layer, k) for k in page_keys]
        try:
            if self.local:
                self.conn.read_pages(kv_out, keys, page_offsets, page_elems)
            else:
                self._ensure_mr(kv_out)
                blocks = [(k, int(o)) for k, o in zip(keys, page_offsets)]
                self.conn.read_cache(kv_out, blocks, page_elems)
            self.conn.sync()
            return True
        except Exception:
            return False

    def load_layer_async(self, layer: int, kv_out: torch.Tensor,
                         page_keys: List[str], page_offsets, page_elems: int):
        """Ticketed prefetch (local path): push the gather and return a
        ticket; call wait_load(ticket) before touching kv_out. Lets decode
        overlap the next request's KV fetch with current compute. Returns
        None if any page was missing or the path is unavailable (caller
        falls back to load_layer / recompute)."""
        if not self.local:
            return None
        keys = [self._key(layer, k) for k in page_keys]
        try:
            return self.conn.read_pages_async(kv_out, keys, page_offsets,
                                              page_elems)
        except Exception:
            return None

    def wait_load(self, ticket) -> bool:
        try:
            self.conn.wait_read(ticket)
            return True
        except Exception:
            return False

    def evict(self, page_keys: List[str]) -> int:
        """Drop a sequence's pages (all layers)."""
        keys = [self._key(layer, k) for layer in range(self.n_layers)
                for k in page_keys]
        return self.conn.delete_keys(keys)

    # -- internals ------------------------------------------------------------
    def _key(self, layer: int, page_hash: str) -> str:
        return f"{self.model_tag}/L{layer}/{page_hash}"

    def _ensure_mr(self, t: torch.Tensor):
        ptr = t.data_ptr()
        if ptr not in self._registered:
            self.conn.register_mr(t)
            self._registered.add(ptr)


def gpu_page_hashes(kv: torch.Tensor, page_offsets, page_elems: int,
                    prev_digest: Optional[int] = None) -> List[int]:
    """GPU-side alternative to token_page_hashes: fingerprint the KV pages
    themselves with the fingerprint kernel (no host readback). Useful for
    content-addressed dedup rather than token-prefix reuse."""
    fps = lib.fingerprint_blocks(kv, list(page_offsets), page_elems)
    out = []
    prev = prev_digest or 0
    for f in fps:
        prev = (prev * 1099511628211 + f) % (1 << 64)
        out.append(prev)
    return out
