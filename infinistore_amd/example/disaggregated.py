"""End-to-end disaggregated inference demo: prefill → store → decode.

This is the workload the store exists for (reference docs/source/design.rst:
54-63 describe the same split): a PREFILL worker runs the prompt through a
model and streams each layer's paged KV into infinistore; a DECODE worker —
a different process with no access to the prefill activations — looks up
how many leading pages of the prompt are cached, gathers the KV pages
straight into its own cache tensors, and generates the next token. The
demo verifies the decode-side logits against a monolithic forward pass of
the same model, so a cache round trip that corrupted or misordered a
single page would fail loudly.

Runs on CPU (TCP fabric path) or GPU (local IPC + HIP gather path):
    python -m infinistore_amd.example.disaggregated [port]
"""

import random
import sys
import uuid

import torch

from infinistore_amd.vllm_adapter import InfiniStoreKVAdapter

PAGE_TOKENS = 16


class TinyLlama(torch.nn.Module):
    """A minimal GQA transformer with an explicit paged KV cache — just
    enough structure (RMSNorm, rotary-free attention, SwiGLU) to make KV
    correctness meaningful."""

    def __init__(self, vocab=1024, dim=256, n_layers=2, n_heads=8, n_kv=4):
        super().__init__()
        self.dim, self.n_heads, self.n_kv = dim, n_heads, n_kv
        self.hd = dim // n_heads
        self.n_layers = n_layers
        self.emb = torch.nn.Embedding(vocab, dim)
        self.layers = torch.nn.ModuleList()
        for _ in range(n_layers):
            blk = torch.nn.ModuleDict(dict(
                wq=torch.nn.Linear(dim, n_heads * self.hd, bias=False),
                wk=torch.nn.Linear(dim, n_kv * self.hd, bias=False),
                wv=torch.nn.Linear(dim, n_kv * self.hd, bias=False),
                wo=torch.nn.Linear(n_heads * self.hd, dim, bias=False),
                w1=torch.nn.Linear(dim, 4 * dim, bias=False),
                w2=torch.nn.Linear(4 * dim, dim, bias=False),
                ln1=torch.nn.LayerNorm(dim),
                ln2=torch.nn.LayerNorm(dim),
            ))
            self.layers.append(blk)
        self.out = torch.nn.Linear(dim, vocab, bias=False)

    def kv_elems_per_page(self):
        # One page holds K and V for PAGE_TOKENS tokens: [2, T, n_kv, hd].
        return 2 * PAGE_TOKENS * self.n_kv * self.hd

    def forward_collect(self, tokens):
        """Full forward over `tokens`; returns (logits_last, per-layer KV
        tensors shaped [2, seq, n_kv, hd])."""
        x = self.emb(tokens)
        kvs = []
        for blk in self.layers:
            h = blk["ln1"](x)
            s = h.shape[0]
            q = blk["wq"](h).view(s, self.n_heads, self.hd)
            k = blk["wk"](h).view(s, self.n_kv, self.hd)
            v = blk["wv"](h).view(s, self.n_kv, self.hd)
            kvs.append(torch.stack([k, v]))
            x = x + self._attend(blk, q, k, v)
            x = x + blk["w2"](torch.nn.functional.silu(blk["w1"](blk["ln2"](x))))
        return self.out(x[-1]), kvs

    def forward_one(self, token, pos, kv_cache):
        """Decode one token at position `pos` against per-layer caches
        shaped [2, cap, n_kv, hd] whose [ :, :pos] entries are valid."""
        x = self.emb(token.view(1))
        for li, blk in enumerate(self.layers):
            h = blk["ln1"](x)
            q = blk["wq"](h).view(1, self.n_heads, self.hd)
            k = blk["wk"](h).view(1, self.n_kv, self.hd)
            v = blk["wv"](h).view(1, self.n_kv, self.hd)
            kv_cache[li][0, pos] = k[0]
            kv_cache[li][1, pos] = v[0]
            keys = kv_cache[li][0, : pos + 1]
            vals = kv_cache[li][1, : pos + 1]
            x = x + self._attend(blk, q, keys, vals, causal=False)
            x = x + blk["w2"](torch.nn.functional.silu(blk["w1"](blk["ln2"](x))))
        return self.out(x[-1])

    def _attend(self, blk, q, k, v, causal=True):
        rep = self.n_heads // self.n_kv
        kq = k.repeat_interleave(rep, dim=1)
        vq = v.repeat_interleave(rep, dim=1)
        o = torch.nn.functional.scaled_dot_product_attention(
            q.transpose(0, 1), kq.transpose(0, 1), vq.transpose(0, 1),
            is_causal=causal)
        return blk["wo"](o.transpose(0, 1).reshape(q.shape[0], -1))


def prefill_worker(model, tokens, adapter: InfiniStoreKVAdapter, device):
    """Run the prompt, place the paged KV into a vLLM-style block-major
    physical cache with an arbitrary block table, and stream it into the
    store through the adapter's per-layer save hook (uploads overlap later
    layers' compute via async writes)."""
    n_pages = len(tokens) // PAGE_TOKENS
    t = torch.tensor(tokens, device=device)
    with torch.no_grad():
        logits, kvs = model.forward_collect(t)
    dt = next(model.parameters()).dtype
    # Physical cache: more blocks than pages, pages scattered through it by
    # a shuffled block table (exactly what a paged-attention allocator does).
    n_blocks = n_pages + 3
    block_table = list(range(n_blocks))
    random.Random(11).shuffle(block_table)
    block_table = block_table[:n_pages]
    for li, kv in enumerate(kvs):
        cache = torch.zeros(n_blocks, 2, PAGE_TOKENS, model.n_kv, model.hd,
                            device=device, dtype=dt)
        paged = kv[:, : n_pages * PAGE_TOKENS].unflatten(
            1, (n_pages, PAGE_TOKENS)).transpose(0, 1)  # [p, 2, T, kv, hd]
        for p in range(n_pages):
            cache[block_table[p]] = paged[p]
        adapter.save_kv_layer(li, cache, tokens, block_table)
    adapter.wait_for_save()


def decode_worker(model, tokens, next_token, adapter: InfiniStoreKVAdapter,
                  device):
    """Reconstruct the KV cache from the store through the adapter's load
    hooks — into this worker's OWN physical cache with its OWN (different)
    block table — and decode one token. Returns (logits, matched_tokens)."""
    matched = adapter.get_num_new_matched_tokens(tokens)
    n_pages = len(tokens) // PAGE_TOKENS
    assert matched >= n_pages * PAGE_TOKENS, (
        f"prefix lookup matched {matched} tokens, need {n_pages * PAGE_TOKENS}")
    dt = next(model.parameters()).dtype
    # Decode engine's physical placement differs from prefill's on purpose.
    n_blocks = n_pages + 5
    block_table = list(range(n_blocks))
    random.Random(23).shuffle(block_table)
    block_table = block_table[:n_pages]
    caches = [torch.zeros(n_blocks, 2, PAGE_TOKENS, model.n_kv, model.hd,
                          device=device, dtype=dt)
              for _ in range(model.n_layers)]
    got = adapter.start_load_kv(caches, tokens, block_table)
    assert got == n_pages, f"loading {got}/{n_pages} pages"
    cap = len(tokens) + 8
    kv_cache = [torch.zeros(2, cap, model.n_kv, model.hd, device=device,
                            dtype=dt)
                for _ in range(model.n_layers)]
    for li in range(model.n_layers):
        assert adapter.wait_for_layer_load(li), f"layer {li}: load failed"
        # Unpack physical blocks -> linear [2, seq] layout for this toy model
        # (a real paged-attention kernel would read the blocks directly).
        for p in range(n_pages):
            page = caches[li][block_table[p]]  # [2, T, kv, hd]
            kv_cache[li][:, p * PAGE_TOKENS:(p + 1) * PAGE_TOKENS] = page
    with torch.no_grad():
        logits = model.forward_one(
            torch.tensor(next_token, device=device), len(tokens), kv_cache)
    return logits, matched


def main(port=22345, device=None, seed=7, quant=None):
    device = device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    torch.manual_seed(seed)
    model = TinyLlama().to(device)
    model.eval()
    if quant:  # fp8 page compression needs bf16 KV pages
        model = model.to(torch.bfloat16)
    tokens = torch.randint(0, 1024, (4 * PAGE_TOKENS,)).tolist()
    next_token = 17
    tag = f"demo-{uuid.uuid4().hex[:8]}"

    local = device.startswith("cuda")
    elems = model.kv_elems_per_page()
    pre = InfiniStoreKVAdapter("127.0.0.1", port, tag, model.n_layers,
                               PAGE_TOKENS, elems, local=local, quant=quant)
    try:
        prefill_worker(model, tokens, pre, device)
    finally:
        pre.close()

    dec = InfiniStoreKVAdapter("127.0.0.1", port, tag, model.n_layers,
                               PAGE_TOKENS, elems, local=local)
    try:
        logits, hits = decode_worker(model, tokens, next_token, dec, device)
    finally:
        dec.close()

    # Ground truth: one monolithic forward over prompt + next token.
    with torch.no_grad():
        ref_logits, _ = model.forward_collect(
            torch.tensor(tokens + [next_token], device=device))
    # fp8-compressed KV carries ~3 mantissa bits; logits move accordingly.
    atol = 0.25 if quant else (2e-2 if next(model.parameters()).dtype
                               == torch.bfloat16 else 1e-4)
    assert torch.allclose(logits.float(), ref_logits.float(), atol=atol), (
        (logits - ref_logits).abs().max().item())
    print(f"disaggregated decode ok ({'fp8' if quant else 'plain'} pages): "
          f"{hits} cached tokens reused via the vLLM-shaped adapter, logits "
          f"match monolithic forward "
          f"(max diff {(logits - ref_logits).abs().max().item():.2e})")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 22345)
