"""Prefill-integration pattern: overlap per-layer KV-cache uploads with
compute — the documented way an inference engine streams KV pages into the
store layer by layer during prefill, bounding client memory for long
sequences (role of /root/reference/infinistore/example/demo_prefill.py and
docs/source/design.rst:54-63).

A toy transformer runs layer by layer on the GPU; after each layer's
"attention" produces its KV pages, the pages are handed to an uploader
thread which issues `local_gpu_write_cache` for that layer while the next
layer computes. HIP events order the upload after the producing kernel.
"""

import queue
import threading
import uuid

import torch

import infinistore_amd as ifs

N_LAYERS = 14
PAGE_ELEMS = 16384  # 32 KB fp16 pages
PAGES_PER_LAYER = 16


def main(port: int = 22345):
    assert torch.cuda.is_available(), "demo needs a GPU"
    cfg = ifs.ClientConfig(
        host_addr="127.0.0.1", service_port=port,
        connection_type=ifs.TYPE_LOCAL_GPU,
    )
    conn = ifs.InfinityConnection(cfg)
    conn.connect()

    run_id = uuid.uuid4().hex
    kv = torch.zeros(N_LAYERS * PAGES_PER_LAYER * PAGE_ELEMS,
                     dtype=torch.float16, device="cuda:0")
    x = torch.randn(4096, 4096, device="cuda:0", dtype=torch.float16)
    w = torch.randn(4096, 4096, device="cuda:0", dtype=torch.float16)

    upload_q: "queue.Queue" = queue.Queue()
    done = threading.Event()

    def uploader():
        while True:
            item = upload_q.get()
            if item is None:
                break
            layer, ev = item
            ev.synchronize()  # wait for the producing layer's kernels
            base = layer * PAGES_PER_LAYER * PAGE_ELEMS
            blocks = [
                (f"{run_id}-L{layer}-p{p}", base + p * PAGE_ELEMS)
                for p in range(PAGES_PER_LAYER)
            ]
            conn.local_gpu_write_cache(kv, blocks, PAGE_ELEMS)
        conn.sync()
        done.set()

    t = threading.Thread(target=uploader)
    t.start()

    for layer in range(N_LAYERS):
        # "attention": produce this layer's KV pages
        x = x @ w
        s = layer * PAGES_PER_LAYER * PAGE_ELEMS
        kv[s : s + PAGES_PER_LAYER * PAGE_ELEMS] = x.flatten()[
            : PAGES_PER_LAYER * PAGE_ELEMS
        ]
        ev = torch.cuda.Event()
        ev.record()
        upload_q.put((layer, ev))  # upload overlaps the next layer's compute

    upload_q.put(None)
    t.join()
    done.wait()

    # Decode side: prefix-match then read back one layer.
    all_keys = [f"{run_id}-L{l}-p{p}" for l in range(N_LAYERS)
                for p in range(PAGES_PER_LAYER)]
    match = conn.get_match_last_index(all_keys)
    print(f"prefix match: {match + 1}/{len(all_keys)} pages cached")

    out = torch.zeros(PAGES_PER_LAYER * PAGE_ELEMS, dtype=torch.float16,
                      device="cuda:0")
    blocks = [(f"{run_id}-L0-p{p}", p * PAGE_ELEMS) for p in range(PAGES_PER_LAYER)]
    conn.read_cache(out, blocks, PAGE_ELEMS)
    conn.sync()
    assert torch.equal(out, kv[: PAGES_PER_LAYER * PAGE_ELEMS])
    print("layer-0 readback verified")
    conn.close()


if __name__ == "__main__":
    import sys

    main(int(sys.argv[1]) if len(sys.argv) > 1 else 22345)
