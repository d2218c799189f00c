#!/usr/bin/env python3
"""Sustained mixed-workload soak: concurrent writers/readers/deleters plus
periodic compaction against one server, watching for errors, leaks, and
throughput collapse. Run on a GPU box:

    python scripts/soak.py --seconds 60 --threads 6
"""

import argparse
import json
import random
import sys
import threading
import time
import uuid

import numpy as np
import torch

sys.path.insert(0, ".")
import infinistore_amd as ifs  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=int, default=60)
    p.add_argument("--threads", type=int, default=6)
    p.add_argument("--port", type=int, default=23751)
    p.add_argument("--pool-gb", type=int, default=6)
    p.add_argument("--block-kb", type=int, default=128)
    p.add_argument("--quant-frac", type=float, default=0.0,
                   help="fraction of write generations stored fp8-compressed "
                        "(GPU path): exercises mixed plain/fp8 entries under "
                        "eviction + compaction churn")
    args = p.parse_args()

    have_gpu = torch.cuda.is_available()
    ifs.register_server(ifs.ServerConfig(
        service_port=args.port, manage_port=args.port + 1,
        prealloc_size=args.pool_gb, minimal_allocate_size=args.block_kb,
        cpu_only=not have_gpu, auto_evict=True,
    ))

    stop = time.time() + args.seconds
    errors = []
    ops = [0] * args.threads
    bytes_moved = [0] * args.threads

    page_elems = args.block_kb * 1024 // 2
    nb = 256

    def worker(tid):
        rng = random.Random(tid)
        try:
            cfg = ifs.ClientConfig(
                host_addr="127.0.0.1", service_port=args.port,
                connection_type=ifs.TYPE_LOCAL_GPU if have_gpu else ifs.TYPE_RDMA,
                link_type="TCP",
            )
            conn = ifs.InfinityConnection(cfg)
            conn.connect()
            dev = "cuda:0" if have_gpu else "cpu"
            src = torch.randn(nb * page_elems, dtype=torch.bfloat16, device=dev)
            dst = torch.zeros_like(src)
            if not have_gpu:
                conn.register_mr(src)
                conn.register_mr(dst)
            offs = np.arange(nb, dtype=np.uint64) * page_elems
            live = []
            gen = 0
            while time.time() < stop:
                gen += 1
                keys = [f"t{tid}-g{gen}-{uuid.uuid4().hex[:8]}-{i}" for i in range(nb)]
                q = "fp8" if (have_gpu and rng.random() < args.quant_frac) else None
                if have_gpu:
                    conn.write_pages(src, keys, offs, page_elems, sync=True,
                                     quant=q)
                else:
                    blocks = conn.allocate_rdma(keys, page_elems * 2)
                    conn.rdma_write_cache(src, [int(o) for o in offs], page_elems, blocks)
                    conn.sync()
                live.append(keys)
                ops[tid] += 1
                bytes_moved[tid] += nb * page_elems * 2
                # read a random live generation back
                rk = rng.choice(live)
                try:
                    if have_gpu:
                        conn.read_pages(dst, rk, offs, page_elems)
                    else:
                        conn.read_cache(dst, list(zip(rk, [int(o) for o in offs])),
                                        page_elems)
                    conn.sync()
                    bytes_moved[tid] += nb * page_elems * 2
                except Exception:
                    pass  # may have been evicted/deleted — allowed
                # occasionally delete an old generation
                if len(live) > 4 and rng.random() < 0.5:
                    victim = live.pop(rng.randrange(len(live) - 2))
                    conn.delete_keys(victim)
                # The freshest generation should normally be fully present,
                # but at a ~full pool the evictor may legitimately reclaim
                # pages between our sync and this query (the freshness guard
                # makes it rare, not impossible). Consistency check instead
                # of strict presence: the match answer must agree with
                # check_exist at the boundary.
                try:
                    m = conn.get_match_last_index(live[-1])
                except Exception:
                    m = -1  # whole generation evicted (raises on no match)
                if m != nb - 1:
                    assert m < nb - 1
                    assert not conn.check_exist(live[-1][m + 1])
            conn.close()
        except Exception as e:
            errors.append(f"t{tid}: {type(e).__name__}: {e}")

    threads = [threading.Thread(target=worker, args=(t,)) for t in range(args.threads)]
    t0 = time.time()
    for t in threads:
        t.start()

    while time.time() < stop:
        time.sleep(5)
        stats = json.loads(ifs.get_server_stats())
        moved = ifs.compact_pool()
        print(f"[soak] t={time.time()-t0:.0f}s kv={stats['kv_len']} "
              f"used={stats['used_blocks']}/{stats['total_blocks']} "
              f"ops={sum(ops)} compact_moved={moved[0]}", flush=True)
    for t in threads:
        t.join(timeout=60)

    wall = time.time() - t0
    gb = sum(bytes_moved) / 1e9
    print(f"[soak] DONE: {sum(ops)} write-gens, {gb:.1f} GB moved, "
          f"{gb / wall:.1f} GB/s sustained, errors={len(errors)}")
    for e in errors[:5]:
        print("[soak] ERROR:", e)
    ifs.unregister_server()
    return 1 if errors else 0


if __name__ == "__main__":
    sys.exit(main())
