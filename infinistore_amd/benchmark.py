"""infinistore-amd benchmark harness.

Superset of the reference's harness (/root/reference/infinistore/benchmark.py:
same workload shape — `--size` MB written as uuid-keyed pages of `--block-size`
KB in `--steps` layer-batches, then read back, printing write/read MB/s) plus
the measurements BASELINE.md requires that the reference lacks: p50/p99
round-trip latency and a concurrent-client saturation mode.

Examples:
    python -m infinistore_amd.benchmark --server 127.0.0.1 --port 22345 \
        --size 1024 --block-size 128 --local-gpu
    python -m infinistore_amd.benchmark ... --clients 8   # saturation
"""

import argparse
import json
import statistics
import sys
import time
import uuid

import torch

from . import lib


def parse_args():
    p = argparse.ArgumentParser(description="infinistore-amd benchmark")
    p.add_argument("--server", default="127.0.0.1")
    p.add_argument("--port", type=int, default=22345)
    p.add_argument("--size", type=int, default=128, help="total MB per iteration")
    p.add_argument("--block-size", type=int, default=32, help="page size in KB")
    p.add_argument("--steps", type=int, default=32,
                   help="layer-batches per iteration (simulates per-layer writes)")
    p.add_argument("--iteration", type=int, default=3)
    p.add_argument("--local-gpu", action="store_true",
                   help="use the local IPC path (default: fabric path)")
    p.add_argument("--src-gpu", type=int, default=0)
    p.add_argument("--dst-gpu", type=int, default=0)
    p.add_argument("--latency-ops", type=int, default=0,
                   help="additionally measure N single-page round-trips")
    p.add_argument("--clients", type=int, default=1,
                   help="concurrent client processes (saturation mode)")
    p.add_argument("--shape", choices=["llama3-8b", "llama3-70b"],
                   help="preset: per-layer paged-KV shapes (16-token pages, "
                        "8 KV heads x 128 dim, bf16 -> 64 KB pages; 8B=32 "
                        "layers, 70B=80 layers); --steps becomes n_layers and "
                        "--size is derived from --seq-len x --batch")
    p.add_argument("--seq-len", type=int, default=8192)
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--quant", choices=["fp8"], default=None,
                   help="store pages fp8-compressed (local-GPU path, bf16)")
    p.add_argument("--verify", action="store_true")
    p.add_argument("--json", action="store_true", help="print a JSON summary")
    p.add_argument("--spawn-server", action="store_true",
                   help="start a local server subprocess for the benchmark")
    p.add_argument("--prealloc-size", type=int, default=8,
                   help="pool GB per shard for --spawn-server")
    return p.parse_args()


def make_conn(args, local):
    cfg = lib.ClientConfig(
        host_addr=args.server,
        service_port=args.port,
        connection_type=lib.TYPE_LOCAL_GPU if local else lib.TYPE_RDMA,
        link_type="TCP",
    )
    conn = lib.InfinityConnection(cfg)
    conn.connect()
    return conn


def make_buffers(args, conn, local, device):
    """src/dst buffers per worker, reused across iterations (fresh tensors
    every iteration would re-export IPC handles and churn the allocator —
    at 64 clients that dominates the wall). Split into sub-tensors under
    ~1.25 GiB each: the local path cannot IPC-import allocations >= 2 GiB
    (hipIpcOpenMemHandle hangs under dmabuf IPC on this driver stack —
    scripts/ipc_size_probe.py), and engines hold per-layer pools anyway.
    Returns (srcs, dsts, blocks_per_sub)."""
    block_bytes = args.block_size << 10
    total_bytes = args.size << 20
    n_blocks = total_bytes // block_bytes
    per_step = max(1, n_blocks // args.steps)
    dt = torch.bfloat16 if args.quant else torch.float32
    es = 2 if args.quant else 4

    cap = (1 << 30) + (1 << 28)  # 1.25 GiB
    n_sub = max(1, -(-total_bytes // cap))
    blocks_per_sub = -(-n_blocks // n_sub)
    blocks_per_sub = -(-blocks_per_sub // per_step) * per_step  # step-aligned
    n_sub = -(-n_blocks // blocks_per_sub)

    dst_dev = f"cuda:{args.dst_gpu}" if local else device
    srcs, dsts = [], []
    for t in range(n_sub):
        nb = min(blocks_per_sub, n_blocks - t * blocks_per_sub)
        elems = nb * block_bytes // es
        srcs.append(torch.rand(elems, dtype=dt, device=device))
        dsts.append(torch.zeros(elems, dtype=dt, device=dst_dev))
        if not local:
            conn.register_mr(srcs[-1])
            conn.register_mr(dsts[-1])
    return srcs, dsts, blocks_per_sub


def run_once(args, conn, local, device, bufs=None):
    block_bytes = args.block_size << 10
    total_bytes = args.size << 20
    n_blocks = total_bytes // block_bytes
    es = 2 if args.quant else 4
    page_elems = block_bytes // es

    srcs, dsts, blocks_per_sub = (bufs if bufs is not None
                                  else make_buffers(args, conn, local, device))

    run = uuid.uuid4().hex
    keys = [f"{run}-{i}" for i in range(n_blocks)]
    per_step = max(1, n_blocks // args.steps)

    def sub_of(i):  # block index -> (tensor index, element offset within it)
        return i // blocks_per_sub, (i % blocks_per_sub) * page_elems

    # ---- write (layer-by-layer batches, like prefill; each step's slice
    # lives in ONE sub-tensor by construction) ----
    t0 = time.perf_counter()
    for s0 in range(0, n_blocks, per_step):
        hi = min(s0 + per_step, n_blocks)
        t_idx, _ = sub_of(s0)
        pairs = [(keys[i], sub_of(i)[1]) for i in range(s0, hi)]
        if local:
            if args.quant:
                conn.write_pages(srcs[t_idx], [p[0] for p in pairs],
                                 [p[1] for p in pairs], page_elems,
                                 quant=args.quant)
            else:
                conn.local_gpu_write_cache(srcs[t_idx], pairs, page_elems)
        else:
            blocks = conn.allocate_rdma([p[0] for p in pairs], block_bytes)
            conn.rdma_write_cache(srcs[t_idx], [p[1] for p in pairs],
                                  page_elems, blocks)
    conn.sync()
    w_time = time.perf_counter() - t0

    # ---- read ----
    # Local path: ticketed async reads in ~64 MB chunks keep several
    # collect+kernel pipelines in flight per connection (one monolithic
    # request serializes its key-collect phase in front of everything).
    t0 = time.perf_counter()
    if local:
        import numpy as np

        chunk_blocks = max(1, min(blocks_per_sub, (64 << 20) // block_bytes))
        tickets = []
        for t_idx in range(len(dsts)):
            lo = t_idx * blocks_per_sub
            hi = min(lo + blocks_per_sub, n_blocks)
            for c0 in range(lo, hi, chunk_blocks):
                c1 = min(c0 + chunk_blocks, hi)
                ks = [keys[i] for i in range(c0, c1)]
                offs = np.asarray([sub_of(i)[1] for i in range(c0, c1)],
                                  dtype=np.uint64)
                tickets.append(conn.read_pages_async(dsts[t_idx], ks, offs,
                                                     page_elems))
        for tk in tickets:
            conn.wait_read(tk)
    else:
        for t_idx in range(len(dsts)):
            lo = t_idx * blocks_per_sub
            hi = min(lo + blocks_per_sub, n_blocks)
            conn.read_cache(dsts[t_idx],
                            [(keys[i], sub_of(i)[1]) for i in range(lo, hi)],
                            page_elems)
    conn.sync()
    r_time = time.perf_counter() - t0

    if args.verify:
        for a, b in zip(srcs, dsts):
            assert torch.equal(a.cpu(), b.cpu()), "verification failed"

    # Steady-state footprint: drop this iteration's keys (each iteration
    # writes a fresh key set; an engine similarly evicts finished
    # sequences). Runs after the timed windows but inside the wall clock.
    try:
        conn.delete_keys(keys)
    except Exception:
        pass

    return total_bytes / w_time / 1e6, total_bytes / r_time / 1e6


def run_latency(args, conn, local, device):
    block_bytes = args.block_size << 10
    page_elems = block_bytes // 4
    src = torch.rand(page_elems, dtype=torch.float32, device=device)
    dst = torch.zeros_like(src)
    if not local:
        conn.register_mr(src)
        conn.register_mr(dst)
    put_us, get_us = [], []
    run = uuid.uuid4().hex
    for i in range(args.latency_ops):
        key = f"lat-{run}-{i}"
        t0 = time.perf_counter()
        if local:
            conn.local_gpu_write_cache(src, [(key, 0)], page_elems)
        else:
            blocks = conn.allocate_rdma([key], block_bytes)
            conn.rdma_write_cache(src, [0], page_elems, blocks)
        conn.sync()
        t1 = time.perf_counter()
        conn.read_cache(dst, [(key, 0)], page_elems)
        conn.sync()
        t2 = time.perf_counter()
        put_us.append((t1 - t0) * 1e6)
        get_us.append((t2 - t1) * 1e6)
    return put_us, get_us


def pct(v, q):
    if not v:
        return 0.0
    return statistics.quantiles(v, n=100)[q - 1] if len(v) >= 10 else max(v)


def _worker(args_dict, q, barrier, wid):
    ns = argparse.Namespace(**args_dict)
    local = ns.local_gpu and torch.cuda.is_available()
    n_dev = torch.cuda.device_count() if torch.cuda.is_available() else 1
    device = f"cuda:{(ns.src_gpu + wid) % n_dev}" if local else "cpu"
    conn = None
    try:
        conn = make_conn(ns, local)
        bufs = make_buffers(ns, conn, local, device)
        run_once(ns, conn, local, device, bufs)  # warm: IPC opens, allocator
        barrier.wait(timeout=600)  # start all clients together (steady state)
        t0 = time.perf_counter()
        w = r = 0.0
        iters = max(1, ns.iteration)
        for _ in range(iters):
            w, r = run_once(ns, conn, local, device, bufs)
        wall = time.perf_counter() - t0
        q.put((w, r, wall, iters))
    except Exception as e:  # report instead of leaving the parent hanging
        q.put(("error", f"worker {wid}: {e}", 0.0, 0))
    finally:
        if conn is not None:
            conn.close()


def _spawn_server(args):
    import socket
    import subprocess
    import sys
    import time as _t

    proc = subprocess.Popen(
        [sys.executable, "-m", "infinistore_amd.server",
         "--service-port", str(args.port),
         "--manage-port", str(args.port + 1),
         "--prealloc-size", str(args.prealloc_size),
         "--minimal-allocate-size", str(args.block_size),
         "--no-manage"],
    )
    t0 = _t.time()
    while _t.time() - t0 < 60:
        try:
            socket.create_connection(("127.0.0.1", args.port), timeout=1).close()
            return proc
        except OSError:
            _t.sleep(0.3)
    raise RuntimeError("spawned server did not come up")


SHAPES = {
    # (n_layers, kv_page_bytes): 16-token pages, 8 KV heads, head_dim 128,
    # bf16, K+V -> 2*16*8*128*2 = 64 KiB per layer-page (Llama-3 GQA).
    "llama3-8b": (32, 64 << 10),
    "llama3-70b": (80, 64 << 10),
}


def apply_shape(args):
    n_layers, page_bytes = SHAPES[args.shape]
    pages_per_layer = (args.seq_len + 15) // 16 * args.batch
    args.block_size = page_bytes >> 10
    args.steps = n_layers  # one layer-batch of pages per step (prefill order)
    args.size = n_layers * pages_per_layer * page_bytes >> 20
    print(f"shape {args.shape}: {n_layers} layers x {pages_per_layer} pages "
          f"x {page_bytes >> 10} KB = {args.size} MB per iteration")


def main():
    args = parse_args()
    if args.shape:
        apply_shape(args)
    if args.spawn_server and args.clients > 1:
        # Keys live until each iteration's trailing delete; size the pool
        # for all clients' live iterations plus slack, capped so the pool +
        # the clients' own src/dst tensors (2x size each) still fit in HBM.
        need = args.clients * args.size * 2 * 1.3 / 1024 + 1
        if torch.cuda.is_available() and args.local_gpu:
            total_gb = torch.cuda.get_device_properties(0).total_memory / (1 << 30)
            client_gb = args.clients * args.size * 2 / 1024
            need = min(need, max(4, total_gb * 0.92 - client_gb))
        args.prealloc_size = max(args.prealloc_size, int(need))
    server_proc = _spawn_server(args) if args.spawn_server else None
    local = args.local_gpu and torch.cuda.is_available()
    device = f"cuda:{args.src_gpu}" if local else "cpu"

    if args.clients > 1:
        import multiprocessing

        ctx = multiprocessing.get_context("spawn")
        q = ctx.Queue()
        barrier = ctx.Barrier(args.clients)
        procs = [
            ctx.Process(target=_worker, args=(vars(args), q, barrier, wid))
            for wid in range(args.clients)
        ]
        for p in procs:
            p.start()
        results = [q.get(timeout=600) for _ in procs]
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
        errors = [r for r in results if r[0] == "error"]
        if errors:
            for e in errors:
                print(e[1], file=sys.stderr)
            if server_proc:
                server_proc.terminate()
            raise SystemExit(1)
        wall = max(r[2] for r in results)  # steady-state window (post-barrier)
        iters = results[0][3]
        agg = args.clients * (args.size << 20) * 2 * iters / wall / 1e6
        print(f"saturation: {args.clients} clients, aggregate {agg:.2f} MB/s "
              f"(per-client write {statistics.mean(r[0] for r in results):.2f} MB/s, "
              f"read {statistics.mean(r[1] for r in results):.2f} MB/s)")
        # Server-side truth (the client-side walls include Python/CPU
        # contention in the worker processes):
        try:
            probe = make_conn(args, False)
            stats = json.loads(probe.conn.get_stats())
            print(f"server counters: bytes_in={stats['bytes_in']/1e9:.1f} GB "
                  f"bytes_out={stats['bytes_out']/1e9:.1f} GB op_us={stats.get('op_us')}")
            probe.close()
        except Exception:
            pass
        if server_proc is not None:
            server_proc.terminate()
            server_proc.wait(timeout=20)
        return

    conn = make_conn(args, local)
    try:
        writes, reads = [], []
        for it in range(args.iteration):
            w, r = run_once(args, conn, local, device)
            writes.append(w)
            reads.append(r)
            print(f"[iter {it}] write cache: {w:.2f} MB/s, read cache: {r:.2f} MB/s")
        summary = {
            "write_MBps": round(statistics.mean(writes), 2),
            "read_MBps": round(statistics.mean(reads), 2),
            "block_kb": args.block_size,
            "size_mb": args.size,
            "path": "local_gpu" if local else "fabric",
        }
        if args.latency_ops:
            put_us, get_us = run_latency(args, conn, local, device)
            summary.update(
                p50_put_us=round(pct(put_us, 50), 1),
                p99_put_us=round(pct(put_us, 99), 1),
                p50_get_us=round(pct(get_us, 50), 1),
                p99_get_us=round(pct(get_us, 99), 1),
            )
            print(f"latency: put p50 {summary['p50_put_us']} us "
                  f"p99 {summary['p99_put_us']} us; get p50 {summary['p50_get_us']} us "
                  f"p99 {summary['p99_get_us']} us")
        if args.json:
            print(json.dumps(summary))
    finally:
        conn.close()
        if server_proc is not None:
            server_proc.terminate()
            server_proc.wait(timeout=20)


if __name__ == "__main__":
    main()
