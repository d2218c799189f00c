"""infinistore-amd server entrypoint: `python -m infinistore_amd.server`.

CLI + management plane parity with the reference server
(/root/reference/infinistore/server.py): argparse flags -> ServerConfig,
FastAPI management endpoints (POST /purge, GET /kvmap_len,
POST /selftest/{port}) on --manage-port, optional --warmup and OOM-score
protection. The data-plane server itself runs on a dedicated C++ thread
(see csrc/server/server.h), so the management plane is a plain uvicorn
asyncio app — no shared uvloop handle.

Extensions: GET /stats (server counters as JSON).
"""

import argparse
import asyncio
import contextlib
import os
import sys

import torch  # noqa: F401  (ensures ROCm runtime is loaded first)

from . import lib
from .lib import Logger, ServerConfig, register_server, unregister_server

try:
    import uvicorn
    from fastapi import FastAPI

    _HAVE_FASTAPI = True
except Exception:  # pragma: no cover
    _HAVE_FASTAPI = False


def make_app(config: ServerConfig):
    app = FastAPI()

    @app.post("/purge")
    async def purge():
        n = lib.purge_kv_map()
        return {"status": "ok", "count": n}

    @app.get("/kvmap_len")
    async def kvmap_len():
        return {"len": lib.get_kvmap_len()}

    @app.get("/stats")
    async def stats():
        import json

        return json.loads(lib.get_server_stats())

    @app.post("/snapshot")
    async def snapshot(path: str):
        loop = asyncio.get_running_loop()
        n, b = await loop.run_in_executor(None, lib.snapshot_pool, path)
        return {"entries": n, "bytes": b}

    @app.post("/restore")
    async def restore(path: str):
        loop = asyncio.get_running_loop()
        n, b = await loop.run_in_executor(None, lib.restore_pool, path)
        return {"entries": n, "bytes": b}

    @app.post("/compact")
    async def compact():
        loop = asyncio.get_running_loop()
        moved, moved_bytes = await loop.run_in_executor(None, lib.compact_pool)
        return {"moved_blocks": moved, "moved_bytes": moved_bytes}

    @app.get("/metrics")
    async def metrics():
        # Prometheus exposition of the server counters (scrape target for
        # production fleets). Gauges are re-set per scrape from the native
        # stats snapshot; per-op handler timings become *_count / *_us_total.
        import json

        from fastapi.responses import PlainTextResponse

        s = json.loads(lib.get_server_stats())
        lines = []

        def g(name, val, help_=""):
            if help_:
                lines.append(f"# HELP infinistore_{name} {help_}")
            lines.append(f"# TYPE infinistore_{name} gauge")
            lines.append(f"infinistore_{name} {val}")

        g("kv_len", s.get("kv_len", 0), "Stored keys")
        g("used_blocks", s.get("used_blocks", 0))
        g("total_blocks", s.get("total_blocks", 0))
        g("bytes_in_total", s.get("bytes_in", 0), "Payload bytes written")
        g("bytes_out_total", s.get("bytes_out", 0), "Payload bytes read")
        g("evicted_total", s.get("evicted", 0))
        for op, st in (s.get("op_us") or {}).items():
            lines.append(f'infinistore_op_count{{op="{op}"}} {st.get("count", 0)}')
            lines.append(f'infinistore_op_avg_us{{op="{op}"}} {st.get("avg_us", 0)}')
            lines.append(f'infinistore_op_max_us{{op="{op}"}} {st.get("max_us", 0)}')
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.post("/selftest/{port}")
    async def selftest(port: int):
        # Loopback roundtrip through the full client stack (CPU tensor).
        import uuid

        def run():
            cfg = lib.ClientConfig(
                host_addr="127.0.0.1",
                service_port=port,
                connection_type=lib.TYPE_RDMA,
                link_type=lib.LINK_TCP,
                log_level=config.log_level,
            )
            conn = lib.InfinityConnection(cfg)
            conn.connect()
            try:
                src = torch.arange(4096, dtype=torch.float32)
                dst = torch.zeros(4096, dtype=torch.float32)
                key = f"selftest-{uuid.uuid4()}"
                blocks = conn.allocate_rdma([key], 4096 * 4)
                conn.register_mr(src)
                conn.rdma_write_cache(src, [0], 4096, blocks)
                conn.sync()
                conn.register_mr(dst)
                conn.read_cache(dst, [(key, 0)], 4096)
                conn.sync()
                return bool(torch.equal(src, dst))
            finally:
                conn.close()

        loop = asyncio.get_running_loop()
        ok = await loop.run_in_executor(None, run)
        return {"status": "ok" if ok else "failed"}

    return app


def parse_args():
    p = argparse.ArgumentParser(description="infinistore-amd server")
    p.add_argument("--service-port", type=int, default=22345)
    p.add_argument("--manage-port", type=int, default=18080)
    p.add_argument("--log-level", default="warning",
                   choices=["error", "debug", "info", "warning"])
    p.add_argument("--prealloc-size", type=int, default=16,
                   help="pool GB per shard (HBM when GPUs present)")
    p.add_argument("--minimal-allocate-size", type=int, default=64,
                   help="allocation granule in KB")
    p.add_argument("--num-stream", type=int, default=4,
                   help="HIP streams per GPU shard")
    p.add_argument("--io-threads", type=int, default=3,
                   help="worker IO loop threads (0 = single loop)")
    p.add_argument("--auto-increase", action="store_true",
                   help="extend the pool automatically when nearly full")
    p.add_argument("--restore-from", default="",
                   help="load a snapshot file into the pool at startup")
    p.add_argument("--ttl-seconds", type=int, default=0,
                   help="expire keys this many seconds after insert "
                        "(0 = keys live until deleted/evicted/purged)")
    p.add_argument("--auto-evict", action="store_true",
                   help="LRU-evict committed idle keys when the pool is full")
    p.add_argument("--devices", default="",
                   help="comma-separated GPU ordinals to shard over (default: all)")
    p.add_argument("--cpu-only", action="store_true",
                   help="force a CPU (DRAM) pool even when GPUs are visible")
    p.add_argument("--dev-name", default="", help="RDMA NIC name (verbs fabric only)")
    p.add_argument("--ib-port", type=int, default=1)
    p.add_argument("--link-type", default="Ethernet", choices=["IB", "Ethernet", "TCP"])
    p.add_argument("--warmup", action="store_true",
                   help="write/read one page on every GPU at startup")
    p.add_argument("--prevent-oom", action="store_true",
                   help="set oom_score_adj=-1000 (requires privileges)")
    p.add_argument("--no-manage", action="store_true",
                   help="skip the HTTP management plane")
    return p.parse_args()


def prevent_oom():
    try:
        with open(f"/proc/{os.getpid()}/oom_score_adj", "w") as f:
            f.write("-1000")
    except OSError as e:
        Logger.warn(f"could not set oom_score_adj: {e}")


def run_warmup(config: ServerConfig):
    from . import warmup

    warmup.warmup(config.service_port)


def main():
    args = parse_args()
    devices = [int(d) for d in args.devices.split(",") if d != ""]
    config = ServerConfig(
        service_port=args.service_port,
        manage_port=args.manage_port,
        log_level=args.log_level,
        prealloc_size=args.prealloc_size,
        minimal_allocate_size=args.minimal_allocate_size,
        num_stream=args.num_stream,
        io_threads=args.io_threads,
        auto_increase=args.auto_increase,
        auto_evict=args.auto_evict,
        ttl_seconds=args.ttl_seconds,
        devices=devices,
        cpu_only=args.cpu_only,
        dev_name=args.dev_name,
        ib_port=args.ib_port,
        link_type=args.link_type,
    )
    config.verify()
    lib.check_supported()
    _restore_path = args.restore_from
    if args.prevent_oom:
        prevent_oom()

    if torch.cuda.is_available() and torch.cuda.device_count() > 1:
        from .warmup import check_p2p_access

        check_p2p_access()
    register_server(config)
    if _restore_path:
        n, b = lib.restore_pool(_restore_path)
        print(f"restored {n} entries ({b >> 20} MB) from {_restore_path}",
              flush=True)
    print(f"infinistore-amd serving on :{config.service_port} "
          f"(manage :{config.manage_port})", flush=True)

    if args.warmup:
        run_warmup(config)

    try:
        if _HAVE_FASTAPI and not args.no_manage:
            app = make_app(config)
            uvicorn.run(app, host="0.0.0.0", port=config.manage_port,
                        log_level="warning")
        else:
            # Data plane only: park the main thread.
            import signal
            import threading

            ev = threading.Event()
            signal.signal(signal.SIGINT, lambda *a: ev.set())
            signal.signal(signal.SIGTERM, lambda *a: ev.set())
            ev.wait()
    except KeyboardInterrupt:
        pass
    finally:
        with contextlib.suppress(Exception):
            unregister_server()


if __name__ == "__main__":
    sys.exit(main())
