"""Multi-process (torch.distributed, gloo) coverage of the distributed
bench path — the same launch shape the scaling driver uses, on CPU:
rank 0 hosts the server, every rank runs a client, coordination over gloo.
"""

import json
import os
import subprocess
import sys

from conftest import free_port

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench_2rank(extra):
    env = os.environ.copy()
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--cpu",
         "--steps", "2", "--warmup", "1", "--blocks", "16",
         "--latency-ops", "3", "--port", str(free_port())] + extra,
        cwd=REPO, env=env, capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    out = json.loads(line)
    assert out["metric"] == "put_get_GBps"
    assert out["steps"] == 2
    assert out["value"] > 0
    return out


def test_bench_two_ranks_gloo():
    _run_bench_2rank([])


def test_bench_two_ranks_gloo_cross():
    # --cross: each rank reads the other rank's keys (the xGMI cross-shard
    # pattern, exercised here over the CPU fabric).
    _run_bench_2rank(["--cross"])
