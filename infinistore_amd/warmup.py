"""Startup warmup: write/read one page on every visible GPU so HIP contexts,
IPC mappings and peer access are created before serving traffic (role of the
reference's infinistore/warmup.py:7-49)."""

import uuid

import torch

from . import lib


def check_p2p_access() -> bool:
    """Report peer-access availability between every GPU pair (role of the
    reference's server.py:99-109 helper). Returns True when every pair can
    peer (xGMI on an MI355X node)."""
    n = torch.cuda.device_count()
    ok = True
    for i in range(n):
        for j in range(n):
            if i == j:
                continue
            can = torch.cuda.can_device_access_peer(i, j)
            if not can:
                lib.Logger.warn(f"no peer access {i} -> {j}")
                ok = False
    if ok and n > 1:
        lib.Logger.info(f"p2p access OK across {n} GPUs")
    return ok


def warmup(service_port: int) -> bool:
    if not torch.cuda.is_available():
        lib.Logger.warn("warmup skipped: no GPU")
        return False
    ok = True
    for dev in range(torch.cuda.device_count()):
        cfg = lib.ClientConfig(
            host_addr="127.0.0.1",
            service_port=service_port,
            connection_type=lib.TYPE_LOCAL_GPU,
        )
        conn = lib.InfinityConnection(cfg)
        conn.connect()
        try:
            src = torch.arange(8192, dtype=torch.float32, device=f"cuda:{dev}")
            dst = torch.zeros(8192, dtype=torch.float32, device=f"cuda:{dev}")
            key = f"warmup-{dev}-{uuid.uuid4()}"
            conn.local_gpu_write_cache(src, [(key, 0)], 8192)
            conn.sync()
            conn.read_cache(dst, [(key, 0)], 8192)
            conn.sync()
            if not torch.equal(src, dst):
                lib.Logger.error(f"warmup mismatch on GPU {dev}")
                ok = False
        finally:
            conn.close()
    return ok


if __name__ == "__main__":
    import sys

    port = int(sys.argv[1]) if len(sys.argv) > 1 else 22345
    sys.exit(0 if warmup(port) else 1)
