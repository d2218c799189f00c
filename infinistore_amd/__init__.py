"""infinistore-amd: MI355X-native GPU-direct KV-cache store.

Public API parity with the reference package's exports
(/root/reference/infinistore/__init__.py:1-31), plus MI355X-native
extensions (fingerprint_blocks, get_server_stats, unregister_server).
"""

from .lib import (
    InfinityConnection,
    DisableTorchCaching,
    ClientConfig,
    ServerConfig,
    TYPE_RDMA,
    TYPE_LOCAL_GPU,
    Logger,
    check_supported,
    LINK_ETHERNET,
    LINK_IB,
    LINK_TCP,
    register_server,
    unregister_server,
    purge_kv_map,
    get_kvmap_len,
    get_server_stats,
    compact_pool,
    snapshot_pool,
    restore_pool,
    fingerprint_blocks,
)

__all__ = [
    "InfinityConnection",
    "DisableTorchCaching",
    "register_server",
    "unregister_server",
    "ClientConfig",
    "ServerConfig",
    "TYPE_RDMA",
    "TYPE_LOCAL_GPU",
    "Logger",
    "check_supported",
    "LINK_ETHERNET",
    "LINK_IB",
    "LINK_TCP",
    "purge_kv_map",
    "get_kvmap_len",
    "get_server_stats",
    "snapshot_pool",
    "restore_pool",
    "compact_pool",
    "fingerprint_blocks",
]

__version__ = "0.1.0"
