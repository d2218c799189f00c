"""infinistore-amd client/server Python API.

API parity with the reference store's `infinistore/lib.py`
(/root/reference/infinistore/lib.py): `InfinityConnection` with the same
method names and unit conventions (offsets/page_size are in tensor
*elements*; conversion to bytes happens here), `ClientConfig`/`ServerConfig`
kwargs classes with `verify()`, `Logger`, `check_supported`,
`register_server`, `purge_kv_map`, `get_kvmap_len`, `DisableTorchCaching`.

Differences (MI355X-native build):
  * The server runs on a dedicated C++ thread — `register_server(config)`
    just starts it; an optional leading asyncio-loop argument is accepted
    and ignored for source compatibility with the reference signature.
  * TYPE_RDMA works with no RDMA NIC: the connection negotiates a data
    fabric at setup time and falls back to the TCP inline-data fabric
    (`link_type` may also be "TCP" to request it explicitly).
  * check_supported() probes ROCm/HIP + amdgpu instead of nv_peer_mem.
"""

import os
import time
import asyncio
from typing import List, Optional, Tuple

import numpy as np
import torch

from . import _build

_build.ensure_native()

from . import _native  # noqa: E402

# connection types
TYPE_LOCAL_GPU = "LOCAL_GPU"
TYPE_RDMA = "RDMA"
# rdma link types ("TCP" = inline-data fabric; auto-negotiated anyway)
LINK_ETHERNET = "Ethernet"
LINK_IB = "IB"
LINK_TCP = "TCP"


class ClientConfig(_native.ClientConfig):
    """Client configuration (kwargs: connection_type, host_addr, dev_name,
    ib_port, link_type, service_port, log_level)."""

    def __init__(self, **kwargs):
        super().__init__()
        self.connection_type = kwargs.get("connection_type", "") or ""
        self.host_addr = kwargs.get("host_addr", "") or ""
        self.dev_name = kwargs.get("dev_name", "")
        self.ib_port = kwargs.get("ib_port", 1)
        self.link_type = kwargs.get("link_type", "Ethernet")
        self.service_port = kwargs.get("service_port", 0) or 0
        if "INFINISTORE_LOG_LEVEL" in os.environ:
            self.log_level = os.environ["INFINISTORE_LOG_LEVEL"]
        else:
            self.log_level = kwargs.get("log_level", "warning")

    def __repr__(self):
        return (
            f"ClientConfig(service_port={self.service_port}, "
            f"log_level='{self.log_level}', host_addr='{self.host_addr}', "
            f"connection_type='{self.connection_type}', "
            f"dev_name='{self.dev_name}', ib_port={self.ib_port}, "
            f"link_type='{self.link_type}')"
        )

    def verify(self):
        if self.connection_type not in [TYPE_LOCAL_GPU, TYPE_RDMA]:
            raise Exception("Invalid connection type")
        if self.host_addr == "":
            raise Exception("Host address is empty")
        if self.service_port == 0:
            raise Exception("Service port is 0")
        if self.log_level not in ["error", "debug", "info", "warning"]:
            raise Exception("log level should be error, debug, info or warning")
        if self.ib_port < 1:
            raise Exception("ib port of device should be greater than 0")
        if self.connection_type == TYPE_RDMA and self.link_type not in [
            LINK_IB,
            LINK_ETHERNET,
            LINK_TCP,
        ]:
            raise Exception("link type should be IB, Ethernet or TCP for RDMA connection")


class ServerConfig(_native.ServerConfig):
    """Server configuration (kwargs: manage_port, service_port, log_level,
    dev_name, ib_port, link_type, prealloc_size (GB per shard),
    minimal_allocate_size (KB), num_stream, auto_increase, devices,
    cpu_only)."""

    def __init__(self, **kwargs):
        super().__init__()
        self.manage_port = kwargs.get("manage_port", 0)
        self.service_port = kwargs.get("service_port", 0)
        self.log_level = kwargs.get("log_level", "warning")
        self.dev_name = kwargs.get("dev_name", "")
        self.ib_port = kwargs.get("ib_port", 1)
        self.link_type = kwargs.get("link_type", "Ethernet")
        self.prealloc_size = kwargs.get("prealloc_size", 16)
        self.minimal_allocate_size = kwargs.get("minimal_allocate_size", 64)
        self.num_stream = kwargs.get("num_stream", 4)
        self.auto_increase = kwargs.get("auto_increase", False)
        self.devices = kwargs.get("devices", [])
        self.cpu_only = kwargs.get("cpu_only", False)
        self.cpu_shards = kwargs.get("cpu_shards", 1)
        self.auto_evict = kwargs.get("auto_evict", False)
        self.ttl_seconds = kwargs.get("ttl_seconds", 0)
        self.io_threads = kwargs.get("io_threads", 3)
        self.extend_size = kwargs.get("extend_size", 10)

    def __repr__(self):
        return (
            f"ServerConfig(service_port={self.service_port}, manage_port={self.manage_port}, "
            f"log_level='{self.log_level}', prealloc_size={self.prealloc_size}, "
            f"minimal_allocate_size={self.minimal_allocate_size}, "
            f"num_stream={self.num_stream}, devices={list(self.devices)}, "
            f"cpu_only={self.cpu_only})"
        )

    def verify(self):
        if self.service_port == 0:
            raise Exception("Service port is 0")
        if self.manage_port == 0:
            raise Exception("Manage port is 0")
        if self.log_level not in ["error", "debug", "info", "warning"]:
            raise Exception("log level should be error, debug, info or warning")
        if self.ib_port < 1:
            raise Exception("ib port of device should be greater than 0")
        if self.link_type not in [LINK_IB, LINK_ETHERNET, LINK_TCP]:
            raise Exception("link type should be IB, Ethernet or TCP")
        if self.minimal_allocate_size < 16:
            raise Exception("minimal allocate size should be greater than 16")


class Logger:
    @staticmethod
    def info(msg):
        _native.log_msg("info", str(msg))

    @staticmethod
    def debug(msg):
        _native.log_msg("debug", str(msg))

    @staticmethod
    def error(msg):
        _native.log_msg("error", str(msg))

    @staticmethod
    def warn(msg):
        _native.log_msg("warning", str(msg))

    @staticmethod
    def set_log_level(level):
        _native.set_log_level(level)


def get_kvmap_len():
    """Number of keys stored in the (in-process) server."""
    return _native.get_kvmap_len()


def purge_kv_map():
    """Drop every key from the (in-process) server; returns the count."""
    return _native.purge_kv_map()


def snapshot_pool(path: str):
    """Dump every committed page to `path` for a warm restart (extension —
    the reference cache is volatile). Returns (entries, payload_bytes)."""
    ok, n, b = _native.server_snapshot(path)
    if not ok:
        raise Exception(f"snapshot to {path} failed")
    return n, b


def restore_pool(path: str):
    """Load a snapshot back into the pool (existing keys win; stops cleanly
    when the pool fills). Returns (entries, payload_bytes)."""
    ok, n, b = _native.server_restore(path)
    if not ok:
        raise Exception(f"restore from {path} failed")
    return n, b


def get_server_stats():
    """JSON string with server counters (extension over the reference)."""
    return _native.server_stats()


def compact_pool():
    """Defragment the (in-process) server's pools by moving committed idle
    blocks to lower addresses with the batched copy kernel. Returns
    (moved_blocks, moved_bytes). Extension over the reference."""
    return _native.server_compact()


def register_server(*args):
    """Start the in-process server.

    Accepts `register_server(config)` or the reference's
    `register_server(loop, config)` (the loop argument is ignored: the
    server owns a dedicated C++ event-loop thread instead of borrowing
    uvloop's uv_loop_t*, cf. reference lib.py:179-205).
    """
    config = args[-1]
    config.verify()
    if not _native.start_server(config):
        raise Exception("Failed to start server")


def unregister_server():
    _native.stop_server()


def check_supported():
    """Probe platform support: ROCm GPUs present? amdgpu loaded? Logs
    warnings (the reference probes nv_peer_mem + ibv_devinfo instead)."""
    if not _native.gpu_available():
        Logger.warn("no ROCm GPU visible — server will run with a CPU (DRAM) pool")
        return False
    try:
        with open("/proc/modules") as f:
            mods = f.read()
        if "amdgpu" not in mods:
            Logger.warn("amdgpu module not listed in /proc/modules")
    except OSError:
        pass
    return True


class DisableTorchCaching:
    """Context manager that disables the torch caching allocator for tensors
    allocated inside it. With this build it is OPTIONAL for the local path:
    the wire protocol carries the offset of a tensor inside its allocation,
    so IPC round-trips work with the caching allocator too. Kept for API
    compatibility with the reference (lib.py:254-274)."""

    def __enter__(self):
        os.environ["PYTORCH_NO_CUDA_MEMORY_CACHING"] = "1"
        return self

    def __exit__(self, exc_type, exc_value, traceback):
        os.environ.pop("PYTORCH_NO_CUDA_MEMORY_CACHING", None)


def _remap_device_id(tensor: torch.Tensor) -> int:
    device_id = tensor.device.index
    if device_id is None:
        return 0
    visible = os.environ.get("CUDA_VISIBLE_DEVICES", "") or os.environ.get(
        "HIP_VISIBLE_DEVICES", ""
    )
    if visible:
        return int(visible.split(",")[device_id])
    return device_id


class InfinityConnection:
    """Connection to an infinistore-amd server (local IPC path or
    RDMA-semantics path over the negotiated fabric)."""

    OP_R = "R"
    OP_W = "W"
    OP_SYNC = "S"
    OP_RDMA_READ = "A"

    def __init__(self, config: ClientConfig):
        config.verify()
        self.conn = _native.Connection()
        self.local_connected = False
        self.rdma_connected = False
        self.config = config
        Logger.set_log_level(config.log_level)

    # -- connect ------------------------------------------------------------
    def connect(self):
        if self.local_connected:
            raise Exception("Already connected to local instance")
        if self.rdma_connected:
            raise Exception("Already connected to remote instance")
        ret = self.conn.init_connection(self.config)
        if ret < 0:
            raise Exception("Failed to initialize remote connection")
        if self.config.connection_type == TYPE_LOCAL_GPU:
            if self.config.host_addr not in ["127.0.0.1", "localhost"]:
                raise Exception("Local GPU connection must be to localhost")
            self.local_connected = True
        else:
            ret = self.conn.setup_rdma(self.config)
            if ret < 0:
                raise Exception("Failed to setup RDMA connection")
            self.rdma_connected = True

    async def connect_async(self):
        if self.config.connection_type == TYPE_LOCAL_GPU:
            raise Exception("Local GPU connection is not supported in async mode")
        loop = asyncio.get_running_loop()

        def blocking_connect():
            if self.conn.init_connection(self.config) < 0:
                raise Exception("Failed to initialize remote connection")
            if self.conn.setup_rdma(self.config) < 0:
                raise Exception("Failed to setup RDMA connection")
            self.rdma_connected = True

        await loop.run_in_executor(None, blocking_connect)

    def close(self):
        self.conn.close_conn()
        self.local_connected = False
        self.rdma_connected = False

    # -- local (IPC) path ---------------------------------------------------
    @staticmethod
    def _pack_blocks(blocks, element_size):
        """(keys_blob, offsets_bytes, n): keys NUL-joined, offsets in bytes as
        u64 — the packed fast-path arguments (avoids per-tuple marshalling)."""
        n = len(blocks)
        if n == 0:
            return b"", b"", 0
        keys, offsets = zip(*blocks)
        blob = "\x00".join(keys).encode()
        offs = np.fromiter(offsets, dtype=np.uint64, count=n) * np.uint64(element_size)
        return blob, offs.tobytes(), n

    def local_gpu_write_cache(
        self, cache: torch.Tensor, blocks: List[Tuple[str, int]], page_size: int
    ):
        """Write pages of `cache` (offsets in elements) under string keys."""
        self._verify(cache)
        assert self.local_connected
        element_size = cache.element_size()
        blob, offs, n = self._pack_blocks(blocks, element_size)
        ret = self.conn.rw_local_fast(
            self.OP_W,
            blob,
            offs,
            n,
            page_size * element_size,
            cache.data_ptr(),
            _remap_device_id(cache),
        )
        if ret < 0:
            raise Exception(f"Failed to write to infinistore, ret = {ret}")
        return 0

    # -- vectorized page API (extension) ------------------------------------
    # Same semantics as local_gpu_write_cache/read_cache but takes keys and
    # offsets as parallel sequences; `offsets` may be a reusable
    # np.ndarray(dtype=uint64) of ELEMENT offsets, skipping per-call tuple
    # marshalling on the hot path. `keys` may also be a pre-serialized bytes
    # blob (the NUL-joined key list, e.g. via pack_keys) — engines that hold
    # a page hash chain can serialize it once and skip the per-request join.
    @staticmethod
    def pack_keys(keys: List[str]) -> bytes:
        """Serialize a key list for the write_pages/read_pages blob form."""
        return "\0".join(keys).encode()

    FLAG_QUANT_FP8 = 2  # wire flag: store bf16 pages fp8-compressed

    def write_pages(self, cache: torch.Tensor, keys, offsets, page_size: int,
                    sync: bool = False, quant: Optional[str] = None):
        """sync=True completes the write in ONE round trip (the response is
        sent when the copy finishes); sync=False (default) returns after the
        server accepts, so uploads overlap compute — call sync() later.

        quant="fp8": store the pages quantized to fp8 e4m3 (one absmax/448
        scale per page) at HALF the HBM footprint; requires a bf16 cache
        tensor, and reads of these keys dequantize back to bf16
        transparently. Doubles effective cache capacity at ~3-bit mantissa
        precision — the usual KV-cache quantization trade."""
        self._verify(cache)
        assert self.local_connected, "write_pages uses the local GPU path"
        flags = 0
        if quant is not None:
            assert quant == "fp8", f"unsupported quant mode {quant!r}"
            assert cache.dtype == torch.bfloat16, "fp8 quant requires a bf16 cache"
            assert page_size % 8 == 0, "fp8 quant needs page_size % 8 == 0"
            flags = self.FLAG_QUANT_FP8
        es = cache.element_size()
        offs = np.asarray(offsets, dtype=np.uint64)
        if isinstance(keys, (bytes, bytearray, memoryview)):
            ret = self.conn.rw_local_blob(
                self.OP_W, keys, offs, es, page_size * es,
                cache.data_ptr(), _remap_device_id(cache), sync, flags,
            )
        else:
            ret = self.conn.rw_local_keys(
                self.OP_W, keys, offs, es, page_size * es,
                cache.data_ptr(), _remap_device_id(cache), sync, flags,
            )
        if ret < 0:
            raise Exception(f"Failed to write to infinistore, ret = {ret}")
        return 0

    def read_pages(self, cache: torch.Tensor, keys, offsets, page_size: int):
        self._verify(cache)
        es = cache.element_size()
        if self.local_connected:
            offs = np.asarray(offsets, dtype=np.uint64)
            if isinstance(keys, (bytes, bytearray, memoryview)):
                ret = self.conn.rw_local_blob(
                    self.OP_R, keys, offs, es, page_size * es,
                    cache.data_ptr(), _remap_device_id(cache),
                )
            else:
                ret = self.conn.rw_local_keys(
                    self.OP_R, keys, offs, es, page_size * es,
                    cache.data_ptr(), _remap_device_id(cache),
                )
        elif self.rdma_connected:
            blocks = [(k, int(o) * es) for k, o in zip(keys, offsets)]
            ret = self.conn.r_rdma(blocks, page_size * es, cache.data_ptr())
        else:
            raise Exception("Not connected to any instance")
        if ret < 0:
            raise Exception(f"Failed to read from infinistore, ret = {ret}")

    def read_pages_async(self, cache: torch.Tensor, keys, offsets, page_size: int):
        """Ticketed read (local path, shm ring): pushes the request and
        returns a ticket; call wait_read(ticket) before using the data. Lets
        an engine overlap its own work (or further requests) with the copy —
        e.g. prefetching the next sequence's KV pages while decoding. Falls
        back to a blocking read (ticket 0) when the ring is unavailable."""
        self._verify(cache)
        assert self.local_connected, "read_pages_async uses the local GPU path"
        es = cache.element_size()
        offs = np.asarray(offsets, dtype=np.uint64)
        if not isinstance(keys, (bytes, bytearray, memoryview)):
            keys = self.pack_keys(keys)
        ret, ticket = self.conn.rw_local_blob_async(
            self.OP_R, keys, offs, es, page_size * es,
            cache.data_ptr(), _remap_device_id(cache),
        )
        if ret < 0:
            raise Exception(f"Failed to read from infinistore, ret = {ret}")
        return ticket

    def wait_read(self, ticket: int):
        ret = self.conn.wait_local_ticket(ticket)
        if ret < 0:
            raise Exception(f"Async read failed, ret = {ret}")

    def read_cache(self, cache: torch.Tensor, blocks: List[Tuple[str, int]], page_size: int):
        """Read pages into `cache` (offsets in elements)."""
        self._verify(cache)
        element_size = cache.element_size()
        if self.local_connected:
            blob, offs, n = self._pack_blocks(blocks, element_size)
            ret = self.conn.rw_local_fast(
                self.OP_R,
                blob,
                offs,
                n,
                page_size * element_size,
                cache.data_ptr(),
                _remap_device_id(cache),
            )
        elif self.rdma_connected:
            blocks_in_bytes = [(key, offset * element_size) for key, offset in blocks]
            ret = self.conn.r_rdma(
                blocks_in_bytes, page_size * element_size, cache.data_ptr()
            )
        else:
            raise Exception("Not connected to any instance")
        if ret < 0:
            raise Exception(f"Failed to read from infinistore, ret = {ret}")

    async def read_cache_async(
        self, cache: torch.Tensor, blocks: List[Tuple[str, int]], page_size: int
    ):
        if not self.rdma_connected:
            raise Exception("this function is only valid for connected rdma")
        self._verify(cache)
        element_size = cache.element_size()
        blocks_in_bytes = [(key, offset * element_size) for key, offset in blocks]
        loop = asyncio.get_running_loop()
        future = loop.create_future()

        def _callback():
            loop.call_soon_threadsafe(future.set_result, 0)

        ret = self.conn.r_rdma_async(
            blocks_in_bytes, page_size * element_size, cache.data_ptr(), _callback
        )
        if ret < 0:
            raise Exception(f"Failed to read from infinistore, ret = {ret}")
        return await future

    # -- RDMA-semantics path --------------------------------------------------
    def register_mr(self, cache: torch.Tensor):
        self._verify(cache)
        if not self.rdma_connected:
            raise Exception("this function is only valid for connected rdma")
        ret = self.conn.register_mr(cache.data_ptr(), cache.numel() * cache.element_size())
        if ret < 0:
            raise Exception("register memory region failed")
        return ret

    def allocate_rdma(self, keys: List[str], page_size_in_bytes: int):
        if not self.rdma_connected:
            raise Exception("this function is only valid for connected rdma")
        ret = self.conn.allocate_rdma(keys, page_size_in_bytes)
        if len(ret) == 0:
            raise Exception("allocate memory failed")
        return ret

    async def allocate_rdma_async(self, keys: List[str], page_size_in_bytes: int):
        if not self.rdma_connected:
            raise Exception("this function is only valid for connected rdma")
        loop = asyncio.get_running_loop()
        future = loop.create_future()

        def _callback(remote_blocks):
            loop.call_soon_threadsafe(future.set_result, remote_blocks)

        self.conn.allocate_rdma_async(keys, page_size_in_bytes, _callback)
        return await future

    def rdma_write_cache(
        self, cache: torch.Tensor, offsets: List[int], page_size, remote_blocks: List
    ):
        assert self.rdma_connected
        self._verify(cache)
        element_size = cache.element_size()
        offsets_in_bytes = [offset * element_size for offset in offsets]
        ret = self.conn.w_rdma(
            offsets_in_bytes,
            page_size * element_size,
            [tuple(b) for b in remote_blocks],
            cache.data_ptr(),
        )
        if ret < 0:
            raise Exception(f"Failed to write to infinistore, ret = {ret}")
        return 0

    async def rdma_write_cache_async(
        self, cache: torch.Tensor, offsets: List[int], page_size, remote_blocks: List
    ):
        if not self.rdma_connected:
            raise Exception("this function is only valid for connected rdma")
        self._verify(cache)
        element_size = cache.element_size()
        offsets_in_bytes = [offset * element_size for offset in offsets]
        loop = asyncio.get_running_loop()
        future = loop.create_future()

        def _callback():
            loop.call_soon_threadsafe(future.set_result, 0)

        self.conn.w_rdma_async(
            offsets_in_bytes,
            page_size * element_size,
            [tuple(b) for b in remote_blocks],
            cache.data_ptr(),
            _callback,
        )
        return await future

    # -- sync / queries -------------------------------------------------------
    def sync(self):
        if self.local_connected:
            n = 0
            timeout = 1.0
            start = time.time()
            while True:
                ret = self.conn.sync_local()
                if ret < 0:
                    raise Exception(f"Failed to sync with infinistore, ret = {ret}")
                if ret == 0:
                    return
                if time.time() - start > timeout:
                    raise Exception("Timeout waiting for inflight requests")
                time.sleep(ret * 0.0005)
                n += 1
        elif self.rdma_connected:
            ret = self.conn.sync_rdma()
            if ret < 0:
                raise Exception(f"Failed to sync with infinistore, ret = {ret}")
        else:
            raise Exception("Not connected to any instance")

    def check_exist(self, key: str):
        ret = self.conn.check_exist(key)
        if ret < 0:
            raise Exception("Failed to check if this key exists")
        return ret == 0

    def get_match_last_index(self, keys: List[str]):
        ret = self.conn.get_match_last_index(keys)
        if ret < 0:
            raise Exception("can't find a match")
        return ret

    def get_server_stats_remote(self) -> str:
        """Server stats JSON fetched over the wire (extension) — works from
        any client, no management port needed."""
        return self.conn.get_stats()

    def delete_keys(self, keys: List[str]) -> int:
        """Delete keys from the store; returns the number removed.
        Extension over the reference (which only offers wholesale purge) —
        lets the inference engine evict cold prefixes."""
        ret = self.conn.delete_keys(keys)
        if ret < 0:
            raise Exception("Failed to delete keys")
        return ret

    # -- internal -------------------------------------------------------------
    def _verify(self, cache: torch.Tensor):
        if (not self.rdma_connected) and cache.device.type != "cuda":
            raise Exception("Tensor must be on CUDA device for local GPU connection")
        if cache.is_contiguous() is False:
            raise Exception("Tensor must be contiguous")


def fingerprint_blocks(cache: torch.Tensor, offsets: List[int], page_size: int):
    """GPU-accelerated 64-bit fingerprints of pages of `cache` (offsets and
    page_size in elements) — one HIP kernel launch. Use to build prefix-hash
    key chains for `get_match_last_index` without reading the KV data back to
    the host. Extension over the reference (new capability)."""
    if cache.device.type != "cuda":
        raise Exception("fingerprint_blocks requires a CUDA (ROCm) tensor")
    if not cache.is_contiguous():
        raise Exception("Tensor must be contiguous")
    es = cache.element_size()
    return _native.hash_blocks(
        cache.data_ptr(),
        [o * es for o in offsets],
        page_size * es,
        cache.device.index or 0,
    )
