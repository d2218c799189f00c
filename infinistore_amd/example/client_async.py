"""Async client example (role of the reference's example/client_async.py):
allocate/write/read through the asyncio API."""

import asyncio
import uuid

import torch

import infinistore_amd as ifs


async def main(port: int = 22345):
    cfg = ifs.ClientConfig(
        host_addr="127.0.0.1", service_port=port,
        connection_type=ifs.TYPE_RDMA, link_type="TCP",
    )
    conn = ifs.InfinityConnection(cfg)
    await conn.connect_async()

    page = 8192
    n = 8
    src = torch.rand(page * n)
    dst = torch.zeros(page * n)
    conn.register_mr(src)
    conn.register_mr(dst)
    run_id = uuid.uuid4().hex
    keys = [f"{run_id}-{i}" for i in range(n)]
    offsets = [i * page for i in range(n)]

    blocks = await conn.allocate_rdma_async(keys, page * src.element_size())
    await conn.rdma_write_cache_async(src, offsets, page, blocks)
    conn.sync()
    await conn.read_cache_async(dst, list(zip(keys, offsets)), page)
    conn.sync()
    assert torch.equal(src, dst)
    print("async roundtrip ok")
    conn.close()


if __name__ == "__main__":
    import sys

    asyncio.run(main(int(sys.argv[1]) if len(sys.argv) > 1 else 22345))
