"""Synchronous client example: put/get KV pages through both data paths
(role of /root/reference/infinistore/example/client.py, covering the
src->dst device combinations that apply on this platform).

Start a server first:
    python -m infinistore_amd.server --service-port 22345 --manage-port 18080
"""

import uuid

import torch

import infinistore_amd as ifs


def run(conn, src_dev, dst_dev, page_elems=8192, n_pages=4):
    src = torch.rand(page_elems * n_pages, device=src_dev)
    dst = torch.zeros(page_elems * n_pages, device=dst_dev)
    run_id = uuid.uuid4().hex
    keys = [f"{run_id}-{i}" for i in range(n_pages)]
    offsets = [i * page_elems for i in range(n_pages)]

    if conn.local_connected:
        conn.local_gpu_write_cache(src, list(zip(keys, offsets)), page_elems)
    else:
        conn.register_mr(src)
        blocks = conn.allocate_rdma(keys, page_elems * src.element_size())
        conn.rdma_write_cache(src, offsets, page_elems, blocks)
    conn.sync()

    if not conn.local_connected:
        conn.register_mr(dst)
    conn.read_cache(dst, list(zip(keys, offsets)), page_elems)
    conn.sync()
    assert torch.equal(src.cpu(), dst.cpu())
    print(f"ok: {src_dev} -> store -> {dst_dev}")


def main():
    port = 22345
    if torch.cuda.is_available():
        cfg = ifs.ClientConfig(
            host_addr="127.0.0.1", service_port=port,
            connection_type=ifs.TYPE_LOCAL_GPU,
        )
        conn = ifs.InfinityConnection(cfg)
        conn.connect()
        run(conn, "cuda:0", "cuda:0")
        if torch.cuda.device_count() > 1:
            run(conn, "cuda:0", "cuda:1")  # cross-GPU read over xGMI
        conn.close()

    cfg = ifs.ClientConfig(
        host_addr="127.0.0.1", service_port=port,
        connection_type=ifs.TYPE_RDMA, link_type="TCP",
    )
    conn = ifs.InfinityConnection(cfg)
    conn.connect()
    run(conn, "cpu", "cpu")
    conn.close()


if __name__ == "__main__":
    main()
