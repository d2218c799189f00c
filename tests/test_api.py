"""Config validation and API-surface parity tests."""

import pytest

import infinistore_amd as ifs


def test_exports_match_reference_surface():
    # The reference package's public names must all exist here
    # (cf. /root/reference/infinistore/__init__.py:17-31).
    for name in [
        "InfinityConnection",
        "DisableTorchCaching",
        "register_server",
        "ClientConfig",
        "ServerConfig",
        "TYPE_RDMA",
        "TYPE_LOCAL_GPU",
        "Logger",
        "check_supported",
        "LINK_ETHERNET",
        "LINK_IB",
        "purge_kv_map",
        "get_kvmap_len",
    ]:
        assert hasattr(ifs, name), name


def test_connection_methods_match_reference():
    for m in [
        "connect",
        "connect_async",
        "local_gpu_write_cache",
        "rdma_write_cache",
        "rdma_write_cache_async",
        "read_cache",
        "read_cache_async",
        "sync",
        "check_exist",
        "get_match_last_index",
        "register_mr",
        "allocate_rdma",
        "allocate_rdma_async",
    ]:
        assert hasattr(ifs.InfinityConnection, m), m


def test_client_config_verify():
    with pytest.raises(Exception):
        ifs.ClientConfig(connection_type="bogus", host_addr="x", service_port=1).verify()
    with pytest.raises(Exception):
        ifs.ClientConfig(connection_type=ifs.TYPE_RDMA, host_addr="", service_port=1).verify()
    with pytest.raises(Exception):
        ifs.ClientConfig(
            connection_type=ifs.TYPE_RDMA, host_addr="x", service_port=0
        ).verify()
    with pytest.raises(Exception):
        ifs.ClientConfig(
            connection_type=ifs.TYPE_RDMA,
            host_addr="x",
            service_port=1,
            log_level="nope",
        ).verify()
    # valid
    ifs.ClientConfig(
        connection_type=ifs.TYPE_RDMA, host_addr="127.0.0.1", service_port=1234
    ).verify()


def test_server_config_verify():
    with pytest.raises(Exception):
        ifs.ServerConfig(service_port=0, manage_port=1).verify()
    with pytest.raises(Exception):
        ifs.ServerConfig(service_port=1, manage_port=0).verify()
    with pytest.raises(Exception):
        ifs.ServerConfig(service_port=1, manage_port=2, minimal_allocate_size=8).verify()
    ifs.ServerConfig(service_port=1, manage_port=2).verify()


def test_local_gpu_requires_localhost():
    cfg = ifs.ClientConfig(
        connection_type=ifs.TYPE_LOCAL_GPU, host_addr="10.0.0.1", service_port=1234
    )
    conn = ifs.InfinityConnection(cfg)
    with pytest.raises(Exception):
        conn.connect()


def test_connect_refused():
    cfg = ifs.ClientConfig(
        connection_type=ifs.TYPE_RDMA,
        host_addr="127.0.0.1",
        service_port=1,  # nothing listens here
        link_type="TCP",
    )
    conn = ifs.InfinityConnection(cfg)
    with pytest.raises(Exception):
        conn.connect()
