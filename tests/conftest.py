import socket

import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X (ROCm) GPU")
    config.addinivalue_line("markers", "benchmark: long-running benchmark test")


def free_port() -> int:
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture
def ports():
    return free_port(), free_port()


@pytest.fixture
def cpu_server(ports):
    """In-process CPU-pool server; yields the service port."""
    import infinistore_amd as ifs

    service_port, manage_port = ports
    cfg = ifs.ServerConfig(
        service_port=service_port,
        manage_port=manage_port,
        prealloc_size=1,
        minimal_allocate_size=16,
        cpu_only=True,
    )
    ifs.register_server(cfg)
    yield service_port
    ifs.unregister_server()


def make_client(port, conn_type=None, **kw):
    import infinistore_amd as ifs

    cfg = ifs.ClientConfig(
        host_addr="127.0.0.1",
        service_port=port,
        connection_type=conn_type or ifs.TYPE_RDMA,
        link_type="TCP",
        **kw,
    )
    conn = ifs.InfinityConnection(cfg)
    conn.connect()
    return conn
