"""Syntax/type-check the ibverbs fabric against the mock rdma-core header
(this image ships no rdma-core; real builds pick up the system header via
__has_include). Catches bit-rot in the verbs module on every CPU test run."""

import os
import subprocess
import sysconfig

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


import pybind11


@pytest.mark.parametrize("src", [
    "csrc/fabric/verbs_fabric.cpp",
    "csrc/server/server_verbs.cpp",
    "csrc/client/client_verbs.cpp",
])
def test_verbs_sources_compile_against_mock(src):
    """With the mock on the include path, __has_include turns the real verbs
    branch ON — so this checks the code that a real rdma-core build would
    compile."""
    mock = os.path.join(REPO, "tests", "mock_verbs")
    cmd = [
        "g++", "-fsyntax-only", "-std=c++20", "-Wall", "-Werror",
        "-Wno-error=unused-variable",
        f"-I{mock}",
        f"-I{os.path.join(REPO, 'csrc')}",
        "-I/opt/rocm/include",
        "-D__HIP_PLATFORM_AMD__",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        os.path.join(REPO, src),
    ]
    r = subprocess.run(cmd, capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stderr


def test_stub_build_reports_not_compiled_in():
    # The shipped .so was built without rdma-core: compiled_in() is exposed
    # indirectly — verbs simply isn't offered, TCP fabric handles TYPE_RDMA.
    import infinistore_amd  # noqa: F401  (import sanity)
