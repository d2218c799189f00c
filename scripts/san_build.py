#!/usr/bin/env python3
"""Build a sanitizer-instrumented _native and run a test subset under it.

Usage:  python scripts/san_build.py asan [pytest args...]
        python scripts/san_build.py tsan [pytest args...]
        python scripts/san_build.py mockverbs [pytest args...]

mockverbs builds against tests/mock_verbs (an in-process loopback ibverbs
provider) so the verbs fabric — unreachable in this NIC-less environment —
executes for real: QP bring-up, SEND metadata, RDMA_WRITE chains,
WRITE_WITH_IMM reads, WrFlow. No sanitizer in that mode.

Host C++ TUs are rebuilt with g++ -fsanitize={address,thread}; the HIP TU
(csrc/gpu/gpu.hip) is compiled uninstrumented (device code cannot carry the
runtime). The instrumented module is placed in a temp dir that shadows
infinistore_amd/_native via PYTHONPATH, and pytest runs with the sanitizer
runtime LD_PRELOADed (python itself is uninstrumented).

Defaults: ASAN runs the protocol/pool/e2e/stress suites; TSAN runs the
fast concurrent subset (stress without the slow soak cases).
"""

import os
import subprocess
import sys
import sysconfig

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

SRCS = [
    "csrc/core/log.cpp", "csrc/core/protocol.cpp", "csrc/core/mempool.cpp",
    "csrc/fabric/verbs_fabric.cpp", "csrc/server/shard.cpp",
    "csrc/server/server.cpp", "csrc/server/server_verbs.cpp",
    "csrc/client/client.cpp", "csrc/client/client_verbs.cpp",
    "csrc/pybind.cpp",
]

DEFAULT_TESTS = {
    "asan": ["tests/test_wire.py", "tests/test_mempool.py", "tests/test_e2e_cpu.py",
             "tests/test_stress.py", "tests/test_shm_ring.py"],
    "tsan": ["tests/test_stress.py::test_many_threads_one_server",
             "tests/test_stress.py::test_async_ops_interleaved",
             "tests/test_shm_ring.py"],
    "mockverbs": ["tests/test_verbs_loopback.py"],
    "mockverbs-tsan": ["tests/test_verbs_loopback.py"],
    "mockverbs-asan": ["tests/test_verbs_loopback.py"],
}


def main():
    mode = sys.argv[1] if len(sys.argv) > 1 else "asan"
    assert mode in ("asan", "tsan", "mockverbs", "mockverbs-tsan", "mockverbs-asan"), \
        "mode must be asan|tsan|mockverbs|mockverbs-tsan|mockverbs-asan"
    san = {"asan": "address", "tsan": "thread", "mockverbs": None,
           "mockverbs-tsan": "thread", "mockverbs-asan": "address"}[mode]
    tests = sys.argv[2:] or DEFAULT_TESTS[mode]

    import pybind11

    build = f"/tmp/{mode}_build"
    os.makedirs(build + "/infinistore_amd", exist_ok=True)
    flags = ["-O1", "-g", "-std=c++20", "-fPIC",
             "-fno-omit-frame-pointer", "-D__HIP_PLATFORM_AMD__", "-Icsrc",
             f"-I{pybind11.get_include()}",
             f"-I{sysconfig.get_paths()['include']}", "-I/opt/rocm/include"]
    if san:
        flags.insert(4, f"-fsanitize={san}")
    srcs = list(SRCS)
    if mode.startswith("mockverbs"):
        flags.insert(0, "-Itests/mock_verbs")  # <infiniband/verbs.h> -> mock
        srcs.append("tests/mock_verbs/mock_verbs.cpp")
    # Rebuild when any header changed too (TUs inline shm_ring/wr_flow/etc.;
    # mixing object vintages across an inline-function change is ODR UB).
    hdr_mtime = 0.0
    for root, _d, files in os.walk(os.path.join(REPO, "csrc")):
        for f in files:
            if f.endswith((".h", ".hpp")):
                hdr_mtime = max(hdr_mtime, os.path.getmtime(os.path.join(root, f)))
    if mode.startswith("mockverbs"):
        for root, _d, files in os.walk(os.path.join(REPO, "tests/mock_verbs")):
            for f in files:
                if f.endswith(".h"):
                    hdr_mtime = max(hdr_mtime, os.path.getmtime(os.path.join(root, f)))

    objs = []
    for s in srcs:
        o = f"{build}/{s.replace('/', '_')}.o"
        src = os.path.join(REPO, s)
        if (not os.path.exists(o) or os.path.getmtime(o) < os.path.getmtime(src)
                or os.path.getmtime(o) < hdr_mtime):
            r = subprocess.run(["g++", "-c", src, "-o", o] + flags,
                               capture_output=True, text=True, cwd=REPO)
            if r.returncode:
                print(r.stderr[:4000])
                return 1
            print(f"  [{mode}] {s}")
        objs.append(o)
    gpu_o = f"{build}/gpu.o"
    gpu_src = os.path.join(REPO, "csrc/gpu/gpu.hip")
    if not os.path.exists(gpu_o) or os.path.getmtime(gpu_o) < os.path.getmtime(gpu_src):
        nosan = [f for f in flags if "sanitize" not in f and f != "-fno-omit-frame-pointer"]
        r = subprocess.run(["/opt/rocm/bin/hipcc", "-c", "csrc/gpu/gpu.hip", "-o",
                            gpu_o, "--offload-arch=gfx950"] + nosan,
                           capture_output=True, text=True, cwd=REPO)
        if r.returncode:
            print(r.stderr[:4000])
            return 1
    objs.append(gpu_o)

    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out = f"{build}/infinistore_amd/_native{suffix}"
    link = ["g++", "-shared"] + ([f"-fsanitize={san}"] if san else []) + ["-o", out] + objs +            ["-luv", "-L/opt/rocm/lib", "-lamdhip64"]
    r = subprocess.run(link, capture_output=True, text=True, cwd=REPO)
    if r.returncode:
        print(r.stderr[:4000])
        return 1

    # Shadow package: real python sources (incl. subpackages) + the
    # instrumented _native.
    pkg = os.path.join(REPO, "infinistore_amd")
    for root, _dirs, files in os.walk(pkg):
        rel = os.path.relpath(root, pkg)
        os.makedirs(os.path.join(build, "infinistore_amd", rel), exist_ok=True)
        for f in files:
            if f.endswith(".py"):
                src = os.path.join(root, f)
                dst = os.path.join(build, "infinistore_amd", rel, f)
                with open(src) as a, open(dst, "w") as b:
                    b.write(a.read())

    env = dict(os.environ)
    env["PYTHONPATH"] = build
    env["IFS_SKIP_BUILD"] = "1"
    if san:
        librt = subprocess.run(
            ["gcc", f"-print-file-name=lib{'asan' if san == 'address' else 'tsan'}.so"],
            capture_output=True, text=True).stdout.strip()
        env["LD_PRELOAD"] = librt
    if san == "address":
        env["ASAN_OPTIONS"] = "detect_leaks=0:abort_on_error=1"
    elif san:
        supp = os.path.join(REPO, "scripts/tsan.supp")
        if not os.path.exists(supp):
            open(supp, "w").write("# TSAN suppressions (python runtime noise)\n"
                                  "race:_Py\nrace:Py\n")
        env["TSAN_OPTIONS"] = "report_bugs=1:halt_on_error=0:suppressions=" + supp
    # Run from the BUILD dir: `python -m pytest` puts the cwd at the front
    # of sys.path, so running from the repo would import the repo's own
    # infinistore_amd and silently test the NORMAL build instead of the
    # instrumented one (import the module in-test and check __file__ if in
    # doubt). Test paths become absolute for the same reason.
    abs_tests = [t if os.path.isabs(t.split("::")[0]) else os.path.join(REPO, t)
                 for t in tests]
    print(f"  [{mode}] running: pytest {' '.join(tests)}")
    return subprocess.call([sys.executable, "-m", "pytest", "-x", "-q"] + abs_tests,
                           env=env, cwd=build)


if __name__ == "__main__":
    sys.exit(main())
