#!/usr/bin/env python3
"""Build a sanitizer-instrumented _native and run a test subset under it.

Usage:  python scripts/san_build.py asan [pytest args...]
        python scripts/san_build.py tsan [pytest args...]

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
}


def main():
    mode = sys.argv[1] if len(sys.argv) > 1 else "asan"
    assert mode in ("asan", "tsan"), "mode must be asan|tsan"
    san = "address" if mode == "asan" else "thread"
    tests = sys.argv[2:] or DEFAULT_TESTS[mode]

    import pybind11

    build = f"/tmp/{mode}_build"
    os.makedirs(build + "/infinistore_amd", exist_ok=True)
    flags = ["-O1", "-g", "-std=c++20", "-fPIC", f"-fsanitize={san}",
             "-fno-omit-frame-pointer", "-D__HIP_PLATFORM_AMD__", "-Icsrc",
             f"-I{pybind11.get_include()}",
             f"-I{sysconfig.get_paths()['include']}", "-I/opt/rocm/include"]
    objs = []
    for s in SRCS:
        o = f"{build}/{s.replace('/', '_')}.o"
        src = os.path.join(REPO, s)
        if not os.path.exists(o) or os.path.getmtime(o) < os.path.getmtime(src):
            r = subprocess.run(["g++", "-c", src, "-o", o] + flags,
                               capture_output=True, text=True, cwd=REPO)
            if r.returncode:
                print(r.stderr[:4000])
                return 1
            print(f"  [{mode}] {s}")
        objs.append(o)
    gpu_o = f"{build}/gpu.o"
    if not os.path.exists(gpu_o):
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
    r = subprocess.run(["g++", "-shared", f"-fsanitize={san}", "-o", out] + objs +
                       ["-luv", "-L/opt/rocm/lib", "-lamdhip64"],
                       capture_output=True, text=True, cwd=REPO)
    if r.returncode:
        print(r.stderr[:4000])
        return 1

    # Shadow package: real python sources + instrumented _native.
    for f in os.listdir(os.path.join(REPO, "infinistore_amd")):
        if f.endswith(".py"):
            src = os.path.join(REPO, "infinistore_amd", f)
            dst = f"{build}/infinistore_amd/{f}"
            with open(src) as a, open(dst, "w") as b:
                b.write(a.read())

    librt = subprocess.run(
        ["gcc", f"-print-file-name=lib{'asan' if mode == 'asan' else 'tsan'}.so"],
        capture_output=True, text=True).stdout.strip()
    env = dict(os.environ)
    env["PYTHONPATH"] = build
    env["LD_PRELOAD"] = librt
    env["IFS_SKIP_BUILD"] = "1"
    if mode == "asan":
        env["ASAN_OPTIONS"] = "detect_leaks=0:abort_on_error=1"
    else:
        supp = os.path.join(REPO, "scripts/tsan.supp")
        if not os.path.exists(supp):
            open(supp, "w").write("# TSAN suppressions (python runtime noise)\n"
                                  "race:_Py\nrace:Py\n")
        env["TSAN_OPTIONS"] = "report_bugs=1:halt_on_error=0:suppressions=" + supp
    print(f"  [{mode}] running: pytest {' '.join(tests)}")
    return subprocess.call([sys.executable, "-m", "pytest", "-x", "-q"] + tests,
                           env=env, cwd=REPO)


if __name__ == "__main__":
    sys.exit(main())
