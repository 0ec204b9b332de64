"""Build every native component in-tree.

  python -m ollamamq_amd.build            # HIP kernels + C++ dispatcher
  python -m ollamamq_amd.build --kernels  # just the gfx950 kernel .so

hipcc cross-compiles gfx950 without a GPU, so this runs in CPU-only CI.
Artifacts (git-ignored, but shipped in gpurun snapshots):
  ollamamq_amd/csrc/kernels/_kernels_gfx950.so
  ollamamq_amd/csrc/dispatcher/_dispatch*.so  + ollamamq-server binary
"""
from __future__ import annotations

import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))
KDIR = os.path.join(HERE, "csrc", "kernels")
DDIR = os.path.join(HERE, "csrc", "dispatcher")


def _run(cmd, **kw):
    print("+", " ".join(cmd), flush=True)
    subprocess.run(cmd, check=True, **kw)


def _newer(target, *sources):
    if not os.path.exists(target):
        return False
    t = os.path.getmtime(target)
    return all(os.path.getmtime(s) <= t for s in sources)


def build_kernels(force=False):
    import glob
    srcs = sorted(glob.glob(os.path.join(KDIR, "*.hip")))
    out = os.path.join(KDIR, "_kernels_gfx950.so")
    if not force and _newer(out, *srcs):
        print(f"kernels up to date: {out}")
        return out
    _run([
        "hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
        "-shared", "-fPIC", *srcs, "-o", out,
    ])
    return out


def build_dispatcher(force=False):
    """C++ dispatcher core: pybind11 module + standalone server binary."""
    import glob
    srcs = sorted(glob.glob(os.path.join(DDIR, "*.cpp")))
    if not srcs:
        return None
    import pybind11
    py_inc = subprocess.run(
        [sys.executable, "-c",
         "import sysconfig; print(sysconfig.get_paths()['include'])"],
        capture_output=True, text=True, check=True).stdout.strip()
    ext = os.path.join(DDIR, "_dispatch.so")
    core = [s for s in srcs if not s.endswith("main.cpp")
            and not s.endswith("bindings.cpp")
            and not s.endswith("tsan_stress.cpp")]
    bindings = os.path.join(DDIR, "bindings.cpp")
    if os.path.exists(bindings) and (force or not _newer(ext, *core, bindings)):
        _run(["g++", "-O2", "-std=c++17", "-shared", "-fPIC",
              "-I" + pybind11.get_include(), "-I" + py_inc, "-I" + DDIR,
              bindings, *core, "-o", ext, "-lpthread"])
    main = os.path.join(DDIR, "main.cpp")
    binary = os.path.join(DDIR, "ollamamq-server")
    if os.path.exists(main) and (force or not _newer(binary, *core, main)):
        _run(["g++", "-O2", "-std=c++17", "-I" + DDIR, main, *core,
              "-o", binary, "-lpthread"])
    return ext


def build_tsan_stress(force=False):
    """TSan-instrumented concurrency hammer (SURVEY §5 race detection)."""
    import glob
    srcs = sorted(glob.glob(os.path.join(DDIR, "*.cpp")))
    core = [s for s in srcs if not s.endswith("main.cpp")
            and not s.endswith("bindings.cpp")
            and not s.endswith("tsan_stress.cpp")]
    out = os.path.join(DDIR, "tsan_stress")
    stress = os.path.join(DDIR, "tsan_stress.cpp")
    if force or not _newer(out, stress, *core):
        # clang's TSan runtime: gcc-11's libtsan false-positives on
        # condition_variable timed waits ("double lock of a mutex")
        cxx = "/opt/rocm/lib/llvm/bin/clang++"
        if not os.path.exists(cxx):
            cxx = "g++"
        _run([cxx, "-O1", "-g", "-std=c++17", "-fsanitize=thread",
              "-I" + DDIR, stress, *core, "-o", out, "-lpthread"])
    return out


def build_asan_stress(force=False):
    """ASan+UBSan variant of the same hammer (heap errors, UB)."""
    import glob
    srcs = sorted(glob.glob(os.path.join(DDIR, "*.cpp")))
    core = [s for s in srcs if not s.endswith("main.cpp")
            and not s.endswith("bindings.cpp")
            and not s.endswith("tsan_stress.cpp")]
    out = os.path.join(DDIR, "asan_stress")
    stress = os.path.join(DDIR, "tsan_stress.cpp")
    if force or not _newer(out, stress, *core):
        cxx = "/opt/rocm/lib/llvm/bin/clang++"
        if not os.path.exists(cxx):
            cxx = "g++"
        _run([cxx, "-O1", "-g", "-std=c++17",
              "-fsanitize=address,undefined",
              "-fno-sanitize-recover=all",
              "-I" + DDIR, stress, *core, "-o", out, "-lpthread"])
    return out


def build_all(force=False):
    build_kernels(force)
    build_dispatcher(force)


if __name__ == "__main__":
    force = "--force" in sys.argv
    if "--kernels" in sys.argv:
        build_kernels(force)
    else:
        build_all(force)
