#!/usr/bin/env python3
"""In-tree build of the torch_quiver native extension for gfx950.

Drives hipcc directly (no hipify, no CUDA shims): kernel TUs compile fast
(HIP-only headers), the torch-binding TU carries the libtorch/pybind11 cost.
The resulting torch_quiver.so lands at the repo root so it travels with
gpurun snapshots.
"""
import os
import subprocess
import sys
import sysconfig
import concurrent.futures

ROOT = os.path.dirname(os.path.abspath(__file__))
CSRC = os.path.join(ROOT, "csrc")
BUILD = os.path.join(ROOT, "build")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

KERNEL_SOURCES = [
    "sample_kernels.hip",
    "reindex_kernels.hip",
    "gather_kernels.hip",
    "segment_kernels.hip",
    "wgrad_kernels.hip",
    "gemm_kernels.hip",
]
TORCH_SOURCES = ["module.cpp"]
OUT = os.path.join(ROOT, "torch_quiver.so")


def torch_paths():
    import torch  # noqa: F401  (heavy, but build-time only)
    import torch.utils.cpp_extension as ce

    tlib = ce.library_paths()[0]
    inc = ce.include_paths()
    return inc, tlib


def needs_rebuild(src, obj):
    if not os.path.exists(obj):
        return True
    dep = [src] + [os.path.join(CSRC, "qk_common.h")]
    omt = os.path.getmtime(obj)
    return any(os.path.getmtime(d) > omt for d in dep)


def run(cmd):
    print("+", " ".join(cmd), flush=True)
    subprocess.check_call(cmd)


def build(verbose=True):
    os.makedirs(BUILD, exist_ok=True)
    inc, tlib = torch_paths()
    py_inc = sysconfig.get_paths()["include"]

    common = [
        "hipcc", f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-DNDEBUG", "-Wno-unused-result",
    ]
    torch_flags = [
        "-DTORCH_EXTENSION_NAME=torch_quiver",
        "-D_GLIBCXX_USE_CXX11_ABI=1",
        "-DUSE_ROCM",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
    ] + [f"-I{p}" for p in inc] + [f"-I{py_inc}"]

    objs = []
    jobs = []
    for s in KERNEL_SOURCES:
        src = os.path.join(CSRC, s)
        obj = os.path.join(BUILD, s.replace(".hip", ".o"))
        objs.append(obj)
        if needs_rebuild(src, obj):
            jobs.append(common + ["-c", src, "-o", obj])
    for s in TORCH_SOURCES:
        src = os.path.join(CSRC, s)
        obj = os.path.join(BUILD, s.replace(".cpp", ".o"))
        objs.append(obj)
        if needs_rebuild(src, obj):
            jobs.append(common + torch_flags + ["-x", "hip", "-c", src, "-o", obj])

    if jobs:
        with concurrent.futures.ThreadPoolExecutor(max_workers=4) as ex:
            list(ex.map(run, jobs))

    if jobs or not os.path.exists(OUT):
        link = common + ["-shared", "-o", OUT] + objs + [
            f"-L{tlib}", f"-Wl,-rpath,{tlib}",
            "-ltorch", "-ltorch_python", "-lc10", "-lc10_hip", "-ltorch_hip",
            "-lamdhip64", "-lrccl",
        ]
        run(link)
    print(f"built {OUT}")

    # standalone C++ kernel test binary (no torch); travels with gpurun
    # snapshots, run by tests/test_gpu_cpp.py on the GPU box
    cpp_test_src = os.path.join(ROOT, "tests", "cpp", "test_kernels.hip")
    cpp_test_bin = os.path.join(BUILD, "qk_tests")
    kernel_objs = [os.path.join(BUILD, s.replace(".hip", ".o"))
                   for s in KERNEL_SOURCES]
    if (not os.path.exists(cpp_test_bin)
            or os.path.getmtime(cpp_test_src) > os.path.getmtime(cpp_test_bin)
            or any(os.path.getmtime(o) > os.path.getmtime(cpp_test_bin)
                   for o in kernel_objs)):
        cpp_test_obj = os.path.join(BUILD, "test_kernels.o")
        # -Wno-unused-value: assert-style test code ignores hipFree returns
        run(common + ["-Wno-unused-value", "-c", cpp_test_src,
                      "-o", cpp_test_obj])
        run(common + [cpp_test_obj] + kernel_objs + ["-o", cpp_test_bin])
    print(f"built {cpp_test_bin}")


if __name__ == "__main__":
    build()
