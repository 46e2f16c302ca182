"""In-tree build of the gfx950 HIP extension.

Usage: ``python -m gossipy_amd.ops.build``. Cross-compiles on CPU-only
machines (hipcc needs no GPU); the resulting ``_gossip_hip.so`` sits next
to ``gossipy_amd/ops/__init__.py`` so it travels with the repo snapshot to
GPU boxes.
"""

from __future__ import annotations

import os
import shutil
import sys


def build_sched(verbose: bool = False) -> str:
    """Build the native CPU scheduler (plain C++, no HIP)."""
    from torch.utils import cpp_extension

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "..", "csrc", "scheduler.cpp")
    build_dir = os.path.join(here, "hip", "build_sched")
    os.makedirs(build_dir, exist_ok=True)
    cpp_extension.load(
        name="_gossip_sched",
        sources=[src],
        build_directory=build_dir,
        extra_cflags=["-O3"],
        verbose=verbose,
        is_python_module=False,
        with_cuda=False,
    )
    built = os.path.join(build_dir, "_gossip_sched.so")
    target = os.path.join(here, "_gossip_sched.so")
    shutil.copy2(built, target)
    return target


def build_sched_asan(verbose: bool = False) -> str:
    """AddressSanitizer build of the native C++ scheduler (pure CPU).
    The sanitizer pass runs the scheduler parity + fuzz suites under it:
    ``LD_PRELOAD=$(g++ -print-file-name=libasan.so) ASAN_OPTIONS=detect_leaks=0
    GOSSIPY_SCHED_SO=.../_gossip_sched_asan.so pytest tests/test_native_sched.py``."""
    import subprocess
    import sysconfig

    import pybind11

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "..", "csrc", "scheduler.cpp")
    target = os.path.join(here, "_gossip_sched_asan.so")
    # direct g++ (the scheduler is pure pybind11, no torch): building via
    # cpp_extension would dlopen the result immediately, which an
    # ASAN-linked .so refuses without the runtime preloaded
    cmd = [
        "g++", "-O1", "-g", "-std=c++17", "-shared", "-fPIC",
        "-fsanitize=address", "-fno-omit-frame-pointer",
        # PYBIND11_MODULE(_gossip_sched, m) -> rename the token so the
        # PyInit_ symbol matches this .so's filename
        "-D_gossip_sched=_gossip_sched_asan",
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_paths()['include']}",
        src, "-o", target,
    ]
    subprocess.run(cmd, check=True, capture_output=not verbose)
    return target


def build_asan(verbose: bool = False) -> str:
    """Device-AddressSanitizer build of the kernel extension (gfx950 with
    xnack+ page migration, which device ASAN requires). Loaded via
    ``GOSSIPY_HIP_SO`` + ``HSA_XNACK=1`` by the sanitizer pass over the
    GPU kernel tests (SURVEY.md §5 race-detection/sanitizer gap)."""
    import subprocess

    import torch
    from torch.utils import cpp_extension as ce

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "hip", "gossip_kernels.hip")
    build_dir = os.path.join(here, "hip", "build_asan")
    os.makedirs(build_dir, exist_ok=True)
    torch_lib = os.path.join(os.path.dirname(torch.__file__), "lib")
    includes = ce.include_paths(device_type="cuda")
    import sysconfig

    # host AND device instrumented (clang ASAN, shared runtime): the
    # device-asan runtime reports through the host one, which the test
    # process preloads (LD_PRELOAD=libclang_rt.asan-x86_64.so
    # ASAN_OPTIONS=detect_leaks=0). hipcc drives BOTH compile and link so
    # clang's sanitizer runtime is the one linked (g++ would pull GCC's).
    obj = os.path.join(build_dir, "gossip_kernels_asan.o")
    out = os.path.join(build_dir, "_gossip_hip_asan.so")
    common = [
        "-DWITH_HIP", "-DTORCH_EXTENSION_NAME=_gossip_hip_asan",
        "-DTORCH_API_INCLUDE_EXTENSION_H", "-D__HIP_PLATFORM_AMD__=1",
        "-DUSE_ROCM=1", "-DHIPBLAS_V2", "-fPIC", "-DCUDA_HAS_FP16=1",
        "-D__HIP_NO_HALF_OPERATORS__=1", "-D__HIP_NO_HALF_CONVERSIONS__=1",
        "-DHIP_ENABLE_WARP_SYNC_BUILTINS=1", "-std=c++17",
        "--offload-arch=gfx950:xnack+", "-fno-gpu-rdc", "-O1",
        "-fsanitize=address", "-fgpu-sanitize", "-shared-libsan",
    ]
    inc = [f"-isystem{p}" for p in includes] + [
        f"-isystem{sysconfig.get_paths()['include']}"
    ]
    subprocess.run(
        ["hipcc", *common, *inc, "-c", src, "-o", obj], check=True,
        capture_output=not verbose,
    )
    subprocess.run(
        ["hipcc", "-shared", *common, obj, f"-L{torch_lib}", "-lc10",
         "-lc10_hip", "-ltorch_cpu", "-ltorch_hip", "-ltorch",
         "-ltorch_python", "-L/opt/rocm/lib", "-lamdhip64", "-o", out],
        check=True, capture_output=not verbose,
    )
    target = os.path.join(here, "_gossip_hip_asan.so")
    shutil.copy2(out, target)
    return target


def build(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils import cpp_extension

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "hip", "gossip_kernels.hip")
    build_dir = os.path.join(here, "hip", "build")
    os.makedirs(build_dir, exist_ok=True)
    cpp_extension.load(
        name="_gossip_hip",
        sources=[src],
        build_directory=build_dir,
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
        is_python_module=False,
        with_cuda=True,
    )
    built = os.path.join(build_dir, "_gossip_hip.so")
    target = os.path.join(here, "_gossip_hip.so")
    shutil.copy2(built, target)
    build_sched(verbose)
    return target


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print("built:", path)
