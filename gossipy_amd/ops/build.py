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


def build_asan(verbose: bool = False) -> str:
    """Device-AddressSanitizer build of the kernel extension (gfx950 with
    xnack+ page migration, which device ASAN requires). Loaded via
    ``GOSSIPY_HIP_SO`` + ``HSA_XNACK=1`` by the sanitizer pass over the
    GPU kernel tests (SURVEY.md §5 race-detection/sanitizer gap)."""
    os.environ["PYTORCH_ROCM_ARCH"] = "gfx950:xnack+"
    from torch.utils import cpp_extension

    here = os.path.dirname(os.path.abspath(__file__))
    src = os.path.join(here, "hip", "gossip_kernels.hip")
    build_dir = os.path.join(here, "hip", "build_asan")
    os.makedirs(build_dir, exist_ok=True)
    cpp_extension.load(
        name="_gossip_hip_asan",
        sources=[src],
        build_directory=build_dir,
        extra_cuda_cflags=[
            "-O1", "--offload-arch=gfx950:xnack+",
            "-Xarch_device", "-fsanitize=address", "-fgpu-sanitize",
        ],
        verbose=verbose,
        is_python_module=False,
        with_cuda=True,
    )
    built = os.path.join(build_dir, "_gossip_hip_asan.so")
    target = os.path.join(here, "_gossip_hip_asan.so")
    shutil.copy2(built, target)
    os.environ["PYTORCH_ROCM_ARCH"] = "gfx950"
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
