"""HIP/CDNA4 kernel extension loader.

The kernels live in ``hip/gossip_kernels.hip`` and are built **in-tree**
for gfx950 (``python -m gossipy_amd.ops.build`` or the repo's
``__graft_entry__.build()``), producing ``_gossip_hip.so`` next to this
file so the snapshot that travels to the GPU box carries it.

On a GPU box the extension is mandatory: :func:`load_extension` raises if
the ``.so`` is absent rather than silently falling back to eager PyTorch.
"""

from __future__ import annotations

import os

_EXT = None
_SCHED = None


def extension_path() -> str:
    # GOSSIPY_HIP_SO: alternate build of the same extension (e.g. the
    # device-AddressSanitizer build the sanitizer CI pass loads)
    override = os.environ.get("GOSSIPY_HIP_SO")
    if override:
        return override
    return os.path.join(os.path.dirname(__file__), "_gossip_hip.so")


def sched_path() -> str:
    override = os.environ.get("GOSSIPY_SCHED_SO")
    if override:
        return override
    return os.path.join(os.path.dirname(__file__), "_gossip_sched.so")


def load_sched():
    """Load (once) the native CPU scheduler, or None if not built."""
    global _SCHED
    if _SCHED is not None:
        return _SCHED
    import importlib.util

    path = sched_path()
    if not os.path.exists(path):
        return None
    modname = os.path.splitext(os.path.basename(path))[0]
    spec = importlib.util.spec_from_file_location(
        "gossipy_amd.ops." + modname, path
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _SCHED = mod
    return _SCHED


def load_extension():
    """Load (once) and return the compiled kernel module."""
    global _EXT
    if _EXT is not None:
        return _EXT
    import importlib.util

    import torch  # noqa: F401  (the .so links against torch libs)

    path = extension_path()
    if not os.path.exists(path):
        raise ImportError(
            "gossipy_amd HIP extension not built: %s missing. "
            "Run `python -m gossipy_amd.ops.build` (needs hipcc; "
            "cross-compiles for gfx950 without a GPU)." % path
        )
    # module name must match the .so's PyInit_<name> (the ASAN build
    # exports PyInit__gossip_hip_asan)
    modname = os.path.splitext(os.path.basename(path))[0]
    spec = importlib.util.spec_from_file_location(
        "gossipy_amd.ops." + modname, path
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    _EXT = mod
    return _EXT
