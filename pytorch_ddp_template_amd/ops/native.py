"""Loader for the in-tree HIP extension (gfx950).

The compute path of this framework is hand-written CDNA4 HIP kernels
(``ops/csrc``) compiled in-tree to ``_hip_ops*.so``.  On a GPU box the
native extension is REQUIRED: any op asked to run on a CUDA (ROCm) tensor
without the extension raises loudly instead of silently falling back to
eager PyTorch.  On CPU-only machines (the dev container has no GPU) the
ops use plain PyTorch reference implementations, which are also what the
numerics tests compare the HIP kernels against.
"""

from __future__ import annotations

import importlib
import os

_ext = None
_ext_err: Exception | None = None


def _try_load():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return
    try:
        _ext = importlib.import_module("pytorch_ddp_template_amd._hip_ops")
    except Exception as e:  # noqa: BLE001
        _ext_err = e


def native_available() -> bool:
    _try_load()
    return _ext is not None


def native():
    """Return the HIP extension module, or raise if it is missing.

    Called on the GPU path of every op — this is the loud failure that
    prevents a silent eager fallback on a GPU box.
    """
    _try_load()
    if _ext is None:
        raise RuntimeError(
            "pytorch_ddp_template_amd._hip_ops is not built but a GPU tensor "
            "reached a native op. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950) "
            f"— original import error: {_ext_err!r}"
        )
    return _ext


def native_or_none():
    _try_load()
    return _ext


def use_native(*tensors) -> bool:
    """True iff all tensors are on a ROCm device (→ HIP kernels required)."""
    if os.environ.get("DDP_AMD_FORCE_EAGER") == "1":
        # escape hatch for A/B debugging only; never the default
        return False
    return all(t.is_cuda for t in tensors if t is not None)
