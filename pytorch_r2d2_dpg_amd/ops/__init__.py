"""Hot-op layer.

Every op has exactly two implementations and NO runtime multi-backend
dispatch (north-star constraint):

* a plain PyTorch fp32 reference in :mod:`.reference` — the CPU path and the
  numerics oracle for tests,
* a hand-written CDNA4 HIP kernel in ``csrc/`` (gfx950-only), loaded from the
  in-tree extension ``_r2d2dpg_hip``.

On a GPU device the HIP extension is REQUIRED: if a tensor is on a HIP device
and the extension is missing, ops raise instead of silently falling back to
eager PyTorch (the round-end harness records which .so files actually loaded).
"""

from __future__ import annotations


import torch

_EXT = None
_EXT_ERR: Exception | None = None


def _try_load():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return
    try:
        import importlib

        _EXT = importlib.import_module("pytorch_r2d2_dpg_amd.ops._r2d2dpg_hip")
    except Exception as e:  # noqa: BLE001
        _EXT_ERR = e


def hip_ext():
    """Return the HIP extension module, raising loudly if unavailable."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "pytorch_r2d2_dpg_amd HIP extension (_r2d2dpg_hip) is not built/loadable "
            "but a GPU op was requested. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Original import error: {_EXT_ERR!r}"
        )
    return _EXT


def ext_available() -> bool:
    _try_load()
    return _EXT is not None
