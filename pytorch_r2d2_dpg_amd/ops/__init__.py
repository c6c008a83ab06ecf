"""Hot-op layer.

Every op has a plain PyTorch fp32 reference in :mod:`.reference` — the CPU
path and the numerics oracle for tests — and hand-written CDNA4 HIP kernels
in ``csrc/`` (gfx950-only), loaded from the in-tree extension
``_r2d2dpg_hip``. There is NO runtime multi-backend dispatch (north-star
constraint): on GPU the HIP path is the only path. Where csrc keeps an
alternate NATIVE kernel for an op (slab vs granule backward exchange,
pipelined forward), selection is a process-wide env choice measured and
documented in docs/KERNELS.md, never a per-call fallback.

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


_TUNABLE_DONE = False


def enable_tunableop() -> None:
    """Enable PyTorch TunableOp with the shipped gfx950 GEMM solution table.

    The learner's gates/weight-grad GEMMs are skinny shapes where
    hipBLASLt's default heuristic picks ~4% off-peak solutions (measured
    5.00 -> 4.81 ms/step on the flagship); the table was tuned across all
    five BASELINE configs' shapes on MI355X. Tuning itself stays OFF (the
    table is read-only; unknown shapes fall back to the heuristic), every
    DP rank reads the same file, and a version-mismatched install makes
    TunableOp ignore the table gracefully (validator lines). Opt out with
    R2D2_NO_TUNABLEOP=1; explicit PYTORCH_TUNABLEOP_* env takes precedence.
    """
    global _TUNABLE_DONE
    if _TUNABLE_DONE:
        return
    _TUNABLE_DONE = True
    import os

    if os.environ.get("R2D2_NO_TUNABLEOP"):
        return
    if any(k.startswith("PYTORCH_TUNABLEOP") for k in os.environ):
        return  # user drives TunableOp themselves (e.g. a tuning run)
    if not torch.cuda.is_available():
        return  # set_filename needs a device
    try:
        t = torch.cuda.tunable
    except AttributeError:  # older torch
        return
    csv = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                       "tunableop_gfx950.csv")
    if not os.path.exists(csv):
        return
    t.set_filename(csv, insert_device_ordinal=False)
    t.tuning_enable(False)
    t.enable(True)
