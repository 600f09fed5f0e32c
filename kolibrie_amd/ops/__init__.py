"""Native CDNA4 kernel dispatch.

On an MI355X the HIP kernels in `_native` are the execution path; the torch
composites in engine/ are the CPU oracle.  If a CUDA/HIP device is visible
but the extension is missing we fail LOUDLY instead of silently falling back
to eager torch (a silent fallback would fake GPU results at far lower
performance).
"""
from __future__ import annotations

import os

import torch

_native = None
_native_err: str | None = None

try:
    import importlib
    _native = importlib.import_module("kolibrie_amd.ops._native")
except ImportError as e:  # extension not built
    _native_err = str(e)

HAS_NATIVE = _native is not None

# escape hatch for A/B benchmarking the torch fallback on GPU
_FORCE_FALLBACK = os.environ.get("KOLIBRIE_FORCE_TORCH_FALLBACK", "") == "1"


def native_for(t: torch.Tensor):
    """Return the native module if `t` is a device tensor, else None.

    Raises if a device tensor is passed but the extension is unavailable.
    """
    if not t.is_cuda or _FORCE_FALLBACK:
        return None
    if _native is None:
        raise RuntimeError(
            "kolibrie_amd native kernels are required on GPU but the "
            "extension is not built (import error: %s). Build with "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`."
            % _native_err
        )
    return _native
