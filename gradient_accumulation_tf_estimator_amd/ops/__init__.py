"""Engine ops: HIP/CDNA4 kernels on GPU, plain-PyTorch fp32 on CPU.

On a CUDA (ROCm) device the hand-written gfx950 kernels in
``csrc/ga_kernels.hip`` are REQUIRED: if the in-tree extension is missing we
raise instead of silently falling back to eager PyTorch -- a GPU run must
exercise the native path.
"""

from __future__ import annotations

import importlib
from typing import Optional

from . import eager

_hip_mod = None
_hip_err: Optional[Exception] = None


def _load_hip():
    global _hip_mod, _hip_err
    if _hip_mod is not None or _hip_err is not None:
        return _hip_mod
    try:
        _hip_mod = importlib.import_module(
            "gradient_accumulation_tf_estimator_amd.ops._ga_hip"
        )
    except Exception as e:  # pragma: no cover - exercised on GPU boxes
        _hip_err = e
        _hip_mod = None
    return _hip_mod


def hip_available() -> bool:
    return _load_hip() is not None


def require_hip():
    mod = _load_hip()
    if mod is None:
        raise RuntimeError(
            "gradient_accumulation_tf_estimator_amd HIP extension (_ga_hip) is not "
            "built; run `python -m gradient_accumulation_tf_estimator_amd.ops.build` "
            f"(import error: {_hip_err})"
        )
    return mod
