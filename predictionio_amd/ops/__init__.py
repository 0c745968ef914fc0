"""Compute ops: gfx950 HIP kernels with CPU torch reference fallbacks.

Policy (per the framework's GPU contract):
- On a GPU box, the HIP extension MUST load — a missing/broken extension
  raises immediately rather than silently falling back to eager torch.
- On CPU-only machines (CI, dev), the pure-torch reference implementations
  in als.py / topk.py serve both as the CPU path and as the numerics
  reference the GPU kernels are tested against.
"""

from __future__ import annotations

import importlib.util
import os
import sys

_ext = None
_ext_err: Exception | None = None


def _load_ext():
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    so = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "_pio_hip.so")
    try:
        if not os.path.exists(so):
            raise ImportError(f"HIP extension not built: {so} missing "
                              "(run predictionio_amd/ops/build.py)")
        spec = importlib.util.spec_from_file_location("_pio_hip", so)
        mod = importlib.util.module_from_spec(spec)
        spec.loader.exec_module(mod)
        sys.modules["_pio_hip"] = mod
        _ext = mod
    except Exception as e:  # noqa: BLE001
        _ext_err = e
        _ext = None
    return _ext


def hip_ext():
    """The loaded HIP extension. Raises (loudly) if unavailable —
    callers only reach this on a CUDA/ROCm device path."""
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "predictionio_amd HIP extension (_pio_hip.so) failed to load on "
            f"a GPU path: {_ext_err!r}. Build it with "
            "`python -m predictionio_amd.ops.build`.") from _ext_err
    return ext


def hip_available() -> bool:
    import torch
    return torch.cuda.is_available() and _load_ext() is not None
