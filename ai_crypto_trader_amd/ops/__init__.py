"""GPU op layer: HIP/CDNA4 kernels + CPU golden references.

Policy (enforced, see _try_load): on a machine WITH a GPU the HIP extension
is mandatory — ops raise rather than silently falling back to eager
PyTorch, so GPU tests can never pass on a hidden CPU path. On CPU-only
machines the numpy/torch reference implementations serve as the (tested)
fallback.
"""

from __future__ import annotations

import importlib
import os

_cached_mod = None
_load_error: Exception | None = None


def _try_load():
    global _cached_mod, _load_error
    if _cached_mod is not None:
        return _cached_mod
    try:
        _cached_mod = importlib.import_module(
            "ai_crypto_trader_amd.ops._hip_ops"
        )
    except ImportError as e:
        _load_error = e
        if os.environ.get("ACT_BUILD_ON_IMPORT") == "1":
            from .build import build
            build()
            _cached_mod = importlib.import_module(
                "ai_crypto_trader_amd.ops._hip_ops"
            )
            _load_error = None
    return _cached_mod


def hip_ops():
    """The raw extension module, or None if unavailable (CPU-only box)."""
    return _try_load()


def require_hip_ops():
    """The extension, mandatory. Raises with a clear message if missing."""
    mod = _try_load()
    if mod is None:
        raise RuntimeError(
            "HIP extension ai_crypto_trader_amd.ops._hip_ops is not built "
            "but a GPU path was requested. Build it with "
            "`python -m ai_crypto_trader_amd.ops.build` "
            f"(last import error: {_load_error!r})"
        )
    return mod


def gpu_available() -> bool:
    try:
        import torch
        return torch.cuda.is_available()
    except Exception:
        return False
