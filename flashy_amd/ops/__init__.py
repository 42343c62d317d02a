# Copyright (c) Flashy-AMD authors.
"""CDNA4 (gfx950) kernel wrappers.

The native extension ``_hip_ops`` is built in-tree by
``python -m flashy_amd.ops.build`` (hipcc --offload-arch=gfx950) and loaded
from this package directory.  Wrappers pass raw device pointers and the
current HIP stream handle, so every launch lands on the torch stream and is
captured by HIP graphs like any torch kernel.

Policy: on a GPU box a missing extension is a HARD error (no silent eager
fallback — the HIP path must be the one that runs); pure-CPU runs (CI) use
torch fallbacks provided by the callers (see flashy_amd/optim.py,
flashy_amd/functional.py).
"""
from __future__ import annotations

import importlib.util
import typing as tp
from pathlib import Path

import torch

_ext = None
_load_error: tp.Optional[str] = None


def _find_so() -> tp.Optional[Path]:
    for pattern in ("_hip_ops*.so",):
        hits = sorted(Path(__file__).parent.glob(pattern))
        if hits:
            return hits[0]
    return None


def load_extension():
    """Load (once) and return the native module; raises with build advice."""
    global _ext, _load_error
    if _ext is not None:
        return _ext
    so = _find_so()
    if so is None:
        _load_error = "extension not built"
        raise RuntimeError(
            "flashy_amd._hip_ops is not built. Build it in-tree with:\n"
            "    python -m flashy_amd.ops.build\n"
            "(requires hipcc; cross-compiles for gfx950 without a GPU).")
    spec = importlib.util.spec_from_file_location("flashy_amd.ops._hip_ops", so)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)  # type: ignore[union-attr]
    _ext = mod
    return _ext


def available() -> bool:
    try:
        load_extension()
        return True
    except (RuntimeError, ImportError, OSError):
        return False


def require() -> tp.Any:
    """On a CUDA device the native kernels are mandatory: fail loudly."""
    return load_extension()


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def _is_bf16(t: torch.Tensor) -> bool:
    if t.dtype == torch.bfloat16:
        return True
    if t.dtype == torch.float32:
        return False
    raise TypeError(f"unsupported dtype {t.dtype} (float32/bfloat16 only)")


# ---------------------------------------------------------------------------
# optimizer kernels (flat fp32 buffers — see flashy_amd/optim.py)
# ---------------------------------------------------------------------------

def fused_sgd(p: torch.Tensor, g: torch.Tensor, m: tp.Optional[torch.Tensor],
              lr: float, momentum: float, wd: float, grad_scale: float = 1.0,
              nesterov: bool = False, p_bf16: tp.Optional[torch.Tensor] = None) -> None:
    ext = require()
    assert p.is_contiguous() and g.is_contiguous()
    ext.fused_sgd(p.data_ptr(), g.data_ptr(),
                  m.data_ptr() if m is not None else 0,
                  p_bf16.data_ptr() if p_bf16 is not None else 0,
                  p.numel(), lr, momentum, wd, grad_scale, nesterov, _stream())


def fused_adam(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor,
               v: torch.Tensor, lr: float, beta1: float, beta2: float,
               eps: float, wd: float, step: int, grad_scale: float = 1.0,
               adamw: bool = False,
               p_bf16: tp.Optional[torch.Tensor] = None) -> None:
    ext = require()
    assert p.is_contiguous() and g.is_contiguous()
    ext.fused_adam(p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr(),
                   p_bf16.data_ptr() if p_bf16 is not None else 0,
                   p.numel(), lr, beta1, beta2, eps, wd, step, grad_scale,
                   adamw, _stream())


# ---------------------------------------------------------------------------
# loss kernels (fused forward+input-grad; see flashy_amd/functional.py)
# ---------------------------------------------------------------------------

def cross_entropy_fwd_bwd(logits: torch.Tensor, target: torch.Tensor,
                          dlogits: torch.Tensor, loss_sum: torch.Tensor,
                          loss_scale: float, grad_scale: float) -> None:
    ext = require()
    B, C = logits.shape
    assert logits.is_contiguous() and target.dtype == torch.int64
    ext.cross_entropy(logits.data_ptr(), target.data_ptr(), dlogits.data_ptr(),
                      loss_sum.data_ptr(), B, C, loss_scale, grad_scale,
                      _is_bf16(logits), _stream())


def bce_logits_fwd_bwd(x: torch.Tensor, dx: torch.Tensor,
                       loss_sum: torch.Tensor, target: float,
                       loss_scale: float, grad_scale: float) -> None:
    ext = require()
    assert x.is_contiguous()
    ext.bce_logits(x.data_ptr(), dx.data_ptr(), loss_sum.data_ptr(), x.numel(),
                   target, loss_scale, grad_scale, _is_bf16(x), _stream())
