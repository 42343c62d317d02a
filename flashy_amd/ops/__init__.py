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
import os
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
               p_bf16: tp.Optional[torch.Tensor] = None,
               step_dev: tp.Optional[torch.Tensor] = None) -> None:
    """step_dev: optional int64 device scalar holding the step count — makes
    the bias correction graph-replay-safe (advance it with adam_step_inc)."""
    ext = require()
    assert p.is_contiguous() and g.is_contiguous()
    ext.fused_adam(p.data_ptr(), g.data_ptr(), m.data_ptr(), v.data_ptr(),
                   p_bf16.data_ptr() if p_bf16 is not None else 0,
                   step_dev.data_ptr() if step_dev is not None else 0,
                   p.numel(), lr, beta1, beta2, eps, wd, step, grad_scale,
                   adamw, _stream())


def adam_step_inc(step_dev: torch.Tensor) -> None:
    require().adam_step_inc(step_dev.data_ptr(), _stream())


def maxpool_fwd(x: torch.Tensor, y: torch.Tensor, argmax: torch.Tensor,
                d: "ConvDims") -> None:
    assert d.C % 8 == 0
    require().maxpool_fwd(x.data_ptr(), y.data_ptr(), argmax.data_ptr(), *d,
                          _stream())


def maxpool_bwd(dy: torch.Tensor, argmax: torch.Tensor, dx: torch.Tensor,
                d: "ConvDims") -> None:
    require().maxpool_bwd(dy.data_ptr(), argmax.data_ptr(), dx.data_ptr(), *d,
                          _stream())


# ---------------------------------------------------------------------------
# convolution kernels (NHWC bf16 implicit GEMM; see flashy_amd/nn.py)
# All activation tensors are logical-NHWC [N, H, W, C]; weights [K, R, S, C].
# ---------------------------------------------------------------------------

class ConvDims(tp.NamedTuple):
    N: int; H: int; W: int; C: int
    K: int; R: int; S: int
    Ho: int; Wo: int
    stride: int; pad: int

    @staticmethod
    def infer(x: torch.Tensor, w: torch.Tensor, stride: int, pad: int) -> "ConvDims":
        N, H, W, C = x.shape
        K, R, S, Cw = w.shape
        assert C == Cw, (x.shape, w.shape)
        Ho = (H + 2 * pad - R) // stride + 1
        Wo = (W + 2 * pad - S) // stride + 1
        return ConvDims(N, H, W, C, K, R, S, Ho, Wo, stride, pad)


def _splitk_plan(M: int, ktiles: int, red_stages: int):
    """Split the reduction over grid.z into fp32 workspace slices when (a)
    the single-pass grid underfills the 256-CU chip, or (b) the K loop is
    long (>= 48 stages) — measured: splitting a 72-stage reduction to
    ~14 stages/slice wins 20-25% even on a full grid (shorter serial
    chains, more concurrent waves).  Returns zn (or 0 = single pass)."""
    if red_stages < 4:
        return 0
    blocks64 = ((M + 63) // 64) * ktiles
    if blocks64 >= 208:
        fill = 0
    else:
        fill = max(2, min((256 + blocks64 - 1) // max(1, blocks64),
                          red_stages // 2))
    longk = red_stages // 14 if red_stages >= 48 else 0
    zn = min(max(fill, longk), 8)
    return zn if zn >= 2 else 0


def conv_fwd(x: torch.Tensor, w: torch.Tensor, y: torch.Tensor, d: ConvDims,
             relu: bool = False,
             want_stats: bool = False) -> tp.Optional[tp.Tuple[torch.Tensor, int]]:
    """Forward conv.  With ``want_stats`` (non-split-K implicit-GEMM path
    only) the epilogue also emits BatchNorm partials [2][K][msplit] summed
    over each block's output rows; returns (partials, msplit) for
    :func:`bn_finalize`, else None."""
    ext = require()
    if d.C % 8 == 0:
        rsc = d.R * d.S * d.C
        assert d.K % 64 == 0 and rsc % 32 == 0, d
        M = d.N * d.Ho * d.Wo
        stages = (rsc + 63) // 64
        zn = 0 if ext.conv8_eligible(*d, False) else \
            _splitk_plan(M, d.K // 64, stages)
        if zn == 0 and stages >= 16 and not ext.conv8_eligible(*d, False):
            # small-M long-reduction (r4-class 1x1): the 8-wave grid
            # underfills without a K split — slice it so the 8-wave
            # split-K launcher engages
            mt8 = (M + 255) // 256
            bn8 = 128 if d.K % 128 == 0 else 64
            tiles8 = mt8 * (d.K // bn8)
            if tiles8 and tiles8 < 160:
                zn = min(max(2, -(-160 // tiles8)), 8, stages // 8)
                zn = 0 if zn < 2 else zn
        if zn:
            spz = (stages + zn - 1) // zn
            zeff = (stages + spz - 1) // spz
            ws = torch.empty(zeff * M * d.K, dtype=torch.float32, device=x.device)
            ext.conv_fwd_splitk(x.data_ptr(), w.data_ptr(), ws.data_ptr(), *d,
                                spz, _stream())
            ext.splitk_combine(ws.data_ptr(), y.data_ptr(), M * d.K, zeff, relu,
                               _stream())
        else:
            stats = None
            bn_ptr = 0
            if want_stats:
                msplit = ext.conv_fwd_msplit(*d)
                partials = torch.empty(2 * d.K * msplit, dtype=torch.float32,
                                       device=x.device)
                stats = (partials, msplit)
                bn_ptr = partials.data_ptr()
            ext.conv_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), *d, relu,
                         bn_ptr, _stream())
            return stats
    else:
        assert not relu
        M = d.N * d.Ho * d.Wo
        if d.C == 3 and d.K % 64 == 0 and d.R <= 8 and d.S <= 8 \
                and M >= 100_000:
            return _stem_fwd8(ext, x, w, y, d, want_stats)
        assert d.R * d.S * d.C <= 160 and d.K == 64, d
        ext.conv_stem_fwd(x.data_ptr(), w.data_ptr(), y.data_ptr(), *d, _stream())
    return None


def _stem_pad_x(ext, x: torch.Tensor, d: ConvDims) -> tp.Tuple[torch.Tensor, int, int]:
    """Zero-padded C'=4 / 8x8-tap copy of the C=3 stem input (see
    csrc/conv_fwd8.hip stem helpers)."""
    Hp = (d.Ho - 1) * d.stride + 8
    Wp = (d.Wo - 1) * d.stride + 8
    xp = torch.empty(d.N, Hp, Wp, 4, dtype=torch.bfloat16, device=x.device)
    ext.stem_pad_x(x.data_ptr(), xp.data_ptr(), d.N, d.H, d.W, Hp, Wp, d.pad,
                   _stream())
    return xp, Hp, Wp


def _stem_fwd8(ext, x, w, y, d: ConvDims, want_stats: bool):
    """Large-M C=3 stem through the 8-wave kernel: pad channels to 4 and
    taps to 8x8 (zero weights), rsc' = 256 — the direct small-C kernel runs
    at ~47 TF/s on the 224px stem; this path reaches the implicit-GEMM rate
    at 147/256 useful work."""
    xp, Hp, Wp = _stem_pad_x(ext, x, d)
    wp = torch.empty(d.K, 8, 8, 4, dtype=torch.bfloat16, device=x.device)
    ext.stem_pad_w(w.data_ptr(), wp.data_ptr(), d.K, d.R, d.S, _stream())
    M = d.N * d.Ho * d.Wo
    mtiles = (M + 255) // 256
    stats = None
    bn_ptr = 0
    if want_stats:
        partials = torch.empty(2 * d.K * mtiles, dtype=torch.float32,
                               device=x.device)
        stats = (partials, mtiles)
        bn_ptr = partials.data_ptr()
    ext.conv_fwd8_direct(xp.data_ptr(), wp.data_ptr(), y.data_ptr(),
                         d.N, Hp, Wp, 4, d.K, 8, 8, d.Ho, d.Wo, d.stride, 0,
                         False, bn_ptr, 64, mtiles, _stream())
    return stats


def conv_stem_dgrad(dout: torch.Tensor, w_krsc: torch.Tensor,
                    dx: torch.Tensor, d: ConvDims) -> None:
    """dX of a small-C edge conv (C <= 8, K == 64); weights in KRSC layout."""
    assert d.C <= 8 and d.K == 64 and d.R * d.S * d.C <= 160, d
    require().conv_stem_dgrad(dout.data_ptr(), w_krsc.data_ptr(),
                              dx.data_ptr(), *d, _stream())


def conv_dgrad(dout: torch.Tensor, w_rsck: torch.Tensor, dx: torch.Tensor,
               d: ConvDims) -> None:
    ext = require()
    assert d.C % 64 == 0 and d.K % 32 == 0, d
    M = d.N * d.H * d.W
    rsk = d.R * d.S * d.K
    stages = (rsk + 63) // 64
    zn = 0 if ext.conv8_eligible(*d, True) else \
        _splitk_plan(M, d.C // 64, stages)
    if zn == 0 and stages >= 16 and d.stride == 1 \
            and not ext.conv8_eligible(*d, True):
        mt8 = (M + 255) // 256
        bn8 = 128 if d.C % 128 == 0 else 64
        tiles8 = mt8 * (d.C // bn8)
        if tiles8 and tiles8 < 160:
            zn = min(max(2, -(-160 // tiles8)), 8, stages // 8)
            zn = 0 if zn < 2 else zn
    if zn:
        spz = (stages + zn - 1) // zn
        zeff = (stages + spz - 1) // spz
        ws = torch.empty(zeff * M * d.C, dtype=torch.float32, device=dout.device)
        ext.conv_dgrad_splitk(dout.data_ptr(), w_rsck.data_ptr(), ws.data_ptr(),
                              *d, spz, _stream())
        ext.splitk_combine(ws.data_ptr(), dx.data_ptr(), M * d.C, zeff, False,
                           _stream())
    else:
        ext.conv_dgrad(dout.data_ptr(), w_rsck.data_ptr(), dx.data_ptr(), *d,
                       _stream())


def weight_transpose_batched(src: torch.Tensor, dst: torch.Tensor,
                             meta: torch.Tensor, n_convs: int,
                             max_elems: int) -> None:
    require().weight_transpose_batched(src.data_ptr(), dst.data_ptr(),
                                       meta.data_ptr(), n_convs, max_elems,
                                       _stream())


def weight_transpose(w: torch.Tensor, wt: torch.Tensor) -> None:
    ext = require()
    K = w.shape[0]
    rsc = w.numel() // K
    ext.weight_transpose(w.data_ptr(), wt.data_ptr(), K, rsc, _stream())


def conv_wgrad(x: torch.Tensor, dout: torch.Tensor, dw: torch.Tensor,
               d: ConvDims, n_splits: tp.Optional[int] = None) -> None:
    """Accumulates (+=) fp32 weight grads; dw must be zeroed or hold the
    running gradient (matches autograd accumulate semantics)."""
    ext = require()
    rsc = d.R * d.S * d.C
    if d.C % 8 == 0:
        assert d.K % 64 == 0 and rsc % 64 == 0, d
        if n_splits is None:
            forced = os.environ.get("FLASHY_WGRAD_SPLITS")
            if forced:  # e.g. =1 -> deterministic wgrad (no atomic splits)
                n_splits = int(forced)
            else:
                tiles = (d.K // 64) * (rsc // 64)
                n_splits = max(1, min(1024 // tiles if tiles else 1, 128))
                M = d.N * d.Ho * d.Wo
                n_splits = max(1, min(n_splits, M // 32 or 1))
        ext.conv_wgrad(x.data_ptr(), dout.data_ptr(), dw.data_ptr(), *d,
                       n_splits, _stream())
    else:
        M = d.N * d.Ho * d.Wo
        if d.C == 3 and d.K % 64 == 0 and d.R <= 8 and d.S <= 8 \
                and M >= 100_000:
            # padded-stem wgrad: pad x to C'=4 / 8x8 taps, run the regular
            # implicit-GEMM wgrad on rsc'=256, unpad-accumulate into dw
            xp, Hp, Wp = _stem_pad_x(ext, x, d)
            dwp = torch.zeros(d.K * 256, dtype=torch.float32, device=x.device)
            dp = ConvDims(d.N, Hp, Wp, 4, d.K, 8, 8, d.Ho, d.Wo, d.stride, 0)
            tiles = (d.K // 64) * 4
            ns = max(1, min(1024 // tiles, 128, M // 32 or 1))
            forced = os.environ.get("FLASHY_WGRAD_SPLITS")
            if forced:
                ns = int(forced)
            ext.conv_wgrad(xp.data_ptr(), dout.data_ptr(), dwp.data_ptr(),
                           *dp, ns, _stream())
            ext.stem_unpad_dw(dwp.data_ptr(), dw.data_ptr(), d.K, d.R, d.S,
                              _stream())
            return
        assert rsc <= 160, d  # small-C stem kernels stage taps/weights in LDS
        ext.conv_stem_wgrad(x.data_ptr(), dout.data_ptr(), dw.data_ptr(), *d,
                            _stream())


# ---------------------------------------------------------------------------
# batchnorm kernels (NHWC training BN; see flashy_amd/nn.py)
# ---------------------------------------------------------------------------

def bn_msplit(M: int, C: int) -> int:
    """Blocks along M for the BN reductions: ~256 blocks saturate the chip
    while keeping the partial-combine kernels cheap."""
    cols = max(1, C // 64)
    msplit = max(1, min(512 // cols, 512))
    msplit = max(1, min(msplit, (M + 31) // 32))
    if msplit >= 4:
        msplit &= ~3  # multiple of 4: per-channel partial rows stay 16B-aligned
    return msplit


def bn_stats(x: torch.Tensor, partials: torch.Tensor, M: int, C: int,
             msplit: int) -> None:
    require().bn_stats(x.data_ptr(), partials.data_ptr(), M, C, msplit, _stream())


def bn_finalize(partials, msplit: int, gamma, beta, rmean, rvar, work,
                M: int, C: int, eps: float, momentum: float,
                update_running: bool) -> None:
    require().bn_finalize(partials.data_ptr(), msplit, gamma.data_ptr(),
                          beta.data_ptr(), rmean.data_ptr(), rvar.data_ptr(),
                          work.data_ptr(), M, C, eps, momentum,
                          update_running, _stream())


def bn_apply(x, res, y, work, M: int, C: int, relu: bool,
             slope: float = 0.0) -> None:
    require().bn_apply(x.data_ptr(), res.data_ptr() if res is not None else 0,
                       y.data_ptr(), work.data_ptr(), M, C, relu, slope,
                       _stream())


def bn_bwd_reduce(dy, y, x, work, dz_out, partials, M: int, C: int,
                  msplit: int, relu: bool, slope: float = 0.0) -> None:
    require().bn_bwd_reduce(dy.data_ptr(), y.data_ptr(), x.data_ptr(),
                            work.data_ptr(), dz_out.data_ptr(),
                            partials.data_ptr(), M, C, msplit, relu, slope,
                            _stream())


def bn_bwd_grads(partials, msplit: int, bsums, dgamma, dbeta, C: int) -> None:
    require().bn_bwd_grads(partials.data_ptr(), msplit, bsums.data_ptr(),
                           dgamma.data_ptr(), dbeta.data_ptr(), C, _stream())


def bn_bwd_apply(dz, x, work, bsums, dx, M: int, C: int) -> None:
    require().bn_bwd_apply(dz.data_ptr(), x.data_ptr(), work.data_ptr(),
                           bsums.data_ptr(), dx.data_ptr(), M, C, _stream())


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


def mse_fwd_bwd(x: torch.Tensor, t: torch.Tensor, dx: torch.Tensor,
                loss_sum: torch.Tensor, loss_scale: float,
                grad_scale: float) -> None:
    ext = require()
    assert x.is_contiguous() and t.is_contiguous() and x.dtype == t.dtype
    ext.mse(x.data_ptr(), t.data_ptr(), dx.data_ptr(), loss_sum.data_ptr(),
            x.numel(), loss_scale, grad_scale, _is_bf16(x), _stream())


def accuracy_count(logits: torch.Tensor, target: torch.Tensor,
                   out: torch.Tensor) -> None:
    """out[0] += number of rows whose argmax == target (caller zeroes out)."""
    ext = require()
    B, C = logits.shape
    assert logits.is_contiguous() and target.dtype == torch.int64
    ext.accuracy(logits.data_ptr(), target.data_ptr(), out.data_ptr(), B, C,
                 _is_bf16(logits), _stream())


# ---------------------------------------------------------------------------
# linear (fc) kernels — fp32, small shapes (see csrc/linear.hip)
# ---------------------------------------------------------------------------

def linear_fwd(x: torch.Tensor, w: torch.Tensor,
               b: tp.Optional[torch.Tensor], y: torch.Tensor) -> None:
    ext = require()
    B, I = x.shape
    O = w.shape[0]
    assert x.dtype == torch.float32 and w.shape == (O, I)
    ext.linear_fwd(x.data_ptr(), w.data_ptr(),
                   b.data_ptr() if b is not None else 0,
                   y.data_ptr(), B, I, O, _stream())


def linear_dx(dy: torch.Tensor, w: torch.Tensor, dx: torch.Tensor) -> None:
    ext = require()
    B, O = dy.shape
    I = w.shape[1]
    ext.linear_dx(dy.data_ptr(), w.data_ptr(), dx.data_ptr(), B, I, O,
                  _stream())


def linear_dw(x: torch.Tensor, dy: torch.Tensor, dw: torch.Tensor,
              db: tp.Optional[torch.Tensor]) -> None:
    """Accumulates (+=) into dw / db (autograd accumulate semantics)."""
    ext = require()
    B, I = x.shape
    O = dy.shape[1]
    ext.linear_dw(x.data_ptr(), dy.data_ptr(), dw.data_ptr(),
                  db.data_ptr() if db is not None else 0, B, I, O, _stream())
