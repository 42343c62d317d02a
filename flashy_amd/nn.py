# Copyright (c) Flashy-AMD authors.
"""Native NHWC modules backed by the gfx950 kernels.

Layout convention: activations are logical-NHWC tensors ``[N, H, W, C]``
(plain contiguous — the same bytes as torch channels_last), conv weights are
``[K, R, S, C]``.  These modules are GPU-only (bf16 activations, fp32 master
params) and pair with :class:`flashy_amd.optim.FlatOptimizer`:

* weight gradients are fp32 and written (accumulated) directly into
  ``param.grad`` when it exists — with a flat optimizer that is the flat
  gradient buffer, so conv/BN backward needs no extra accumulate kernels;
* when the optimizer keeps a bf16 mirror of the flat params
  (``bf16_mirror=True``), forward reads the mirror view stashed on the
  parameter (``param._bf16_mirror``) — zero weight-cast kernels per step.

Replaces the ResNet conv+bn+relu torch chains of the reference workload
 with MFMA implicit-GEMM conv and fused NHWC BatchNorm
(+residual +ReLU) kernels.
"""
from __future__ import annotations

import math
import os
import typing as tp

import torch
from torch import nn

from . import ops


def _weight_bf16(w: torch.Tensor) -> torch.Tensor:
    mirror = getattr(w, "_bf16_mirror", None)
    if mirror is not None:
        return mirror
    return w.detach().to(torch.bfloat16)


def _grad_target(p: torch.Tensor) -> tp.Tuple[torch.Tensor, bool]:
    """(fp32 accumulation buffer, direct) — direct=True writes into
    ``p.grad`` in place (flat-optimizer fast path), else a temp returned to
    autograd for accumulation."""
    if p.grad is not None and p.grad.is_contiguous():
        return p.grad, True
    return torch.zeros_like(p, memory_format=torch.contiguous_format), False


class _ConvFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, stride: int, pad: int,
                input_grad: bool, wt_cached: tp.Optional[torch.Tensor],
                want_stats: bool = False):
        w16 = _weight_bf16(w)
        d = ops.ConvDims.infer(x, w16, stride, pad)
        y = x.new_empty((d.N, d.Ho, d.Wo, d.K))
        stats = ops.conv_fwd(x, w16, y, d, want_stats=want_stats)
        if stats is not None:
            # consumed by the following BatchNorm2d (skips its stats pass)
            y._bn_stats = stats
        ctx.save_for_backward(x, w16)
        ctx.dims = d
        ctx.input_grad = input_grad
        ctx.w_ref = w
        ctx.wt_cached = wt_cached
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w16 = ctx.saved_tensors
        d: ops.ConvDims = ctx.dims
        dy = dy.contiguous()
        dw_buf = None
        direct = True
        if ctx.needs_input_grad[1]:  # skip wgrad entirely for frozen weights
            dw_buf, direct = _grad_target(ctx.w_ref)
            ops.conv_wgrad(x, dy, dw_buf, d)
        dx = None
        if ctx.input_grad:
            dx = x.new_empty(x.shape)
            if d.C % 64 == 0:
                wt = ctx.wt_cached  # RSCK copy refreshed per step (WtCache)
                if wt is None:
                    wt = w16.new_empty((d.R, d.S, d.C, d.K))
                    ops.weight_transpose(w16, wt)
                ops.conv_dgrad(dy, wt, dx, d)
            else:  # small-C edge conv (e.g. a discriminator RGB stem)
                ops.conv_stem_dgrad(dy, w16, dx, d)
        return (dx, dw_buf if dw_buf is not None and not direct else None,
                None, None, None, None, None)


class Conv2d(nn.Module):
    """NHWC bf16 conv (no bias, as in ResNet).  Weight: [K, R, S, C] fp32."""

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding: int = 0, input_grad: bool = True,
                 feeds_bn: bool = False):
        super().__init__()
        self.stride = stride
        self.padding = padding
        self.input_grad = input_grad
        # when the conv output goes straight into a training BatchNorm2d,
        # the conv epilogue emits the BN sum/sumsq partials for free
        self.feeds_bn = feeds_bn
        self._wt_view: tp.Optional[torch.Tensor] = None
        k = kernel_size
        self.weight = nn.Parameter(
            torch.empty(out_channels, k, k, in_channels))
        fan_out = out_channels * k * k
        nn.init.normal_(self.weight, std=math.sqrt(2.0 / fan_out))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _ConvFn.apply(x, self.weight, self.stride, self.padding,
                             self.input_grad and x.requires_grad,
                             self._wt_view, self.feeds_bn and self.training)


class WtCache:
    """One arena + ONE batched kernel refreshing every conv's RSCK weight
    copy per step (replaces a transpose launch per conv per backward).

    Requires the flat optimizer's bf16 mirror (``bf16_mirror=True``): the
    kernel reads each weight's mirror view straight out of the flat buffer.
    Call :meth:`refresh` once per step before backward — NativeResNet does it
    at the top of ``forward``.
    """

    def __init__(self, model: nn.Module):
        convs = [m for m in model.modules()
                 if isinstance(m, Conv2d) and m.input_grad
                 and m.weight.shape[-1] % 64 == 0]
        metas, total = [], 0
        mirrors = [getattr(c.weight, "_bf16_mirror", None) for c in convs]
        if not convs or any(m is None for m in mirrors):
            self.meta = None
            return
        base = mirrors[0]
        for c, mir in zip(convs, mirrors):
            assert mir.untyped_storage().data_ptr() == \
                base.untyped_storage().data_ptr(), \
                "all conv weights must share one flat bf16 mirror"
            src_off = (mir.data_ptr() - base.data_ptr()) // 2
            assert src_off >= 0
            K, R, S, C = c.weight.shape
            metas.append((src_off, total, K, R * S * C))
            total += K * R * S * C
        dev = convs[0].weight.device
        self.src = mirrors[0]
        self.arena = torch.empty(total, dtype=torch.bfloat16, device=dev)
        self.meta = torch.tensor([list(m) for m in metas],
                                 dtype=torch.int32, device=dev).flatten()
        self.max_elems = max(m[2] * m[3] for m in metas)
        self.n = len(convs)
        for c, m in zip(convs, metas):
            K, R, S, C = c.weight.shape
            c._wt_view = self.arena[m[1]:m[1] + K * R * S * C].view(R, S, C, K)

    @property
    def active(self) -> bool:
        return self.meta is not None

    def refresh(self) -> None:
        if self.meta is not None:
            ops.weight_transpose_batched(self.src, self.arena, self.meta,
                                         self.n, self.max_elems)


class _BnFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                res: tp.Optional[torch.Tensor], relu: bool, slope: float,
                module):
        N, H, W, C = x.shape
        M = N * H * W
        y = torch.empty_like(x)
        if module.training:
            fused = getattr(x, "_bn_stats", None)
            if fused is not None:   # producing conv already emitted partials
                partials, msplit = fused
            else:
                msplit = ops.bn_msplit(M, C)
                partials = torch.empty(msplit * 2 * C, dtype=torch.float32,
                                       device=x.device)
                ops.bn_stats(x, partials, M, C, msplit)
            work = torch.empty(4 * C, dtype=torch.float32, device=x.device)
            ops.bn_finalize(partials, msplit, gamma, beta, module.running_mean,
                            module.running_var, work, M, C, module.eps,
                            module.momentum, update_running=True)
        else:
            invstd = torch.rsqrt(module.running_var + module.eps)
            scale = gamma.detach() * invstd
            shift = beta.detach() - module.running_mean * scale
            work = torch.cat([module.running_mean, invstd, scale, shift])
        ops.bn_apply(x, res, y, work, M, C, relu, slope)
        ctx.save_for_backward(x, y, work)
        ctx.relu = relu
        ctx.slope = slope
        ctx.has_res = res is not None
        ctx.refs = (gamma, beta)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, y, work = ctx.saved_tensors
        gamma, beta = ctx.refs
        N, H, W, C = x.shape
        M = N * H * W
        dy = dy.contiguous()
        msplit = ops.bn_msplit(M, C)
        partials = torch.empty(msplit * 2 * C, dtype=torch.float32,
                               device=x.device)
        bsums = torch.empty(2 * C, dtype=torch.float32, device=x.device)
        dz = torch.empty_like(dy)
        ops.bn_bwd_reduce(dy, y, x, work, dz, partials, M, C, msplit,
                          ctx.relu, ctx.slope)
        # bn_bwd_grads also produces bsums (needed for dx), so the kernel
        # always runs; frozen gamma/beta just accumulate into a discarded
        # temp instead of param.grad (never pollute frozen params' grads).
        need_g, need_b = ctx.needs_input_grad[1], ctx.needs_input_grad[2]
        dgamma, g_direct = _grad_target(gamma) if need_g \
            else (torch.empty_like(gamma), True)
        dbeta, b_direct = _grad_target(beta) if need_b \
            else (torch.empty_like(beta), True)
        ops.bn_bwd_grads(partials, msplit, bsums, dgamma, dbeta, C)
        dx = torch.empty_like(x)
        ops.bn_bwd_apply(dz, x, work, bsums, dx, M, C)
        return (dx,
                dgamma if need_g and not g_direct else None,
                dbeta if need_b and not b_direct else None,
                dz if ctx.has_res else None,
                None, None, None)


class _ConvTransposeFn(torch.autograd.Function):
    """Transposed conv via the equivalent regular conv C (weight [K=C_in]
    [R][S][C=C_out]):  fwd = C's dgrad,  dx = C's fwd,  dw = C's wgrad with
    the operand roles swapped (dL/dw = wgrad(x=dy, dout=x))."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, stride: int, pad: int,
                input_grad: bool):
        w16 = _weight_bf16(w)
        N, Hi, Wi, Cin = x.shape
        K, R, S, Cout = w16.shape
        assert K == Cin, (x.shape, w16.shape)
        Ho = (Hi - 1) * stride - 2 * pad + R
        Wo = (Wi - 1) * stride - 2 * pad + S
        # equivalent-conv dims: "input" = our OUTPUT, "output" = our input
        d = ops.ConvDims(N, Ho, Wo, Cout, Cin, R, S, Hi, Wi, stride, pad)
        y = x.new_empty((N, Ho, Wo, Cout))
        if Cout % 64 == 0:
            wt = w16.new_empty((R, S, Cout, Cin))
            ops.weight_transpose(w16, wt)
            ops.conv_dgrad(x, wt, y, d)
        else:  # RGB head: small-C direct kernel, KRSC weights as stored
            ops.conv_stem_dgrad(x, w16, y, d)
        ctx.save_for_backward(x, w16)
        ctx.dims = d
        ctx.input_grad = input_grad
        ctx.w_ref = w
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w16 = ctx.saved_tensors
        d: ops.ConvDims = ctx.dims
        dy = dy.contiguous()
        dw_buf = None
        direct = True
        if ctx.needs_input_grad[1]:  # skip wgrad entirely for frozen weights
            dw_buf, direct = _grad_target(ctx.w_ref)
            ops.conv_wgrad(dy, x, dw_buf, d)   # roles swapped vs regular conv
        dx = None
        if ctx.input_grad:
            dx = x.new_empty(x.shape)
            ops.conv_fwd(dy, w16, dx, d)
        return (dx, dw_buf if dw_buf is not None and not direct else None,
                None, None, None)


class ConvTranspose2d(nn.Module):
    """NHWC bf16 transposed conv.  Weight [C_in, R, S, C_out] fp32
    (= the equivalent regular conv's [K][R][S][C]).
    Trunk layers need C_in % 64 == 0 and C_out % 64 == 0; a small-C output
    head (e.g. RGB, C_out <= 8 with C_in == 64) uses the direct edge
    kernels."""

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding: int = 0, input_grad: bool = True,
                 feeds_bn: bool = False):
        super().__init__()
        self.stride = stride
        self.padding = padding
        self.input_grad = input_grad
        # when the conv output goes straight into a training BatchNorm2d,
        # the conv epilogue emits the BN sum/sumsq partials for free
        self.feeds_bn = feeds_bn
        k = kernel_size
        self.weight = nn.Parameter(
            torch.empty(in_channels, k, k, out_channels))
        fan_out = out_channels * k * k
        nn.init.normal_(self.weight, std=math.sqrt(2.0 / fan_out))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _ConvTransposeFn.apply(x, self.weight, self.stride,
                                      self.padding,
                                      self.input_grad and x.requires_grad)


# Above this many FLOPs per product the fc goes to rocBLAS (the guide's
# rule for PLAIN library GEMMs — no fusion opportunity here); below it the
# one-kernel-per-product native path wins on launch overhead.  The ResNet-50
# head (64x2048 -> 1000, 262 MFLOP) sits far above; the example MLPs far
# below.  Measured: the naive fc kernels cost 435 us/step on the R50 head
# vs ~15 us through rocBLAS (profiles/r02h).
_LINEAR_GEMM_CUTOFF = int(os.environ.get("FLASHY_LINEAR_CUTOFF", 1 << 23))


class _LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor,
                b: tp.Optional[torch.Tensor]):
        B, I = x.shape
        O = w.shape[0]
        big = 2 * B * I * O >= _LINEAR_GEMM_CUTOFF and x.dtype == torch.float32
        if big:
            x = x.contiguous()
            y = torch.addmm(b, x, w.t()) if b is not None else x.mm(w.t())
        else:
            y = x.new_empty((B, O))
            ops.linear_fwd(x.contiguous(), w, b, y)
        ctx.save_for_backward(x)
        ctx.refs = (w, b)
        ctx.big = big
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (x,) = ctx.saved_tensors
        w, b = ctx.refs
        dy = dy.contiguous()
        dw = db = None
        need_w = ctx.needs_input_grad[1]
        need_b = b is not None and ctx.needs_input_grad[2]
        if ctx.big:
            if need_w:
                dw, w_direct = _grad_target(w)
                dw.addmm_(dy.t(), x)  # += dy^T @ x (rocBLAS, fp32)
                dw = None if w_direct else dw
            if need_b:
                db, b_direct = _grad_target(b)
                db.add_(dy.sum(0))
                db = None if b_direct else db
        elif need_w or need_b:
            # one kernel produces both; frozen side goes to a discarded temp
            dw, w_direct = _grad_target(w) if need_w \
                else (torch.empty_like(w), True)
            db, b_direct = (_grad_target(b) if need_b else (None, True))
            ops.linear_dw(x, dy, dw, db)
            dw = dw if need_w and not w_direct else None
            db = db if need_b and not b_direct else None
        dx = None
        if ctx.needs_input_grad[0]:
            if ctx.big:
                dx = dy.mm(w)
            else:
                dx = x.new_empty(x.shape)
                ops.linear_dx(dy, w, dx)
        return dx, dw, db


class Linear(nn.Module):
    """fp32 fully-connected layer.  Tiny (launch-bound) shapes run the
    native fc kernels, one kernel per product; large heads (e.g. ResNet-50's
    2048->1000) go through rocBLAS — a plain dense GEMM with no fusion
    opportunity belongs on the library path (guide rule; 435 us -> ~15 us
    per step measured).  Weight [O, I], bias [O]; CPU falls back to
    torch.nn.functional.linear."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        bound = 1.0 / math.sqrt(in_features)
        nn.init.uniform_(self.weight, -bound, bound)
        self.bias: tp.Optional[nn.Parameter] = None
        if bias:
            self.bias = nn.Parameter(torch.empty(out_features))
            nn.init.uniform_(self.bias, -bound, bound)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.device.type != "cuda":
            return torch.nn.functional.linear(x, self.weight, self.bias)
        return _LinearFn.apply(x, self.weight, self.bias)


class _MaxPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, kernel: int, stride: int, pad: int):
        N, H, W, C = x.shape
        Ho = (H + 2 * pad - kernel) // stride + 1
        Wo = (W + 2 * pad - kernel) // stride + 1
        d = ops.ConvDims(N, H, W, C, C, kernel, kernel, Ho, Wo, stride, pad)
        y = x.new_empty((N, Ho, Wo, C))
        argmax = torch.empty(N * Ho * Wo * C, dtype=torch.uint8, device=x.device)
        ops.maxpool_fwd(x, y, argmax, d)
        ctx.save_for_backward(argmax)
        ctx.dims = d
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        (argmax,) = ctx.saved_tensors
        d: ops.ConvDims = ctx.dims
        dx = dy.new_empty((d.N, d.H, d.W, d.C))
        ops.maxpool_bwd(dy.contiguous(), argmax, dx, d)
        return dx, None, None, None


class MaxPool2d(nn.Module):
    """NHWC bf16 max pool with deterministic gather-based backward."""

    def __init__(self, kernel_size: int, stride: int, padding: int = 0):
        super().__init__()
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return _MaxPoolFn.apply(x, self.kernel_size, self.stride, self.padding)


class BatchNorm2d(nn.Module):
    """NHWC training BatchNorm with optional fused residual-add + ReLU."""

    def __init__(self, num_features: int, eps: float = 1e-5, momentum: float = 0.1):
        super().__init__()
        self.num_features = num_features
        self.eps = eps
        self.momentum = momentum
        self.weight = nn.Parameter(torch.ones(num_features))
        self.bias = nn.Parameter(torch.zeros(num_features))
        self.register_buffer("running_mean", torch.zeros(num_features))
        self.register_buffer("running_var", torch.ones(num_features))

    def forward(self, x: torch.Tensor,
                res: tp.Optional[torch.Tensor] = None,
                relu: bool = False, slope: float = 0.0) -> torch.Tensor:
        """``relu=True, slope=s`` applies LeakyReLU(s) (s=0: plain ReLU)."""
        return _BnFn.apply(x, self.weight, self.bias, res, relu, slope, self)
