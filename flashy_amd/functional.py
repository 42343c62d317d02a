# Copyright (c) Flashy-AMD authors.
"""Autograd-integrated functional ops backed by the gfx950 kernels.

Each loss computes its input gradient during the forward pass (training
always runs backward, so fusing saves a full re-read of the logits), wrapped
in ``torch.autograd.Function`` so ``loss.backward()`` works as usual.
On CPU they fall back to the torch implementations (tests/CI); on CUDA the
native extension is required.
"""
from __future__ import annotations

import torch

from . import ops


class _CrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        B, C = logits.shape
        dlogits = torch.empty_like(logits)
        loss = torch.zeros((), dtype=torch.float32, device=logits.device)
        ops.cross_entropy_fwd_bwd(logits.contiguous(), target, dlogits, loss,
                                  loss_scale=1.0 / B, grad_scale=1.0 / B)
        ctx.save_for_backward(dlogits)
        return loss

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (dlogits,) = ctx.saved_tensors
        return dlogits * grad_out, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean cross-entropy over [B, C] logits (fp32 or bf16) and int64 targets.

    GPU: one fused kernel producing loss + dlogits.  CPU: torch fallback.
    """
    if logits.device.type != "cuda":
        return torch.nn.functional.cross_entropy(logits, target)
    return _CrossEntropy.apply(logits, target)


class _BCEWithLogits(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, target_value: float) -> torch.Tensor:
        n = x.numel()
        dx = torch.empty_like(x)
        loss = torch.zeros((), dtype=torch.float32, device=x.device)
        ops.bce_logits_fwd_bwd(x.contiguous(), dx, loss, target_value,
                               loss_scale=1.0 / n, grad_scale=1.0 / n)
        ctx.save_for_backward(dx)
        return loss

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (dx,) = ctx.saved_tensors
        return dx * grad_out, None


def bce_with_logits_const(x: torch.Tensor, target_value: float) -> torch.Tensor:
    """Mean BCE-with-logits against a constant target (GAN labels)."""
    if x.device.type != "cuda":
        target = torch.full_like(x, target_value)
        return torch.nn.functional.binary_cross_entropy_with_logits(x, target)
    return _BCEWithLogits.apply(x, target_value)


class _MSE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, t: torch.Tensor) -> torch.Tensor:
        n = x.numel()
        dx = torch.empty_like(x)
        loss = torch.zeros((), dtype=torch.float32, device=x.device)
        ops.mse_fwd_bwd(x.contiguous(), t.contiguous(), dx, loss,
                        loss_scale=1.0 / n, grad_scale=1.0 / n)
        ctx.save_for_backward(dx)
        return loss

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (dx,) = ctx.saved_tensors
        return dx * grad_out, None


def mse_loss(x: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """Mean squared error (fused fwd+grad kernel on GPU; torch on CPU).

    Replaces F.mse_loss of the teacher-student workload
    (/root/reference/tests/dummy/train.py:93)."""
    if x.device.type != "cuda":
        return torch.nn.functional.mse_loss(x, target)
    return _MSE.apply(x, target.detach())


def accuracy(logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """mean(argmax(logits, 1) == target) as a device scalar — one kernel on
    GPU (vs the argmax+eq+mean 3-launch chain), torch ops on CPU."""
    if logits.device.type != "cuda":
        return (logits.argmax(1) == target).float().mean()
    B = logits.shape[0]
    out = torch.zeros((), dtype=torch.float32, device=logits.device)
    ops.accuracy_count(logits.detach().contiguous(), target, out)
    return out / B
