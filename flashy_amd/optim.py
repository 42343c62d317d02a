# Copyright (c) Flashy-AMD authors.
"""Flat-buffer fused optimizers for MI355X.

Design: at construction ALL trainable parameters of one (device, dtype)
group are packed into a single contiguous fp32 buffer; each ``param.data``
is re-pointed to a view of that buffer, and ``param.grad`` to a view of a
matching flat gradient buffer.  Consequences:

* ``step()`` is ONE HIP kernel over the flat buffers (float4-vectorized,
  grid-strided — flashy_amd/ops/csrc/fused_optim.hip) instead of torch's
  per-parameter loop: the whole ResNet update is a single launch.
* ``zero_grad()`` is one memset.
* Data-parallel gradient sync is ONE RCCL all-reduce of the flat gradient
  buffer — the maximal bucket for the xGMI ring (see
  :func:`flashy_amd.distrib.sync_flat_gradients`).

Replaces the reference's use of torch.optim.SGD/Adam in its workloads
(SURVEY.md §2.10 "SGD step / Adam step (fused over all params)").

On CUDA devices the native extension is REQUIRED (loud failure otherwise);
on CPU the same classes run a torch fallback with identical numerics, so
every test runs in CI.
"""
from __future__ import annotations

import math
import typing as tp

import torch

from . import ops

ParamsT = tp.Iterable[torch.nn.Parameter]


class _Group:
    """One (device, dtype) flat group."""

    def __init__(self, params: tp.List[torch.nn.Parameter],
                 bf16_mirror: bool = False):
        self.params = params
        self.device = params[0].device
        self.dtype = params[0].dtype
        total = sum(p.numel() for p in params)
        self.flat_p = torch.empty(total, device=self.device, dtype=self.dtype)
        offset = 0
        views = []
        for p in params:
            n = p.numel()
            view = self.flat_p[offset:offset + n].view_as(p)
            view.copy_(p.data)
            views.append(view)
            offset += n
        # re-point after all copies (a param could alias another's storage)
        for p, view in zip(params, views):
            p.data = view
        self.flat_g = torch.zeros_like(self.flat_p)
        offset = 0
        for p in params:
            n = p.numel()
            p.grad = self.flat_g[offset:offset + n].view_as(p)
            offset += n
        # optional bf16 mirror of the packed params, refreshed by the fused
        # optimizer kernel in the same pass as the update; modules read the
        # per-param view via `param._bf16_mirror` (flashy_amd/nn.py) so the
        # step needs zero weight-cast kernels.
        self.flat_p16: tp.Optional[torch.Tensor] = None
        if bf16_mirror:
            self.flat_p16 = self.flat_p.to(torch.bfloat16)
            offset = 0
            for p in params:
                n = p.numel()
                p._bf16_mirror = self.flat_p16[offset:offset + n].view_as(p)
                offset += n

    def refresh_bf16(self) -> None:
        if self.flat_p16 is not None:
            self.flat_p16.copy_(self.flat_p)

    def buffers_like(self) -> torch.Tensor:
        return torch.zeros_like(self.flat_p)


class FlatOptimizer:
    """Base: flat parameter/grad packing + state dict plumbing."""

    def __init__(self, params: ParamsT, defaults: tp.Dict[str, tp.Any],
                 bf16_mirror: bool = False):
        params = [p for p in params if p.requires_grad]
        if not params:
            raise ValueError("no trainable parameters")
        for p in params:
            if p.dtype != torch.float32:
                raise TypeError(
                    f"FlatOptimizer packs fp32 master params, got {p.dtype}")
        by_key: tp.Dict[tp.Any, tp.List[torch.nn.Parameter]] = {}
        for p in params:
            by_key.setdefault((p.device, p.dtype), []).append(p)
        self.groups = [_Group(ps, bf16_mirror) for ps in by_key.values()]
        self.defaults = dict(defaults)
        self.step_count = 0
        self._use_hip = any(g.device.type == "cuda" for g in self.groups)
        if self._use_hip:
            ops.require()  # fail loudly now, not at the first step

    # -- torch-optimizer-compatible surface ---------------------------------
    def zero_grad(self, set_to_none: bool = False) -> None:
        # grads are views into the flat buffer: never set to None
        del set_to_none
        for g in self.groups:
            g.flat_g.zero_()

    @property
    def grad_buffers(self) -> tp.List[torch.Tensor]:
        """Flat gradient buffers (one per group) — the DP sync payload."""
        return [g.flat_g for g in self.groups]

    @property
    def param_buffers(self) -> tp.List[torch.Tensor]:
        return [g.flat_p for g in self.groups]

    def refresh_bf16(self) -> None:
        """Re-sync the bf16 mirrors after any out-of-band param mutation
        (checkpoint restore, broadcast_model)."""
        for g in self.groups:
            g.refresh_bf16()

    def step(self, closure: tp.Optional[tp.Callable] = None) -> None:
        loss = closure() if closure is not None else None
        self.step_count += 1
        for g in self.groups:
            self._step_group(g)
        return loss

    def _step_group(self, group: _Group) -> None:
        raise NotImplementedError

    # -- checkpointing -------------------------------------------------------
    def _extra_state(self) -> tp.Dict[str, tp.Any]:
        return {}

    def _load_extra_state(self, state: tp.Dict[str, tp.Any]) -> None:
        del state

    def state_dict(self) -> tp.Dict[str, tp.Any]:
        return {
            "defaults": dict(self.defaults),
            "step_count": self.step_count,
            "extra": self._extra_state(),
        }

    def load_state_dict(self, state: tp.Mapping[str, tp.Any]) -> None:
        self.defaults.update(state.get("defaults", {}))
        self.step_count = state.get("step_count", 0)
        self._load_extra_state(state.get("extra", {}))
        # model params are usually restored in the same breath
        # (StateManager restores the model first); re-sync the bf16 mirrors
        self.refresh_bf16()


class FusedSGD(FlatOptimizer):
    """torch.optim.SGD semantics (momentum / weight decay / nesterov) as one
    fused HIP kernel over the flat buffers."""

    def __init__(self, params: ParamsT, lr: float, momentum: float = 0.0,
                 weight_decay: float = 0.0, nesterov: bool = False,
                 bf16_mirror: bool = False):
        super().__init__(params, dict(lr=lr, momentum=momentum,
                                      weight_decay=weight_decay,
                                      nesterov=nesterov), bf16_mirror)
        self._momentum_buffers = [
            g.buffers_like() if momentum != 0 else None for g in self.groups]

    def _step_group(self, group: _Group) -> None:
        d = self.defaults
        m = self._momentum_buffers[self.groups.index(group)]
        if group.device.type == "cuda":
            ops.fused_sgd(group.flat_p, group.flat_g, m, d["lr"],
                          d["momentum"], d["weight_decay"],
                          nesterov=d["nesterov"], p_bf16=group.flat_p16)
            return
        # CPU fallback, identical math
        with torch.no_grad():
            g = group.flat_g
            if d["weight_decay"] != 0:
                g = g.add(group.flat_p, alpha=d["weight_decay"])
            if d["momentum"] != 0:
                m.mul_(d["momentum"]).add_(g)
                g = g.add(m, alpha=d["momentum"]) if d["nesterov"] else m
            group.flat_p.add_(g, alpha=-d["lr"])
            group.refresh_bf16()

    def _extra_state(self):
        return {"momentum_buffers": self._momentum_buffers}

    def _load_extra_state(self, state):
        saved = state.get("momentum_buffers")
        if saved:
            for mine, got in zip(self._momentum_buffers, saved):
                if mine is not None and got is not None:
                    mine.copy_(got.to(mine.device))


class FusedAdam(FlatOptimizer):
    """torch.optim.Adam/AdamW semantics as one fused HIP kernel per group."""

    def __init__(self, params: ParamsT, lr: float = 1e-3,
                 betas: tp.Tuple[float, float] = (0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0,
                 adamw: bool = False, bf16_mirror: bool = False):
        super().__init__(params, dict(lr=lr, beta1=betas[0], beta2=betas[1],
                                      eps=eps, weight_decay=weight_decay,
                                      adamw=adamw), bf16_mirror)
        self._exp_avg = [g.buffers_like() for g in self.groups]
        self._exp_avg_sq = [g.buffers_like() for g in self.groups]
        # device-side step counter: the kernel reads it for bias correction,
        # so the whole step is HIP-graph-replay-safe (a captured host step
        # count would freeze).  Starts at 1 = the value used by step #1.
        self._step_dev = {}
        for g in self.groups:
            if g.device.type == "cuda" and g.device not in self._step_dev:
                self._step_dev[g.device] = torch.ones(
                    1, dtype=torch.int64, device=g.device)

    def _step_group(self, group: _Group) -> None:
        d = self.defaults
        i = self.groups.index(group)
        m, v = self._exp_avg[i], self._exp_avg_sq[i]
        if group.device.type == "cuda":
            ops.fused_adam(group.flat_p, group.flat_g, m, v, d["lr"],
                           d["beta1"], d["beta2"], d["eps"],
                           d["weight_decay"], self.step_count,
                           adamw=d["adamw"], p_bf16=group.flat_p16,
                           step_dev=self._step_dev[group.device])
            return
        with torch.no_grad():
            g = group.flat_g
            p = group.flat_p
            if d["adamw"]:
                p.mul_(1 - d["lr"] * d["weight_decay"])
            elif d["weight_decay"] != 0:
                g = g.add(p, alpha=d["weight_decay"])
            m.mul_(d["beta1"]).add_(g, alpha=1 - d["beta1"])
            v.mul_(d["beta2"]).addcmul_(g, g, value=1 - d["beta2"])
            bc1 = 1 - d["beta1"] ** self.step_count
            bc2 = 1 - d["beta2"] ** self.step_count
            denom = (v / bc2).sqrt_().add_(d["eps"])
            p.addcdiv_(m / bc1, denom, value=-d["lr"])
            group.refresh_bf16()

    def step(self, closure=None):
        out = super().step(closure)
        for dev_step in self._step_dev.values():
            ops.adam_step_inc(dev_step)
        return out

    def state_dict(self):
        if self._step_dev:
            # The device counter is authoritative: under HIP-graph capture
            # step() runs only at capture time, so the host count freezes
            # while the device counter advances per replay.  It holds the
            # NEXT step number (starts at 1 before step #1).
            self.step_count = max(
                int(s.item()) for s in self._step_dev.values()) - 1
        return super().state_dict()

    def _extra_state(self):
        return {"exp_avg": self._exp_avg, "exp_avg_sq": self._exp_avg_sq}

    def _load_extra_state(self, state):
        for mine, got in zip(self._exp_avg, state.get("exp_avg", [])):
            mine.copy_(got.to(mine.device))
        for mine, got in zip(self._exp_avg_sq, state.get("exp_avg_sq", [])):
            mine.copy_(got.to(mine.device))
        for dev_step in self._step_dev.values():
            dev_step.fill_(self.step_count + 1)
