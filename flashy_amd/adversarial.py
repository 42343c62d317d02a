# Copyright (c) Flashy-AMD authors.
"""Adversarial (GAN) loss wrapper owning the discriminator update.

Capability parity with the reference's ``flashy/adversarial.py:22-89``:
encapsulates the discriminator ("adversary"), its optimizer and the loss
function; ``train_adv(fake, real)`` runs one discriminator step with the
eager (overlapped) distributed gradient sync; calling the module computes
the generator loss with the discriminator temporarily frozen.

Convention (kept): a HIGH discriminator logit means FAKE — the
discriminator is trained towards ``D(fake)=1, D(real)=0`` and the generator
minimizes ``loss(D(fake), 0)``.

Checkpointing: the optimizer state is embedded inside the module's own state
dict, so ``register_stateful('adv')`` captures model + optimizer together.
"""
from __future__ import annotations

import typing as tp

import torch
from torch import nn
from torch.nn import functional as F

from . import distrib
from .utils import readonly

LossFn = tp.Callable[[torch.Tensor, torch.Tensor], torch.Tensor]
# drop-in alias kept from the reference API (flashy/adversarial.py)
LossType = LossFn


def _bce_logits(logits: torch.Tensor, target_is_fake: float) -> torch.Tensor:
    target = torch.full_like(logits, target_is_fake)
    return F.binary_cross_entropy_with_logits(logits, target)


class AdversarialLoss(nn.Module):
    """Owns the adversary + its optimizer; broadcast-synced at construction.

    Args:
        adversary: discriminator module mapping samples -> logits.
        optimizer: optimizer over ``adversary.parameters()``.
        loss: callable ``(logits, target_value)`` -> scalar loss; defaults to
            BCE-with-logits against a constant target.
    """

    def __init__(self, adversary: nn.Module, optimizer: torch.optim.Optimizer,
                 loss: tp.Optional[LossFn] = None):
        super().__init__()
        self.adversary = adversary
        self.optimizer = optimizer
        self.loss = loss or _bce_logits
        # all ranks start from rank-0's discriminator weights
        distrib.broadcast_model(self.adversary)

    # -- checkpoint embedding (optimizer rides inside the module state) ----
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        super()._save_to_state_dict(destination, prefix, keep_vars)
        destination[prefix + "optimizer"] = self.optimizer.state_dict()

    def _load_from_state_dict(self, state_dict, prefix, *args, **kwargs):
        opt_state = state_dict.pop(prefix + "optimizer", None)
        if opt_state is not None:
            self.optimizer.load_state_dict(opt_state)
        super()._load_from_state_dict(state_dict, prefix, *args, **kwargs)

    # -- discriminator step -------------------------------------------------
    def train_adv(self, fake: torch.Tensor, real: torch.Tensor) -> torch.Tensor:
        """One discriminator update on detached samples; returns the D loss."""
        loss = self.loss(self.adversary(fake.detach()), 1.0) + \
            self.loss(self.adversary(real.detach()), 0.0)
        self.optimizer.zero_grad()
        if hasattr(self.optimizer, "grad_buffers"):
            # flat optimizer (native kernels write grads straight into the
            # flat buffer, bypassing autograd hooks): one all-reduce per group
            loss.backward()
            distrib.sync_flat_gradients(self.optimizer)
            if distrib.is_distributed():
                distrib.average_tensors(
                    [b for b in self.adversary.buffers() if b.is_floating_point()])
        else:
            with distrib.eager_sync_model(self.adversary):
                loss.backward()
        self.optimizer.step()
        return loss.detach()

    # -- generator loss ------------------------------------------------------
    def forward(self, fake: torch.Tensor) -> torch.Tensor:
        """Generator loss: fool the discriminator towards the REAL label,
        with D's parameters frozen so G's backward cannot touch them."""
        with readonly(self.adversary):
            return self.loss(self.adversary(fake), 0.0)
