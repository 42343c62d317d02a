# Copyright (c) Flashy-AMD authors.
"""CIFAR-10-shaped ResNet-18 solver: train/valid stages, DP via the bucketed
``sync_model`` path, accuracy formatting, image logging.

Parity: reference examples/cifar/solver.py + train.py (torchvision download
replaced by a deterministic synthetic dataset — this environment has no
network; shapes and the training loop structure are the same).
"""
from __future__ import annotations

import math

import torch
from torch.nn import functional as F

from flashy_amd import BaseSolver, Formatter, distrib
from flashy_amd.functional import accuracy, cross_entropy
from flashy_amd.graph import CapturedStep
from flashy_amd.models.resnet_native import NativeResNet
from flashy_amd.utils import averager


class SyntheticCIFAR:
    """Deterministic CIFAR-10-shaped dataset: sample i is seeded noise with a
    class-dependent mean, so accuracy is learnable above chance."""

    def __init__(self, size: int, num_classes: int = 10, train: bool = True):
        self.size = size
        self.num_classes = num_classes
        self.offset = 0 if train else 1 << 24

    def __len__(self):
        return self.size

    def __getitem__(self, index: int):
        g = torch.Generator().manual_seed(index + self.offset)
        label = int(torch.randint(self.num_classes, (1,), generator=g))
        img = torch.randn(3, 32, 32, generator=g) * 0.5 + label * 0.1
        return img, label


class Solver(BaseSolver):
    def __init__(self, cfg, model, loaders, optim):
        super().__init__()
        self.cfg = cfg
        self.async_checkpoint = bool(cfg.get('async_checkpoint', False))
        self.model = model
        self.loaders = loaders
        self.optim = optim
        self.device = next(model.parameters()).device
        self.native = isinstance(model, NativeResNet)
        self.autocast = (self.device.type == "cuda" and cfg.dtype == "bf16"
                         and not self.native)
        # HIP-graph whole-step capture: native path, single process only
        self.use_graph = (cfg.get("use_graph", True) and self.native
                          and not distrib.is_distributed())
        self._graph = None
        self.register_stateful("model", "optim")

    def get_formatter(self, stage_name):
        return Formatter({"acc": ".1%", "loss": ".5f"})

    def _capture(self, shape):
        self._static_img = torch.zeros(shape, device=self.device)
        self._static_label = torch.zeros(shape[0], dtype=torch.long,
                                         device=self.device)

        def step():
            self.optim.zero_grad(set_to_none=False)
            est = self.model(self._static_img)
            loss = cross_entropy(est, self._static_label)
            loss.backward()
            self.optim.step()
            return loss, est

        self._graph = CapturedStep(step, warmup=3).capture()

    def _graphed_step(self, img, label):
        """One training step as a single hipGraph replay (static buffers)."""
        if self._graph is None:
            self._capture(img.shape)
        self._static_img.copy_(img, non_blocking=True)
        self._static_label.copy_(label, non_blocking=True)
        loss, est = self._graph()
        acc = accuracy(est, self._static_label)
        return loss, acc

    def _step(self, img, label, train: bool):
        if (train and self.use_graph and self.device.type == "cuda"
                and img.shape[0] == self.cfg.batch_size):
            return self._graphed_step(img, label)
        img = img.to(self.device, non_blocking=True)
        label = label.to(self.device, non_blocking=True)
        with torch.autocast("cuda", torch.bfloat16, enabled=self.autocast):
            est = self.model(img)
        if self.native and train:
            loss = cross_entropy(est, label)  # fused fwd+grad kernel
        else:
            loss = F.cross_entropy(est, label)
        acc = accuracy(est, label)
        if train:
            self.optim.zero_grad()
            loss.backward()
            if hasattr(self.optim, "grad_buffers"):
                distrib.sync_flat_gradients(self.optim)
            else:
                distrib.sync_model(self.model)
            self.optim.step()
        return loss, acc

    def do_train_valid(self, train: bool):
        stage = "train" if train else "valid"
        loader = self.loaders[stage]
        self.model.train(train)
        avg = averager()
        lp = self.log_progress(stage, loader, updates=5)
        with torch.set_grad_enabled(train):
            for img, label in lp:
                loss, acc = self._step(img, label, train)
                metrics = avg({"loss": loss.item(), "acc": acc.item()})
                lp.update(**metrics)
        if not math.isfinite(metrics["loss"]):
            # a NaN must halt the run loudly, not be silently averaged —
            # especially with graph capture in the loop
            raise RuntimeError(
                f"non-finite {stage} loss at epoch {self.epoch}: {metrics}")
        return distrib.average_metrics(metrics, len(loader))

    def run(self):
        if self.restore() and hasattr(self.optim, "refresh_bf16"):
            # restored fp32 params -> re-sync the bf16 weight mirrors
            self.optim.refresh_bf16()
        self.log_hyperparams(self.cfg)
        for epoch in range(self.epoch, self.cfg.epochs + 1):
            self.run_stage("train", self.do_train_valid, True)
            self.run_stage("valid", self.do_train_valid, False)
            self.commit()
        self.finalize_checkpoint()   # join an in-flight async write, if any
