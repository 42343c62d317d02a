# Copyright (c) Flashy-AMD authors.
"""Self-contained CPU/gloo fixture workload: teacher-student MSE plus an
adversarial loss over synthetic noise (the "fake backend" — no data download,
no GPU).  Used by the integration tests for the run -> kill -> resume oracle.

Parity: reference tests/dummy/train.py (Network/NoiseDataset/Solver with
``stop_at`` early exit; 4 registered stateful objects incl. the adversarial
wrapper).
"""
from __future__ import annotations

from pathlib import Path

import torch
from torch import nn

import flashy_amd
from flashy_amd import BaseSolver, Formatter, distrib
from flashy_amd.adversarial import AdversarialLoss
from flashy_amd.utils import averager
from flashy_amd import xp as fxp

main = fxp.entry_point("tests.dummy", Path(__file__).parent / "conf")


class Network(nn.Module):
    def __init__(self, dim: int):
        super().__init__()
        self.net = nn.Sequential(nn.Linear(dim, dim), nn.ReLU(), nn.Linear(dim, dim))

    def forward(self, x):
        return self.net(x)


class NoiseDataset:
    """Deterministic synthetic dataset: sample i is seeded noise."""

    def __init__(self, size: int, dim: int):
        self.size = size
        self.dim = dim

    def __len__(self):
        return self.size

    def __getitem__(self, index: int):
        g = torch.Generator().manual_seed(index)
        return torch.randn(self.dim, generator=g)


class Solver(BaseSolver):
    def __init__(self, cfg):
        super().__init__()
        self.cfg = cfg
        self.student = Network(cfg.dim)
        self.teacher = Network(cfg.dim)
        distrib.broadcast_model(self.student)
        distrib.broadcast_model(self.teacher)
        self.optim = torch.optim.Adam(self.student.parameters(), lr=cfg.lr)
        adversary = nn.Sequential(nn.Linear(cfg.dim, cfg.dim), nn.LeakyReLU(0.2),
                                  nn.Linear(cfg.dim, 1))
        adv_optim = torch.optim.Adam(adversary.parameters(), lr=cfg.lr)
        self.adv = AdversarialLoss(adversary, adv_optim)
        self.register_stateful("student", "teacher", "optim", "adv")
        dataset = NoiseDataset(cfg.dataset_size, cfg.dim)
        self.loader = distrib.loader(dataset, batch_size=cfg.batch_size, shuffle=True)

    def get_formatter(self, stage_name):
        return Formatter({"loss": ".6f", "adv": ".6f"}, exclude_keys=["*"])

    def train_stage(self):
        avg = averager()
        lp = self.log_progress("train", self.loader, updates=2)
        for batch in lp:
            target = self.teacher(batch).detach()
            est = self.student(batch)
            loss = torch.nn.functional.mse_loss(est, target)
            adv_loss = self.adv(est)
            self.optim.zero_grad()
            with distrib.eager_sync_model(self.student):
                (loss + 0.1 * adv_loss).backward()
            self.optim.step()
            d_loss = self.adv.train_adv(est.detach(), target)
            metrics = avg({"loss": loss.item(), "adv": d_loss.item()})
            lp.update(**metrics)
        return distrib.average_metrics(metrics, len(self.loader))

    def run(self):
        self.restore()
        for epoch in range(self.epoch, self.cfg.epochs + 1):
            if self.cfg.stop_at is not None and epoch > self.cfg.stop_at:
                break
            self.run_stage("train", self.train_stage)
            self.commit()


@main.bind
def run(cfg):
    flashy_amd.setup_logging()
    distrib.init(cfg.distrib_backend)
    torch.manual_seed(cfg.seed)
    Solver(cfg).run()


if __name__ == "__main__":
    main()
