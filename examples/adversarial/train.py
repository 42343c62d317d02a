# Copyright (c) Flashy-AMD authors.
"""DCGAN-style adversarial workload on 64x64 synthetic images
(BASELINE.json config 4): generator trained against AdversarialLoss's
discriminator with the eager (overlapped) distributed sync."""
from __future__ import annotations

from pathlib import Path

import torch

import flashy_amd
from flashy_amd import BaseSolver, Formatter, distrib
from flashy_amd.adversarial import AdversarialLoss
from flashy_amd.models import (DCGANDiscriminator, DCGANGenerator,
                               NativeDCGANDiscriminator, NativeDCGANGenerator)
from flashy_amd.optim import FusedAdam
from flashy_amd.utils import averager
from flashy_amd import xp as fxp

main = fxp.entry_point("examples.adversarial", Path(__file__).parent / "conf")


class Solver(BaseSolver):
    def __init__(self, cfg):
        super().__init__()
        self.cfg = cfg
        self.device = torch.device("cuda" if torch.cuda.is_available() else "cpu") \
            if cfg.device == "auto" else torch.device(cfg.device)
        self.native = self.device.type == "cuda" and cfg.get("native", True)
        if self.native:  # gfx950 kernel G/D (trunk needs nz % 64 == 0)
            self.nz = max(64, (cfg.nz + 63) // 64 * 64)
            self.generator = NativeDCGANGenerator(self.nz, cfg.ngf).to(self.device)
            distrib.broadcast_model(self.generator)
            self.g_optim = FusedAdam(self.generator.parameters(), lr=cfg.lr,
                                     betas=(cfg.beta1, 0.999))
            disc = NativeDCGANDiscriminator(cfg.ndf).to(self.device)
            d_optim = FusedAdam(disc.parameters(), lr=cfg.lr,
                                betas=(cfg.beta1, 0.999))
        else:
            self.nz = cfg.nz
            self.generator = DCGANGenerator(self.nz, cfg.ngf).to(self.device)
            distrib.broadcast_model(self.generator)
            self.g_optim = torch.optim.Adam(self.generator.parameters(), lr=cfg.lr,
                                            betas=(cfg.beta1, 0.999))
            disc = DCGANDiscriminator(cfg.ndf).to(self.device)
            d_optim = torch.optim.Adam(disc.parameters(), lr=cfg.lr,
                                       betas=(cfg.beta1, 0.999))
        self.adv = AdversarialLoss(disc, d_optim)
        self.register_stateful("generator", "g_optim", "adv")

    def get_formatter(self, stage_name):
        return Formatter({"g_loss": ".4f", "d_loss": ".4f"})

    def _real_batch(self):
        # synthetic "real" images: smooth blobs, deterministic per draw
        x = torch.randn(self.cfg.batch_size, 3, 64, 64, device=self.device)
        return torch.tanh(torch.nn.functional.avg_pool2d(x, 5, 1, 2) * 3)

    def train_stage(self):
        avg = averager()
        lp = self.log_progress("train", range(self.cfg.steps_per_epoch), updates=5)
        for _ in lp:
            real = self._real_batch()
            z = torch.randn(self.cfg.batch_size, self.nz, 1, 1, device=self.device)
            fake = self.generator(z)
            d_loss = self.adv.train_adv(fake, real)
            g_loss = self.adv(fake)
            self.g_optim.zero_grad()
            if hasattr(self.g_optim, "grad_buffers"):
                g_loss.backward()
                distrib.sync_flat_gradients(self.g_optim)
            else:
                with distrib.eager_sync_model(self.generator):
                    g_loss.backward()
            self.g_optim.step()
            metrics = avg({"g_loss": g_loss.item(), "d_loss": d_loss.item()})
            lp.update(**metrics)
        return distrib.average_metrics(metrics, self.cfg.steps_per_epoch)

    def run(self):
        if self.restore():
            for opt in (self.g_optim, self.adv.optimizer):
                if hasattr(opt, "refresh_bf16"):
                    opt.refresh_bf16()
        self.log_hyperparams(self.cfg)
        for epoch in range(self.epoch, self.cfg.epochs + 1):
            self.run_stage("train", self.train_stage)
            if distrib.is_rank_zero():
                with torch.no_grad():
                    z = torch.randn(8, self.nz, 1, 1, device=self.device)
                    sample = self.generator(z).add(1).div(2).clamp(0, 1)
                self.log_image("train", "samples", sample.cpu())
            self.commit()


@main.bind
def run(cfg):
    flashy_amd.setup_logging()
    distrib.init()
    torch.manual_seed(cfg.seed + distrib.rank())
    Solver(cfg).run()


if __name__ == "__main__":
    main()
