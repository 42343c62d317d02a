# Copyright (c) Flashy-AMD authors.
"""CIFAR example entry point (parity: reference examples/cifar/train.py)."""
from __future__ import annotations

from pathlib import Path

import torch

import flashy_amd
from flashy_amd import distrib
from flashy_amd.models import native_resnet18, resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd import xp as fxp

from .solver import Solver, SyntheticCIFAR

main = fxp.entry_point("examples.cifar", Path(__file__).parent / "conf")


def get_solver(cfg):
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu") \
        if cfg.device == "auto" else torch.device(cfg.device)
    native = device.type == "cuda" and cfg.get("native", True)
    if native:  # the gfx950 NHWC kernel path + flat fused optimizer
        model = native_resnet18(cfg.num_classes).to(device)
        distrib.broadcast_model(model)
        optim = FusedSGD(model.parameters(), lr=cfg.lr, momentum=cfg.momentum,
                         weight_decay=cfg.weight_decay, bf16_mirror=True)
        model.enable_wt_cache()
        return Solver(cfg, model, _loaders(cfg), optim)
    model = resnet18(num_classes=cfg.num_classes, small_input=True).to(device)
    distrib.broadcast_model(model)
    optim = torch.optim.SGD(model.parameters(), lr=cfg.lr,
                            momentum=cfg.momentum, weight_decay=cfg.weight_decay)
    return Solver(cfg, model, _loaders(cfg), optim)


def _loaders(cfg):
    return {
        "train": distrib.loader(SyntheticCIFAR(cfg.dataset_size, cfg.num_classes),
                                batch_size=cfg.batch_size, shuffle=True),
        "valid": distrib.loader(SyntheticCIFAR(cfg.valid_size, cfg.num_classes,
                                               train=False),
                                batch_size=cfg.batch_size, shuffle=False),
    }


def get_solver_from_sig(sig: str):
    """Notebook workflow: rebuild + restore the solver of an existing run."""
    xp = main.get_xp_from_sig(sig)
    xp.enter()
    solver = get_solver(xp.cfg)
    solver.restore()
    return solver


@main.bind
def run(cfg):
    flashy_amd.setup_logging()
    distrib.init()
    torch.manual_seed(cfg.seed)
    get_solver(cfg).run()


if __name__ == "__main__":
    main()
