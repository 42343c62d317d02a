# Copyright (c) Flashy-AMD authors.
"""Diagnose the graph-captured-step NaN (GPUTEST_r01: torch resnet18 +
autocast + FusedSGD under CapturedStep goes NaN after 2 replays).

Runs a matrix of variants of the failing test and, for the baseline repro,
checks after every replay which tensor family goes non-finite first
(flat params / flat grads / momentum / BN running stats / loss).

Usage (GPU box):  python scripts/graph_nan_diag.py
"""
import json
import sys

import torch

from flashy_amd.graph import CapturedStep
from flashy_amd.models import resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd.functional import cross_entropy


def build(seed, use_autocast=True, cache_enabled=True, native_ce=True,
          torch_sgd=False, warmup=3):
    torch.manual_seed(seed)
    model = resnet18(num_classes=10, small_input=True).cuda()
    if torch_sgd:
        opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    else:
        opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    static_x = torch.randn(16, 3, 32, 32, device="cuda")
    static_y = torch.randint(10, (16,), device="cuda")

    def step():
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", torch.bfloat16, enabled=use_autocast,
                            cache_enabled=cache_enabled):
            logits = model(static_x)
        if native_ce:
            loss = cross_entropy(logits.float(), static_y)
        else:
            loss = torch.nn.functional.cross_entropy(logits.float(), static_y)
        loss.backward()
        opt.step()
        return loss

    return model, opt, static_x, step, warmup


def finite_report(model, opt):
    rep = {}
    if isinstance(opt, FusedSGD):
        rep["flat_p"] = all(torch.isfinite(g.flat_p).all().item() for g in opt.groups)
        rep["flat_g"] = all(torch.isfinite(g.flat_g).all().item() for g in opt.groups)
        rep["momentum"] = all(m is None or torch.isfinite(m).all().item()
                              for m in opt._momentum_buffers)
    bn_ok = True
    for m in model.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            if not (torch.isfinite(m.running_mean).all() and torch.isfinite(m.running_var).all()):
                bn_ok = False
                break
    rep["bn_stats"] = bn_ok
    return rep


def run_variant(name, replays=6, detail=False, **kw):
    try:
        model, opt, static_x, step, warmup = build(3, **kw)
        graphed = CapturedStep(step, warmup=warmup).capture()
        losses = []
        details = []
        for i in range(replays):
            static_x.normal_()
            loss = graphed()
            torch.cuda.synchronize()
            losses.append(round(float(loss.item()), 4))
            if detail:
                details.append(finite_report(model, opt))
        ok = all(torch.isfinite(torch.tensor(losses)).tolist())
        out = {"variant": name, "ok": ok, "losses": losses}
        if detail:
            out["finite_after_each_replay"] = details
        print(json.dumps(out), flush=True)
    except Exception as e:  # noqa: BLE001
        print(json.dumps({"variant": name, "error": repr(e)}), flush=True)


def run_eager(name, steps=9, **kw):
    """Control: same step run eagerly (no capture)."""
    try:
        model, opt, static_x, step, warmup = build(3, **kw)
        losses = []
        for i in range(steps):
            static_x.normal_()
            loss = step()
            torch.cuda.synchronize()
            losses.append(round(float(loss.item()), 4))
        ok = all(torch.isfinite(torch.tensor(losses)).tolist())
        print(json.dumps({"variant": name, "ok": ok, "losses": losses}), flush=True)
    except Exception as e:  # noqa: BLE001
        print(json.dumps({"variant": name, "error": repr(e)}), flush=True)


if __name__ == "__main__":
    torch.backends.cudnn.benchmark = True
    print("torch", torch.__version__, file=sys.stderr)
    run_eager("eager-control")
    run_variant("repro-baseline", detail=True)
    run_variant("repro-again", detail=True)   # determinism check
    run_variant("no-autocast", use_autocast=False)
    run_variant("autocast-cache-off", cache_enabled=False)
    run_variant("torch-ce", native_ce=False)
    run_variant("torch-sgd", torch_sgd=True)
    run_variant("warmup-10", warmup=10)
