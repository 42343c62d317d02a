# Copyright (c) Flashy-AMD authors.
"""Hunt the intermittent graph-replay NaN (torch resnet18 + autocast +
FusedSGD under CapturedStep; losses go [ok, ok, nan, ...] on some fresh
boxes/runs).  Reproduces the suite context (other graphs captured first),
optionally stresses the allocator between replays, and reports which
tensor family goes non-finite first.

Usage:  PYTHONPATH=. python scripts/graph_nan_hunt.py [trials] [stress]
"""
import sys

import torch

from flashy_amd.graph import CapturedStep
from flashy_amd.models import resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd.functional import cross_entropy

TRIALS = int(sys.argv[1]) if len(sys.argv) > 1 else 6
STRESS = len(sys.argv) > 2 and sys.argv[2] == "stress"


def suite_preamble():
    """Capture-and-discard a couple of graphs like the test suite does
    before the failing test (allocator/pool state)."""
    m = torch.nn.Sequential(torch.nn.Linear(64, 64), torch.nn.ReLU(),
                            torch.nn.Linear(64, 64)).cuda()
    x = torch.randn(32, 64, device="cuda")

    def step():
        return m(x).square().mean()

    g = CapturedStep(step, warmup=2).capture()
    for _ in range(3):
        g()
    torch.cuda.synchronize()
    del g, m, x


def finite_report(model, opt):
    rep = {}
    rep["flat_p"] = all(bool(torch.isfinite(g.flat_p).all()) for g in opt.groups)
    rep["flat_g"] = all(bool(torch.isfinite(g.flat_g).all()) for g in opt.groups)
    rep["momentum"] = all(m is None or bool(torch.isfinite(m).all())
                          for m in opt._momentum_buffers)
    bn_ok = True
    for m in model.modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            if not (torch.isfinite(m.running_mean).all()
                    and torch.isfinite(m.running_var).all()):
                bn_ok = False
    rep["bn_stats"] = bn_ok
    return rep


def trial(i: int) -> bool:
    torch.manual_seed(3)
    model = resnet18(num_classes=10, small_input=True).cuda()
    opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    static_x = torch.randn(16, 3, 32, 32, device="cuda")
    static_y = torch.randint(10, (16,), device="cuda")

    def step():
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", torch.bfloat16):
            logits = model(static_x)
        loss = cross_entropy(logits.float(), static_y)
        loss.backward()
        opt.step()
        return loss

    graphed = CapturedStep(step, warmup=3).capture()
    losses = []
    junk = []
    for r in range(6):
        static_x.normal_()
        loss = graphed()
        torch.cuda.synchronize()
        v = float(loss.item())
        losses.append(round(v, 4))
        if v != v:  # first NaN: forensics
            print(f"trial {i} NAN at replay {r}: {losses} "
                  f"{finite_report(model, opt)}", flush=True)
            return False
        if STRESS:  # churn the general allocator between replays
            junk.append(torch.randn(1 << (14 + r), device="cuda"))
            if len(junk) > 2:
                junk.pop(0)
    print(f"trial {i} ok: {losses}", flush=True)
    return True


if __name__ == "__main__":
    suite_preamble()
    fails = sum(0 if trial(i) else 1 for i in range(TRIALS))
    print(f"{fails}/{TRIALS} trials failed (stress={STRESS})")
    sys.exit(1 if fails else 0)
