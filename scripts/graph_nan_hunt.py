# Copyright (c) Flashy-AMD authors.
"""Hunt the intermittent graph-replay NaN (torch resnet18 + autocast +
FusedSGD under CapturedStep; losses go [ok, ok, nan, ...] on ~50% of
processes on some boxes).  Runs N trials of one variant per invocation and
reports which tensor family goes non-finite first.

Usage:  PYTHONPATH=. python scripts/graph_nan_hunt.py N [flags...]
flags: no-autocast torch-ce torch-sgd torch-bn-eval warmup8 stress
"""
import sys

import torch

from flashy_amd.graph import CapturedStep
from flashy_amd.models import resnet18
from flashy_amd.optim import FusedSGD
from flashy_amd.functional import cross_entropy

TRIALS = int(sys.argv[1]) if len(sys.argv) > 1 else 6
FLAGS = set(sys.argv[2:])


def finite_report(model, opt):
    rep = {}
    if isinstance(opt, FusedSGD):
        rep["flat_p"] = all(bool(torch.isfinite(g.flat_p).all())
                            for g in opt.groups)
        rep["flat_g"] = all(bool(torch.isfinite(g.flat_g).all())
                            for g in opt.groups)
        rep["momentum"] = all(m is None or bool(torch.isfinite(m).all())
                              for m in opt._momentum_buffers)
    else:
        rep["params"] = all(bool(torch.isfinite(p).all())
                            for p in model.parameters())
        rep["grads"] = all(p.grad is None or bool(torch.isfinite(p.grad).all())
                           for p in model.parameters())
    bad_bn = []
    for n, m in model.named_modules():
        if isinstance(m, torch.nn.BatchNorm2d):
            if not (torch.isfinite(m.running_mean).all()
                    and torch.isfinite(m.running_var).all()):
                bad_bn.append(n)
    rep["bad_bn"] = bad_bn[:4]
    return rep


def trial(i: int) -> bool:
    torch.manual_seed(3)
    model = resnet18(num_classes=10, small_input=True).cuda()
    if "no-fc" in FLAGS:   # isolate the autocast bf16 GEMM (hipBLASLt)
        model.fc = torch.nn.Identity()
    if "fp32-fc" in FLAGS:  # keep autocast but run the fc GEMM in fp32
        fc = model.fc

        class F32FC(torch.nn.Module):
            def forward(self, x):
                with torch.autocast("cuda", enabled=False):
                    return fc(x.float())

        model.fc = F32FC()
    if "torch-bn-eval" in FLAGS:
        for m in model.modules():
            if isinstance(m, torch.nn.BatchNorm2d):
                m.eval()
    if "torch-sgd" in FLAGS:
        opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)
    else:
        opt = FusedSGD(model.parameters(), lr=0.01, momentum=0.9)
    static_x = torch.randn(16, 3, 32, 32, device="cuda")
    static_y = torch.randint(10, (16,), device="cuda")
    use_ac = "no-autocast" not in FLAGS

    def step():
        opt.zero_grad(set_to_none=False)
        with torch.autocast("cuda", torch.bfloat16, enabled=use_ac):
            logits = model(static_x)
        if "no-fc" in FLAGS:
            logits = logits[:, :10]   # pooled features stand in for logits
        if "torch-ce" in FLAGS:
            loss = torch.nn.functional.cross_entropy(logits.float(), static_y)
        else:
            loss = cross_entropy(logits.float(), static_y)
        loss.backward()
        opt.step()
        return loss

    wu = 8 if "warmup8" in FLAGS else 3
    graphed = CapturedStep(step, warmup=wu).capture()
    losses = []
    junk = []
    for r in range(6):
        static_x.normal_()
        loss = graphed()
        torch.cuda.synchronize()
        v = float(loss.item())
        losses.append(round(v, 4))
        if v != v:
            print(f"trial {i} NAN at replay {r}: {losses} "
                  f"{finite_report(model, opt)}", flush=True)
            return False
        if "stress" in FLAGS:
            junk.append(torch.randn(1 << (14 + r), device="cuda"))
            if len(junk) > 2:
                junk.pop(0)
    print(f"trial {i} ok: {losses}", flush=True)
    return True


if __name__ == "__main__":
    fails = sum(0 if trial(i) else 1 for i in range(TRIALS))
    print(f"RESULT {sorted(FLAGS)}: {fails}/{TRIALS} failed", flush=True)
    sys.exit(0)
