# Copyright (c) Flashy-AMD authors.
"""ws=1 validation harness for the overlapped in-graph DP step (VERDICT r01
item 2): brings up a real RCCL communicator on one MI355X, forces the
distributed code path (world_size=1 all-reduce is an identity), and captures
the WHOLE flagship step — forward + backward with chunked flat all-reduces
flushing from inside backward + fused SGD — into one HIP graph.

Checks:
  * capture succeeds with RCCL collectives recorded in-graph;
  * replayed losses are finite and match the no-comm graph trajectory
    (identity all-reduce => same numerics);
  * per-step time overhead of the in-graph comm schedule at ws=1.

Run:  MASTER_ADDR=127.0.0.1 RANK=0 WORLD_SIZE=1 PYTHONPATH=. \
      python scripts/overlap_harness.py
"""
import json
import os
import time

import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29651")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")

from flashy_amd import distrib  # noqa: E402
from flashy_amd.functional import cross_entropy  # noqa: E402
from flashy_amd.graph import CapturedStep  # noqa: E402
from flashy_amd.models import native_resnet18  # noqa: E402
from flashy_amd.optim import FusedSGD  # noqa: E402

dist.init_process_group("nccl", init_method="env://")
torch.cuda.set_device(0)

# force the distributed code path: ws=1 collectives are identities, so the
# overlapped schedule is exercised with identical numerics to no-comm
distrib.is_distributed = lambda: True

BATCH, STEPS, WARMUP = 64, 50, 10


def build():
    torch.manual_seed(7)
    model = native_resnet18(num_classes=10, imagenet_stem=False).cuda()
    opt = FusedSGD(model.parameters(), lr=0.1, momentum=0.9,
                   weight_decay=5e-4, bf16_mirror=True)
    model.enable_wt_cache()
    x = torch.randn(BATCH, 3, 32, 32, device="cuda")
    y = torch.randint(10, (BATCH,), device="cuda")
    return model, opt, x, y


def timed(runner, x):
    losses = []
    for _ in range(WARMUP):
        runner()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(STEPS):
        loss = runner()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / STEPS * 1000
    return dt, float(loss.item())


def loss_traj(runner, n=8):
    out = []
    for _ in range(n):
        loss = runner()
        torch.cuda.synchronize()
        out.append(round(float(loss.item()), 4))
    return out


out = {}

# --- baseline: no-comm whole-step graph (the ws=1 flagship path) ----------
model, opt, x, y = build()


def make_step(model, opt, x, y, sync=None):
    def step():
        opt.zero_grad(set_to_none=False)
        logits = model(x)
        loss = cross_entropy(logits, y)
        loss.backward()
        if sync is not None:
            sync.finish()
        opt.step()
        return loss
    return step


base = CapturedStep(make_step(model, opt, x, y), warmup=3).capture()
out["base_traj"] = loss_traj(base)
ms, _ = timed(base, x)
out["base_ms"] = round(ms, 4)

# --- overlapped: chunked in-graph all-reduce --------------------------------
model2, opt2, x2, y2 = build()
sync = distrib.OverlappedFlatSync(opt2)
out["n_chunks"] = sync.n_chunks
try:
    ov = CapturedStep(make_step(model2, opt2, x2, y2, sync), warmup=3).capture()
    out["overlap_traj"] = loss_traj(ov)
    ms, _ = timed(ov, x2)
    out["overlap_ms"] = round(ms, 4)
    drift = max(abs(a - b) for a, b in zip(out["base_traj"], out["overlap_traj"]))
    out["traj_max_drift"] = round(drift, 4)
    out["finite"] = all(torch.isfinite(torch.tensor(out["overlap_traj"])).tolist())
except Exception as e:  # noqa: BLE001
    out["overlap"] = f"FAIL: {e!r}"

# --- smaller chunks (more in-backward flushes) ------------------------------
for mb in (2, 4, 16):
    model3, opt3, x3, y3 = build()
    s3 = distrib.OverlappedFlatSync(opt3, chunk_bytes=mb << 20)
    try:
        g3 = CapturedStep(make_step(model3, opt3, x3, y3, s3), warmup=3).capture()
        ms, loss = timed(g3, x3)
        out[f"chunk{mb}MB"] = {"n_chunks": s3.n_chunks, "ms": round(ms, 4),
                               "loss": round(loss, 4)}
    except Exception as e:  # noqa: BLE001
        out[f"chunk{mb}MB"] = f"FAIL: {e!r}"

print(json.dumps(out))
dist.destroy_process_group()
