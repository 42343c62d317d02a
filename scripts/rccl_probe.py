# Copyright (c) Flashy-AMD authors.
"""RCCL bring-up probe at world_size=1 on a single MI355X:

(a) init_process_group(nccl) + every collective the framework uses
    (all_reduce, broadcast, barrier) — catches env/plumbing faults that a
    gloo CPU test cannot;
(b) an all_reduce captured INSIDE a HIP graph and replayed — validates that
    RCCL collectives are hipGraph-capturable on this stack, the prerequisite
    for the overlapped in-graph DP design (flashy_amd/distrib.py chunked
    sync).

Run:  MASTER_ADDR=127.0.0.1 MASTER_PORT=29617 RANK=0 WORLD_SIZE=1 \
      python scripts/rccl_probe.py
"""
import json
import os

import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29617")
os.environ.setdefault("RANK", "0")
os.environ.setdefault("WORLD_SIZE", "1")

out = {}

dist.init_process_group("nccl", init_method="env://")
torch.cuda.set_device(0)
out["init"] = "ok"

# (a) eager collectives
t = torch.arange(1024, device="cuda", dtype=torch.float32)
dist.all_reduce(t)
assert torch.equal(t, torch.arange(1024, device="cuda", dtype=torch.float32))
dist.broadcast(t, src=0)
dist.barrier()
big = torch.ones(11_000_000, device="cuda")  # ~44 MB fp32, the flagship payload
dist.all_reduce(big)
torch.cuda.synchronize()
out["eager_collectives"] = "ok"

# (b) all_reduce inside a HIP graph
try:
    g = torch.cuda.CUDAGraph()
    static = torch.ones(11_000_000, device="cuda")
    # warmup on side stream (allocator state)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            dist.all_reduce(static)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    with torch.cuda.graph(g):
        dist.all_reduce(static)
    static.fill_(2.0)
    for _ in range(3):
        g.replay()
    torch.cuda.synchronize()
    out["graphed_allreduce"] = "ok" if float(static[0]) == 2.0 else \
        f"wrong value {float(static[0])}"
except Exception as e:  # noqa: BLE001
    out["graphed_allreduce"] = f"FAIL: {e!r}"

# (c) all_reduce on a side stream inside a capture (the overlapped-chunk shape)
try:
    g2 = torch.cuda.CUDAGraph()
    payload = torch.ones(4_000_000, device="cuda")
    comm = torch.cuda.Stream()

    def fn():
        # "compute stream" = whatever stream we are running/capturing on
        cur = torch.cuda.current_stream()
        payload.mul_(1.0)  # compute-stream work producing the payload
        ev = torch.cuda.Event()
        ev.record(cur)
        comm.wait_event(ev)
        with torch.cuda.stream(comm):
            dist.all_reduce(payload)
            done = torch.cuda.Event()
            done.record(comm)
        cur.wait_event(done)
        payload.add_(1.0)

    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    with torch.cuda.graph(g2):
        fn()
    payload.fill_(5.0)
    g2.replay()
    torch.cuda.synchronize()
    out["graphed_sidestream_allreduce"] = "ok" if float(payload[0]) == 6.0 else \
        f"wrong value {float(payload[0])}"
except Exception as e:  # noqa: BLE001
    out["graphed_sidestream_allreduce"] = f"FAIL: {e!r}"

print(json.dumps(out))
dist.destroy_process_group()
