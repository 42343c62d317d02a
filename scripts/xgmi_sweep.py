# Copyright (c) Flashy-AMD authors.
"""xGMI chunk-size / RCCL-tuning sweep for the 8-GPU SCALE run (VERDICT r01
item 7).  Runs the flagship bench at world_size=N across a matrix of
FLASHY_AMD_CHUNK_MB (the OverlappedFlatSync chunk size) and RCCL env
settings, and prints img/s per configuration so the best defaults can be
committed.

Requires an N-GPU box:
    python scripts/xgmi_sweep.py --gpus 8 [--steps 40]

Rationale: each MI355X has 7 point-to-point xGMI links
(~153 GB/s each); a ring all-reduce is bound by ONE link, so chunk sizes
must amortize per-message latency without serializing behind backward,
and NCCL_MIN_NCHANNELS spreads the rings across links.
"""
import argparse
import itertools
import json
import os
import subprocess
import sys

parser = argparse.ArgumentParser()
parser.add_argument("--gpus", type=int, default=8)
parser.add_argument("--steps", type=int, default=40)
parser.add_argument("--warmup", type=int, default=10)
args = parser.parse_args()

CHUNK_MB = [4, 8, 16, 32]
NCCL_ENVS = [
    {},
    {"NCCL_MIN_NCHANNELS": "4"},
    {"NCCL_MIN_NCHANNELS": "8"},
    {"NCCL_ALGO": "Tree"},
]
MODES = ["graph-overlap", "posthoc"]

results = []
for mode, mb, extra in itertools.product(MODES, CHUNK_MB, NCCL_ENVS):
    if mode == "posthoc" and (mb != CHUNK_MB[0]):
        continue  # chunking only matters for the overlapped mode
    env = dict(os.environ, FLASHY_AMD_DP_MODE=mode,
               FLASHY_AMD_CHUNK_MB=str(mb), **extra)
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={args.gpus}", "--master-addr", "127.0.0.1",
           "--master-port", "29513", "bench.py", "--gpus", str(args.gpus),
           "--steps", str(args.steps), "--warmup", str(args.warmup),
           "--no-ckpt"]
    out = subprocess.run(cmd, env=env, capture_output=True, text=True,
                         timeout=600)
    val = None
    for line in out.stdout.splitlines():
        if line.startswith("{"):
            try:
                val = json.loads(line)
            except json.JSONDecodeError:
                pass
    row = {"mode": mode, "chunk_mb": mb, "env": extra,
           "img_s": val["value"] if val else None,
           "parallelism": val["config"]["parallelism"] if val else "FAILED"}
    results.append(row)
    print(json.dumps(row), flush=True)

best = max((r for r in results if r["img_s"]), key=lambda r: r["img_s"])
print("BEST:", json.dumps(best))
