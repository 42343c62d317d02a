# A/B soak for the async checkpointer resume transient:
# N rounds of {run 1 epoch -> new process resumes +1 epoch}, async vs sync,
# with deterministic wgrad (FLASHY_WGRAD_SPLITS=1) so workload noise cannot
# masquerade as checkpoint corruption.  A bad restore shows up as a first-
# resumed-epoch train loss far above the continuation trend.
#
# Usage (GPU box): PYTHONPATH=. python scripts/soak_async_ab.py [rounds]
import json
import os
import pathlib
import shutil
import subprocess
import sys

ROUNDS = int(sys.argv[1]) if len(sys.argv) > 1 else 6


def one_round(mode: str, i: int):
    root = pathlib.Path(f"/tmp/soak_ab_{mode}_{i}")
    shutil.rmtree(root, ignore_errors=True)
    env = dict(os.environ, _FLASHY_AMD_DIR=str(root), PYTHONPATH=".",
               FLASHY_WGRAD_SPLITS="1")
    base = [sys.executable, "-m", "examples.cifar.train",
            "dataset_size=2048", "valid_size=512", "batch_size=64",
            "run.exclude=[device,use_graph,epochs]",
            f"async_checkpoint={'true' if mode == 'async' else 'false'}"]
    rc = subprocess.call(base + ["epochs=1"], env=env,
                         stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    assert rc == 0, (mode, i, "first run failed")
    rc = subprocess.call(base + ["epochs=2"], env=env,
                         stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    assert rc == 0, (mode, i, "resume run failed")
    hist = json.loads(next(root.glob("xps/*/history.json")).read_text())
    assert len(hist) == 2, hist
    return [round(h["train"]["loss"], 4) for h in hist]


out = {}
for mode in ("async", "sync"):
    losses = []
    for i in range(ROUNDS):
        ep = one_round(mode, i)
        losses.append(ep)
        print(mode, i, ep, flush=True)
    # epoch-2 (resumed) loss must continue the trend, not blow past epoch-1
    bad = [ep for ep in losses if ep[1] > ep[0] + 0.5]
    out[mode] = {"runs": losses, "bad_resumes": len(bad)}
print(json.dumps(out))
