# End-to-end solver soak on GPU: run the cifar example 2 epochs, kill,
# resume for 1 more, and verify history continuity + checkpoint presence.
import os
import pathlib
import subprocess
import sys

root = pathlib.Path("/tmp/soak_xp")
env = dict(os.environ, _FLASHY_AMD_DIR=str(root), PYTHONPATH=".")
base = [sys.executable, "-m", "examples.cifar.train",
        "epochs=2", "dataset_size=2048", "valid_size=512", "batch_size=64",
        "run.exclude=[device,use_graph,epochs]"]
if "--async" in sys.argv:   # exercise the experimental async writer
    base.append("async_checkpoint=true")
rc = subprocess.call(base, env=env)
assert rc == 0, rc
rc = subprocess.call([a if a != "epochs=2" else "epochs=3" for a in base], env=env)
assert rc == 0, rc
import json
sigs = list(root.glob("xps/*/history.json"))
assert len(sigs) == 1, sigs   # same signature -> same XP folder
hist = json.loads(sigs[0].read_text())
assert len(hist) == 3, len(hist)
assert all("train" in h and "valid" in h for h in hist)
ckpt = list(root.glob("xps/*/checkpoint.th"))
assert ckpt
print("soak ok: 2 epochs + resume 1, history:", [round(h["train"]["loss"], 3) for h in hist])
