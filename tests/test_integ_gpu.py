# Copyright (c) Flashy-AMD authors.
"""GPU end-to-end integration: the real cifar example CLI on the native
MI355X stack (NHWC HIP kernels + flat fused SGD + hipGraph step capture +
checkpoint commit/restore), with the reference's resume oracle
(reference tests/test_integ.py): partial run -> resume -> history grows
with the persisted prefix intact."""
import json
import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


def _run(tmp: Path, *args: str) -> None:
    env = dict(os.environ)
    env["_FLASHY_AMD_DIR"] = str(tmp)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    subprocess.run(
        [sys.executable, "-m", "flashy_amd.run", "examples.cifar", *args],
        check=True, cwd=REPO, env=env, timeout=420)


@requires_gpu
def test_cifar_native_gpu_resume(tmp_path):
    common = ("run.exclude=[epochs,device,use_graph]",
              "dataset_size=256", "valid_size=128")
    _run(tmp_path, "--clear", "epochs=2", *common)
    xps = list((tmp_path / "xps").iterdir())
    assert len(xps) == 1, xps
    hist_path = xps[0] / "history.json"
    with open(hist_path) as fh:
        hist2 = json.load(fh)
    assert len(hist2) == 2
    import math
    for epoch in hist2:  # native train+valid stages ran and logged finite loss
        assert "train" in epoch and "valid" in epoch, epoch
        assert math.isfinite(float(epoch["train"]["loss"])), epoch
    assert (xps[0] / "checkpoint.th").exists()

    _run(tmp_path, "epochs=4", *common)  # resume from the checkpoint
    with open(hist_path) as fh:
        hist4 = json.load(fh)
    assert len(hist4) == 4
    assert hist4[:2] == hist2  # persisted prefix untouched by the resume
