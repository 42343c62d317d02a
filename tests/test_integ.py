# Copyright (c) Flashy-AMD authors.
"""End-to-end integration via the real CLI (subprocess), mirroring the
reference's resume oracle (reference tests/test_integ.py): run the dummy
workload to epoch 2, re-run without --clear and assert the history grows to
4 with a bit-identical 2-epoch prefix, then exercise 2-worker local DDP."""
import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def _run(tmp: Path, *args: str) -> None:
    env = dict(os.environ)
    env["_FLASHY_AMD_DIR"] = str(tmp)
    env["PYTHONPATH"] = str(REPO) + os.pathsep + env.get("PYTHONPATH", "")
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    subprocess.run([sys.executable, "-m", "flashy_amd.run", "tests.dummy", *args],
                   check=True, cwd=REPO, env=env, timeout=300)


def _history(tmp: Path) -> list:
    xps = list((tmp / "xps").iterdir())
    assert len(xps) == 1, xps
    with open(xps[0] / "history.json") as fh:
        return json.load(fh)


def test_resume_prefix_equality(tmp_path):
    # stop_at is excluded from the signature via run.exclude so partial and
    # full runs share one XP
    excl = "run.exclude=[stop_at]"
    _run(tmp_path, "--clear", excl, "stop_at=2")
    hist2 = _history(tmp_path)
    assert len(hist2) == 2

    _run(tmp_path, excl)  # stop_at back to null -> runs to epochs=4
    hist4 = _history(tmp_path)
    assert len(hist4) == 4
    assert hist4[:2] == hist2  # bit-identical persisted prefix


def test_local_ddp_two_workers(tmp_path):
    _run(tmp_path, "--clear", "-d", "--workers", "2", "epochs=2")
    hist = _history(tmp_path)
    assert len(hist) == 2


def test_bench_distributed_contract(tmp_path):
    """bench.py under torch.distributed.run exactly as the driver launches
    it (one JSON line from rank 0, value aggregated over the world)."""
    import torch
    if torch.cuda.is_available() and torch.cuda.device_count() < 2:
        pytest.skip("2 ranks need 2 GPUs once cuda is visible (RCCL "
                    "refuses two ranks on one device); the CPU/gloo "
                    "container covers the contract")
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--no-ckpt", "--batch", "8"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    result = json.loads(lines[0])
    assert result["n_gpus"] == 2
    assert result["config"]["parallelism"] == "dp2"
    assert result["config"]["global_batch"] == 16
    assert result["value"] > 0


def test_elastic_restart(tmp_path):
    """The elastic wrapper relaunches a failing command until it succeeds."""
    import subprocess, sys
    marker = tmp_path / "tries"
    prog = (
        "import pathlib, sys; p = pathlib.Path(%r); "
        "n = int(p.read_text()) if p.exists() else 0; "
        "p.write_text(str(n + 1)); sys.exit(0 if n >= 2 else 3)" % str(marker))
    rc = subprocess.call([sys.executable, "examples/elastic/run_elastic.py",
                          "--backoff", "0.01", "--",
                          sys.executable, "-c", prog])
    assert rc == 0
    assert marker.read_text() == "3"  # failed twice, succeeded third
