# Copyright (c) Flashy-AMD authors.
"""Launcher CLI — the in-house replacement for ``dora run``.

Usage::

    python -m flashy_amd.run <package> [--clear] [-d] [--workers N] [k=v ...]

``<package>`` must expose ``<package>.train.main`` (an
:class:`flashy_amd.xp.EntryPoint`) or be a module itself exposing ``main``.

* plain run: executes the entry point in-process;
* ``-d --workers N``: spawns N local workers, one process per GPU, with
  torchrun-style env (RANK / WORLD_SIZE / LOCAL_RANK / MASTER_ADDR=127.0.0.1)
  so ``flashy_amd.distrib.init()`` brings up RCCL (or gloo on CPU).

Parity: reference ``dora run [--clear] [-d --ddp_workers=N] [overrides]``
(/root/reference/tests/test_integ.py:18-29).
"""
from __future__ import annotations

import argparse
import importlib
import os
import socket
import subprocess
import sys
import typing as tp


def _find_free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _resolve_main(package: str):
    try:
        module = importlib.import_module(package + ".train")
    except ImportError:
        module = importlib.import_module(package)
    main = getattr(module, "main", None)
    if main is None:
        raise RuntimeError(f"{package} does not expose a `main` entry point")
    return main


def run_workers(package: str, workers: int, args: tp.Sequence[str]) -> int:
    """Spawn `workers` local processes with distributed rendezvous env."""
    port = _find_free_port()
    procs = []
    for worker_rank in range(workers):
        env = dict(os.environ)
        env.update(RANK=str(worker_rank), LOCAL_RANK=str(worker_rank),
                   WORLD_SIZE=str(workers),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
        cmd = [sys.executable, "-m", "flashy_amd.run", package, *args]
        procs.append(subprocess.Popen(cmd, env=env))
    code = 0
    for p in procs:
        p.wait()
        code = code or p.returncode
    return code


def main(argv: tp.Optional[tp.Sequence[str]] = None) -> int:
    parser = argparse.ArgumentParser("flashy_amd.run")
    parser.add_argument("package", help="package exposing train.main")
    parser.add_argument("--clear", action="store_true",
                        help="wipe the XP folder before running")
    parser.add_argument("-d", "--distributed", action="store_true",
                        help="spawn local DDP workers")
    parser.add_argument("--workers", type=int, default=None,
                        help="number of DDP workers (default: GPU count)")
    parser.add_argument("overrides", nargs="*", help="config overrides k=v")
    ns = parser.parse_intermixed_args(argv)

    if ns.distributed and "RANK" not in os.environ:
        import torch
        workers = ns.workers or max(1, torch.cuda.device_count())
        forwarded = list(ns.overrides)
        if ns.clear:
            forwarded.insert(0, "--clear")
        return run_workers(ns.package, workers, forwarded)

    entry = _resolve_main(ns.package)
    # All ranks pass clear=True: EntryPoint.run rendezvous-gates the wipe
    # (rank 0 clears between two barriers) so no rank observes a half-
    # cleared XP.
    entry.run(ns.overrides, clear=ns.clear)
    return 0


if __name__ == "__main__":
    sys.exit(main())
