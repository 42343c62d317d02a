# Copyright (c) Flashy-AMD authors.
"""Elastic restart wrapper: keep relaunching a flashy_amd entry point until
it exits cleanly, resuming from its last committed checkpoint each time.

Training jobs on shared MI355X pools get preempted; because
``BaseSolver.commit()`` writes an atomic checkpoint per epoch and
``restore()`` picks it up on the next run (same signature -> same XP
folder), a crash/preemption only ever loses the current epoch.  This
wrapper is the single-node analogue of a scheduler requeue hook:

    python examples/elastic/run_elastic.py -- \
        python -m examples.cifar.train epochs=20

Exit code 0 stops the loop; anything else (OOM kill, SIGTERM, node
failure simulated by `kill`) triggers a relaunch after a short backoff,
up to --max-restarts.
"""
from __future__ import annotations

import argparse
import subprocess
import sys
import time


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--max-restarts", type=int, default=10)
    ap.add_argument("--backoff", type=float, default=5.0,
                    help="seconds between relaunches")
    ap.add_argument("cmd", nargs=argparse.REMAINDER,
                    help="-- followed by the training command")
    args = ap.parse_args()
    cmd = args.cmd[1:] if args.cmd and args.cmd[0] == "--" else args.cmd
    if not cmd:
        ap.error("no training command given (use: run_elastic.py -- <cmd>)")
    for attempt in range(args.max_restarts + 1):
        if attempt:
            print(f"[elastic] restart {attempt}/{args.max_restarts} "
                  f"in {args.backoff:.0f}s", file=sys.stderr)
            time.sleep(args.backoff)
        rc = subprocess.call(cmd)
        if rc == 0:
            print("[elastic] run finished cleanly", file=sys.stderr)
            return 0
        print(f"[elastic] run exited with {rc}", file=sys.stderr)
    print("[elastic] giving up", file=sys.stderr)
    return 1


if __name__ == "__main__":
    sys.exit(main())
