# Copyright (c) Flashy-AMD authors.
# MI355X-native training-loop framework.
"""Small shared utilities.

Capability parity with the reference's ``flashy/utils.py`` (see
/root/reference/flashy/utils.py:19-69): a running/EMA metric averager, an
atomic write-then-rename file context, and a temporary ``requires_grad_(False)``
context manager.  Implementations are original.
"""
from __future__ import annotations

import os
import typing as tp
from contextlib import contextmanager
from pathlib import Path

# drop-in alias kept from the reference API (flashy/utils.py)
AnyPath = tp.Union[str, Path]

import torch


def averager(beta: float = 1.0) -> tp.Callable[..., tp.Dict[str, float]]:
    """Return a closure maintaining a (weighted) running average per metric key.

    With ``beta == 1`` this is the exact weighted mean of everything fed so
    far; with ``beta < 1`` it is an exponential moving average.  Each call
    ``avg(metrics, weight=1)`` folds in the new values and returns the current
    averages as a plain dict of floats.

    Parity: reference ``flashy/utils.py:19-37``.
    """
    num: tp.Dict[str, float] = {}
    den: tp.Dict[str, float] = {}

    def _update(metrics: tp.Mapping[str, tp.Any], weight: float = 1.0) -> tp.Dict[str, float]:
        for key, value in metrics.items():
            v = float(value)
            num[key] = num.get(key, 0.0) * beta + weight * v
            den[key] = den.get(key, 0.0) * beta + weight
        return {key: num[key] / den[key] for key in num}

    return _update


@contextmanager
def write_and_rename(path: tp.Union[str, Path], mode: str = "wb",
                     suffix: str = ".tmp", pid: bool = False):
    """Open ``path + suffix`` for writing, and atomically rename it onto
    ``path`` when the block exits without error.

    A crash mid-write leaves the previous file intact — the rename is the
    durability point.  ``pid=True`` appends the process id to the temp name so
    concurrent writers cannot collide.

    Parity: reference ``flashy/utils.py:40-54``.
    """
    path = Path(path)
    tmp = Path(str(path) + suffix + (f".{os.getpid()}" if pid else ""))
    tmp.parent.mkdir(parents=True, exist_ok=True)
    with open(tmp, mode) as fh:
        yield fh
        fh.flush()
        os.fsync(fh.fileno())
    os.rename(tmp, path)


@contextmanager
def readonly(model: torch.nn.Module):
    """Temporarily set ``requires_grad_(False)`` on all parameters of ``model``.

    Used by the adversarial loss so the generator backward does not
    accumulate into discriminator parameters.

    Parity: reference ``flashy/utils.py:57-69``.
    """
    states = [p.requires_grad for p in model.parameters()]
    try:
        for p in model.parameters():
            p.requires_grad_(False)
        yield
    finally:
        for p, s in zip(model.parameters(), states):
            p.requires_grad_(s)
