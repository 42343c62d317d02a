# Copyright (c) Flashy-AMD authors.
import os

import pytest
import torch
from torch import nn

from flashy_amd.utils import averager, readonly, write_and_rename


def test_averager_running_mean():
    avg = averager()
    assert avg({"x": 1.0}) == {"x": 1.0}
    assert avg({"x": 3.0}) == {"x": 2.0}
    out = avg({"x": 5.0}, weight=2)
    assert out["x"] == pytest.approx((1 + 3 + 10) / 4)


def test_averager_ema():
    avg = averager(beta=0.5)
    avg({"x": 1.0})
    out = avg({"x": 2.0})
    # total = 1*0.5 + 2 = 2.5 ; fix = 0.5 + 1 = 1.5
    assert out["x"] == pytest.approx(2.5 / 1.5)


def test_write_and_rename_atomic(tmp_path):
    target = tmp_path / "f.bin"
    target.write_bytes(b"old")
    with pytest.raises(RuntimeError):
        with write_and_rename(target) as fh:
            fh.write(b"partial")
            raise RuntimeError("kill mid-write")
    assert target.read_bytes() == b"old"  # untouched on failure
    with write_and_rename(target) as fh:
        fh.write(b"new")
    assert target.read_bytes() == b"new"
    assert set(os.listdir(tmp_path)) == {"f.bin"}  # tmp renamed away


def test_readonly():
    model = nn.Linear(2, 2)
    model.bias.requires_grad_(False)
    with readonly(model):
        assert not any(p.requires_grad for p in model.parameters())
    assert model.weight.requires_grad
    assert not model.bias.requires_grad  # original state restored exactly


def test_readonly_blocks_grad_flow():
    d = nn.Linear(2, 1)
    x = torch.randn(3, 2, requires_grad=True)
    with readonly(d):
        d(x).sum().backward()
    assert d.weight.grad is None
    assert x.grad is not None
