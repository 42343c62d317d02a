# Copyright (c) Flashy-AMD authors.
"""CPU-checkable pieces of the ops layer: dim inference and launch plans."""
import torch

from flashy_amd import ops


def test_convdims_infer_asymmetric():
    x = torch.zeros(2, 17, 9, 8)
    w = torch.zeros(32, 5, 3, 8)            # R=5, S=3
    d = ops.ConvDims.infer(x, w, stride=2, pad=1)
    assert (d.R, d.S) == (5, 3)
    assert d.Ho == (17 + 2 - 5) // 2 + 1
    assert d.Wo == (9 + 2 - 3) // 2 + 1


def test_splitk_plan_regimes():
    # big-M short-K: single pass
    assert ops._splitk_plan(65536, 1, 9) == 0
    # tiny reductions never split
    assert ops._splitk_plan(128, 1, 2) == 0
    # underfilled grid: split for fill
    zn = ops._splitk_plan(1024, 8, 72)
    assert 2 <= zn <= 8
    # full grid + long reduction: split for the K length (measured 20-25%)
    zn = ops._splitk_plan(3136, 8, 72)
    assert zn == 5
    # cap at 8
    assert ops._splitk_plan(64, 1, 200) <= 8


def test_bn_msplit_alignment():
    for M, C in [(65536, 64), (16384, 128), (1024, 512), (48, 64)]:
        ms = ops.bn_msplit(M, C)
        assert ms >= 1
        if ms >= 4:
            assert ms % 4 == 0   # float4-aligned partial rows
        assert ms <= max(1, (M + 31) // 32)
