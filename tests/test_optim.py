# Copyright (c) Flashy-AMD authors.
"""Numerics tests for the flat fused optimizers (CPU fallback path; the GPU
kernels are tested against the same oracles in test_ops_gpu.py)."""
import copy

import pytest
import torch
from torch import nn

from flashy_amd.optim import FusedAdam, FusedSGD


def _models():
    torch.manual_seed(7)
    a = nn.Sequential(nn.Linear(10, 32), nn.ReLU(), nn.Linear(32, 4))
    b = copy.deepcopy(a)
    return a, b


def _train(model, opt, steps=5, seed=3):
    g = torch.Generator().manual_seed(seed)
    for _ in range(steps):
        x = torch.randn(8, 10, generator=g)
        y = torch.randn(8, 4, generator=g)
        loss = torch.nn.functional.mse_loss(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()


@pytest.mark.parametrize("momentum,wd,nesterov", [
    (0.0, 0.0, False), (0.9, 0.0, False), (0.9, 5e-4, False), (0.9, 1e-3, True)])
def test_fused_sgd_matches_torch(momentum, wd, nesterov):
    ours, ref = _models()
    opt_o = FusedSGD(ours.parameters(), lr=0.05, momentum=momentum,
                     weight_decay=wd, nesterov=nesterov)
    opt_r = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=momentum,
                            weight_decay=wd, nesterov=nesterov)
    _train(ours, opt_o)
    _train(ref, opt_r)
    for p, q in zip(ours.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-6), (p - q).abs().max()


@pytest.mark.parametrize("wd,adamw", [(0.0, False), (1e-2, False), (1e-2, True)])
def test_fused_adam_matches_torch(wd, adamw):
    ours, ref = _models()
    opt_o = FusedAdam(ours.parameters(), lr=1e-2, weight_decay=wd, adamw=adamw)
    klass = torch.optim.AdamW if adamw else torch.optim.Adam
    opt_r = klass(ref.parameters(), lr=1e-2, weight_decay=wd)
    _train(ours, opt_o)
    _train(ref, opt_r)
    for p, q in zip(ours.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-6), (p - q).abs().max()


def test_flat_views_alias():
    model, _ = _models()
    opt = FusedSGD(model.parameters(), lr=0.1)
    flat = opt.param_buffers[0]
    total = sum(p.numel() for p in model.parameters())
    assert flat.numel() == total
    # params are views into the flat buffer
    with torch.no_grad():
        flat.zero_()
    assert all(torch.all(p == 0) for p in model.parameters())
    # grads are views into the flat grad buffer
    model(torch.randn(2, 10)).sum().backward()
    assert opt.grad_buffers[0].abs().sum() > 0
    opt.zero_grad()
    assert all(p.grad.abs().sum() == 0 for p in model.parameters())


def test_state_roundtrip():
    ours, _ = _models()
    opt = FusedSGD(ours.parameters(), lr=0.05, momentum=0.9)
    _train(ours, opt, steps=3)
    saved = copy.deepcopy(opt.state_dict())
    before = [m.clone() for m in opt._momentum_buffers]
    _train(ours, opt, steps=2)
    opt.load_state_dict(saved)
    for m, b in zip(opt._momentum_buffers, before):
        assert torch.equal(m, b)
    assert opt.step_count == 3


def test_checkpoint_views_compacted(tmp_path):
    """Saving flat-view params must not serialize the whole flat storage per
    param (flashy_amd/checkpoint.py _compact_cpu)."""
    from flashy_amd import checkpoint as fckpt
    model, _ = _models()
    FusedSGD(model.parameters(), lr=0.1)
    path = tmp_path / "m.th"
    fckpt.save_state(model.state_dict(), path)
    n_bytes = sum(p.numel() * 4 for p in model.parameters())
    assert path.stat().st_size < n_bytes * 2 + 10000
    loaded = fckpt.load_state(path)
    model2, _ = _models()
    model2.load_state_dict(loaded)
    for p, q in zip(model.parameters(), model2.parameters()):
        assert torch.equal(p, q)


def test_bf16_mirror_refresh_on_restore():
    """Restoring fp32 params must re-sync the bf16 weight mirrors
    (load_state_dict path; solver-level restore calls refresh_bf16)."""
    import torch
    model, _ = _models()
    opt = FusedSGD(model.parameters(), lr=0.05, momentum=0.9, bf16_mirror=True)
    p = next(model.parameters())
    assert hasattr(p, "_bf16_mirror")
    assert torch.allclose(p._bf16_mirror.float(), p.float(), atol=1e-2)
    saved_params = [q.detach().clone() for q in model.parameters()]
    saved_opt = opt.state_dict()
    _train(model, opt, steps=2)
    # emulate a restore: params back in place + optimizer state reload
    with torch.no_grad():
        for q, s in zip(model.parameters(), saved_params):
            q.copy_(s)
    opt.load_state_dict(saved_opt)
    assert torch.allclose(p._bf16_mirror.float(),
                          p.detach().to(torch.bfloat16).float())
