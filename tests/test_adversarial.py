# Copyright (c) Flashy-AMD authors.
"""CPU tests for the adversarial (GAN) loss wrapper."""
import copy

import torch
from torch import nn

from flashy_amd.adversarial import AdversarialLoss


def _setup(seed=0):
    torch.manual_seed(seed)
    gen = nn.Linear(4, 8)
    disc = nn.Sequential(nn.Linear(8, 8), nn.LeakyReLU(0.2), nn.Linear(8, 1))
    d_opt = torch.optim.Adam(disc.parameters(), lr=1e-2)
    return gen, AdversarialLoss(disc, d_opt)


def test_train_adv_updates_discriminator_only():
    gen, adv = _setup()
    z = torch.randn(16, 4)
    fake = gen(z)
    real = torch.randn(16, 8) + 2.0
    g_before = [p.clone() for p in gen.parameters()]
    d_before = [p.clone() for p in adv.adversary.parameters()]
    loss = adv.train_adv(fake, real)
    assert torch.isfinite(loss)
    assert all(torch.equal(a, b) for a, b in zip(gen.parameters(), g_before))
    assert not all(torch.equal(a, b)
                   for a, b in zip(adv.adversary.parameters(), d_before))


def test_generator_loss_does_not_touch_discriminator():
    gen, adv = _setup()
    fake = gen(torch.randn(16, 4))
    g_loss = adv(fake)
    g_loss.backward()
    assert all(p.grad is None for p in adv.adversary.parameters())
    assert all(p.grad is not None for p in gen.parameters())
    # requires_grad restored after the readonly context
    assert all(p.requires_grad for p in adv.adversary.parameters())


def test_adversarial_trains_discriminator():
    gen, adv = _setup()
    real = torch.randn(64, 8) + 3.0
    with torch.no_grad():
        fake = gen(torch.randn(64, 4))
    losses = [adv.train_adv(fake, real).item() for _ in range(50)]
    assert losses[-1] < losses[0], losses[::10]


def test_optimizer_embedded_in_state_dict():
    gen, adv = _setup()
    real = torch.randn(8, 8)
    fake = gen(torch.randn(8, 4)).detach()
    adv.train_adv(fake, real)
    state = copy.deepcopy(adv.state_dict())
    assert "optimizer" in state
    # mutate, then restore: weights AND optimizer state come back
    adv.train_adv(fake, real)
    adv2_gen, adv2 = _setup(seed=1)
    adv2.load_state_dict(state)
    for (k, v) in adv.state_dict().items():
        if k == "optimizer":
            continue
    restored = adv2.state_dict()
    for k, v in state.items():
        if k == "optimizer":
            assert restored[k]["state"].keys() == v["state"].keys()
        else:
            assert torch.equal(restored[k], v), k
