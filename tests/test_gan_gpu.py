# Copyright (c) Flashy-AMD authors.
"""GPU tests: transposed conv numerics vs torch, native DCGAN training."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


@requires_gpu
@pytest.mark.parametrize("shape", [
    # (N, Hi, Wi, Cin, Cout, R, stride, pad)
    (2, 8, 8, 128, 64, 4, 2, 1),
    (2, 1, 1, 128, 512, 4, 1, 0),
    (2, 8, 8, 64, 3, 4, 2, 1),     # RGB head (small-C edge kernel)
])
def test_convtranspose_matches_torch(shape):
    from flashy_amd import nn as fnn
    N, Hi, Wi, Cin, Cout, R, stride, pad = shape
    torch.manual_seed(0)
    ct = fnn.ConvTranspose2d(Cin, Cout, R, stride, pad).cuda()
    x16 = torch.randn(N, Hi, Wi, Cin, device="cuda").to(torch.bfloat16)
    x = x16.detach().requires_grad_(True)
    y = ct(x)
    dy = torch.randn_like(y).to(torch.bfloat16)
    y.backward(dy)

    # torch reference: weight [C_in, C_out, R, S]
    wt = ct.weight.detach().float().permute(0, 3, 1, 2)
    xr = x16.float().permute(0, 3, 1, 2).requires_grad_(True)
    wr = wt.clone().requires_grad_(True)
    ref = F.conv_transpose2d(xr, wr, stride=stride, padding=pad)
    ref.backward(dy.float().permute(0, 3, 1, 2))
    ref_y = ref.permute(0, 2, 3, 1)

    scale = ref_y.abs().max().item() + 1e-6
    assert (y.float() - ref_y).abs().max().item() / scale < 2e-2
    gscale = xr.grad.abs().max().item() + 1e-6
    assert (x.grad.float().permute(0, 3, 1, 2) - xr.grad).abs().max().item() \
        / gscale < 2e-2
    dw = ct.weight.grad.permute(0, 3, 1, 2)
    wscale = wr.grad.abs().max().item() + 1e-6
    assert (dw - wr.grad).abs().max().item() / wscale < 2e-2


@requires_gpu
def test_native_dcgan_trains():
    from flashy_amd.adversarial import AdversarialLoss
    from flashy_amd.models import (NativeDCGANDiscriminator,
                                   NativeDCGANGenerator)
    from flashy_amd.optim import FusedAdam
    torch.manual_seed(1)
    gen = NativeDCGANGenerator(nz=128, ngf=64).cuda().train()
    disc = NativeDCGANDiscriminator(ndf=64).cuda().train()
    g_opt = FusedAdam(gen.parameters(), lr=2e-4, betas=(0.5, 0.999))
    d_opt = FusedAdam(disc.parameters(), lr=2e-4, betas=(0.5, 0.999))
    adv = AdversarialLoss(disc, d_opt)
    real = torch.tanh(torch.randn(16, 3, 64, 64, device="cuda"))
    for i in range(3):
        z = torch.randn(16, 128, device="cuda")
        fake = gen(z)
        d_loss = adv.train_adv(fake, real)
        g_loss = adv(fake)
        g_opt.zero_grad()
        g_loss.backward()
        g_opt.step()
        assert torch.isfinite(d_loss) and torch.isfinite(g_loss), (i, d_loss, g_loss)
    assert fake.shape == (16, 3, 64, 64)
