# Copyright (c) Flashy-AMD authors.
"""GPU tests: NHWC maxpool vs torch, ImageNet-stem native ResNet, Adam-in-graph."""
import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


@requires_gpu
@pytest.mark.parametrize("shape", [(2, 8, 8, 64, 3, 2, 1), (3, 7, 7, 128, 3, 2, 1),
                                   (2, 8, 8, 64, 2, 2, 0)])
def test_maxpool_fwd_bwd(shape):
    from flashy_amd import nn as fnn
    N, H, W, C, k, s, p = shape
    torch.manual_seed(0)
    x16 = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
    pool = fnn.MaxPool2d(k, s, p)
    x = x16.detach().requires_grad_(True)
    y = pool(x)
    dy = torch.randn_like(y).to(torch.bfloat16)
    y.backward(dy)

    xr = x16.float().permute(0, 3, 1, 2).requires_grad_(True)
    ref = F.max_pool2d(xr, k, stride=s, padding=p)
    ref.backward(dy.float().permute(0, 3, 1, 2))
    ref_y = ref.permute(0, 2, 3, 1)
    ref_dx = xr.grad.permute(0, 2, 3, 1)
    assert torch.allclose(y.float(), ref_y, atol=1e-2, rtol=1e-2)
    # our dx is bf16 (ref is fp32): pixels summing >1 overlapping window's
    # grad round; single-contribution pixels are bit-exact
    diff = (x.grad.float() - ref_dx).abs()
    assert (diff > 0.01 * ref_dx.abs() + 0.02).float().mean().item() < 0.005, \
        diff.max().item()
    assert torch.allclose(x.grad.float().sum(), ref_dx.sum(), rtol=1e-2, atol=1.0)


@requires_gpu
def test_native_resnet50_imagenet_stem_trains():
    from flashy_amd.models import native_resnet50
    from flashy_amd.optim import FusedSGD
    from flashy_amd.functional import cross_entropy
    torch.manual_seed(1)
    model = native_resnet50(num_classes=1000, imagenet_stem=True).cuda().train()
    opt = FusedSGD(model.parameters(), lr=0.02, momentum=0.9, bf16_mirror=True)
    x = torch.randn(4, 3, 224, 224, device="cuda")
    y = torch.randint(1000, (4,), device="cuda")
    losses = []
    for _ in range(3):
        loss = cross_entropy(model(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert all(torch.isfinite(torch.tensor(losses))), losses


@requires_gpu
def test_fused_adam_graph_bias_correction():
    """The device step counter keeps Adam's bias correction advancing under
    graph replay (a captured host step count would freeze)."""
    import copy
    from flashy_amd.graph import CapturedStep
    from flashy_amd.optim import FusedAdam
    torch.manual_seed(2)
    model = torch.nn.Linear(32, 32).cuda()
    ref = copy.deepcopy(model)
    opt = FusedAdam(model.parameters(), lr=1e-2)
    opt_ref = torch.optim.Adam(ref.parameters(), lr=1e-2)
    x = torch.randn(16, 32, device="cuda")

    def step():
        opt.zero_grad()
        loss = (model(x) ** 2).mean()
        loss.backward()
        opt.step()
        return loss

    graphed = CapturedStep(step, warmup=0).capture()
    n_graph_steps = 6
    for _ in range(n_graph_steps):
        graphed()
    torch.cuda.synchronize()
    for _ in range(n_graph_steps):
        loss = (ref(x) ** 2).mean()
        opt_ref.zero_grad()
        loss.backward()
        opt_ref.step()
    for p, q in zip(model.parameters(), ref.parameters()):
        assert torch.allclose(p, q, atol=1e-4), (p - q).abs().max().item()
