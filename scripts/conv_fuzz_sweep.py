"""Broad randomized conv correctness sweep (standalone, GPU).

Samples the full supported envelope — C,K multiples of 64 (plus C=3/4
stems), R in {1,3,4,5,7}, stride in {1,2}, odd spatials, tiny M — and
checks fwd/dgrad/wgrad against a plain torch fp32 reference of the same
op on the same bf16-rounded inputs.  Covers every dispatch branch:
dedup / mloop / fwd8 / split-K / under-fill forcing / SCAT2 / s2-parity
/ stem pad / old-tile fallbacks.

Run: gpurun -- 'PYTHONPATH=. python scripts/conv_fuzz_sweep.py [n]'
"""
import random
import sys

import torch
import torch.nn.functional as F

from flashy_amd import ops


def sample_shapes(n, rng):
    shapes = []
    while len(shapes) < n:
        stem = rng.random() < 0.12
        if stem:
            R = rng.choice([3, 4, 7])
            # R=7 stems are C=3 only (the pow2-padded 8-wave path); the
            # scalar stem kernels cap rsc at 160
            C = 3 if R == 7 else rng.choice([3, 4])
            K = 64
        else:
            C = 64 * rng.choice([1, 1, 2, 3, 4, 8])
            K = 64 * rng.choice([1, 1, 2, 3, 4, 8])
            R = rng.choice([1, 1, 3, 3, 4, 5, 7])
        stride = rng.choice([1, 1, 2])
        pad = R // 2 if R > 1 else 0
        N = rng.choice([1, 2, 3, 8, 16])
        H = rng.randint(4, 40)
        W = rng.randint(4, 40)
        Ho = (H + 2 * pad - R) // stride + 1
        Wo = (W + 2 * pad - R) // stride + 1
        if Ho < 1 or Wo < 1:
            continue
        if (N * H * W * C + N * Ho * Wo * K) * 2 > 1 << 30:
            continue
        shapes.append((N, H, W, C, K, R, stride, pad, stem))
    return shapes


def check(shape):
    N, H, W, C, K, R, stride, pad, stem = shape
    g = torch.Generator(device="cuda").manual_seed(hash(shape) & 0xffffff)
    x = torch.randn(N, H, W, C, device="cuda", generator=g).to(torch.bfloat16)
    w = (torch.randn(K, R, R, C, device="cuda", generator=g) * 0.1).to(
        torch.bfloat16)
    d = ops.ConvDims.infer(x, w, stride, pad)
    dy = torch.randn(N, d.Ho, d.Wo, K, device="cuda", generator=g).to(
        torch.bfloat16)

    y = x.new_empty((d.N, d.Ho, d.Wo, d.K))
    ops.conv_fwd(x, w, y, d)   # wrapper dispatches the C<8 stem paths too
    dx = x.new_empty(x.shape)
    if stem:
        ops.conv_stem_dgrad(dy, w, dx, d)
    else:
        wt = w.new_empty((d.R, d.S, d.C, d.K))
        ops.weight_transpose(w, wt)
        ops.conv_dgrad(dy, wt, dx, d)
    dw = torch.zeros(K, R, R, C, device="cuda", dtype=torch.float32)
    ops.conv_wgrad(x, dy, dw, d)
    torch.cuda.synchronize()

    xr = x.float().permute(0, 3, 1, 2).requires_grad_(True)
    wr = w.float().permute(0, 3, 1, 2).requires_grad_(True)
    ref = F.conv2d(xr, wr, stride=stride, padding=pad)
    ref.backward(dy.float().permute(0, 3, 1, 2))
    fails = []
    for got, want, tag in [
            (y.float(), ref.detach().permute(0, 2, 3, 1), "fwd"),
            (dx.float(), xr.grad.permute(0, 2, 3, 1), "dgrad"),
            (dw, wr.grad.permute(0, 2, 3, 1), "wgrad")]:
        err = (got - want).abs().max().item()
        scale = want.abs().max().item() + 1e-6
        if err / scale >= 2e-2:
            fails.append((tag, err / scale))
    return fails


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 80
    rng = random.Random(20260914)
    shapes = sample_shapes(n, rng)
    bad = 0
    for i, sh in enumerate(shapes):
        fails = check(sh)
        if fails:
            bad += 1
            print(f"FAIL {sh}: {fails}")
    print(f"{len(shapes) - bad}/{len(shapes)} shapes pass "
          f"(fwd+dgrad+wgrad vs torch fp32)")
    if bad:
        sys.exit(1)


if __name__ == "__main__":
    main()
