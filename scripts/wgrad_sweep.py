# Sweep wgrad n_splits for CIFAR + ResNet-50/224 layer shapes.
import sys
import torch
sys.path.insert(0, ".")
from flashy_amd import ops

SHAPES = {  # name: (N,H,W,C,K,R,stride)
    "c10-l1": (64, 32, 32, 64, 64, 3, 1),
    "c10-l4": (64, 4, 4, 512, 512, 3, 1),
    "r50-l1c2": (64, 56, 56, 64, 64, 3, 1),
    "r50-l2c2": (64, 28, 28, 128, 128, 3, 1),
    "r50-l1c3": (64, 56, 56, 64, 256, 1, 1),
    "r50-l3c2": (64, 14, 14, 256, 256, 3, 1),
}
for name, (N, H, W, C, K, R, st) in SHAPES.items():
    pad = R // 2
    x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
    w = (torch.randn(K, R, R, C, device="cuda") * 0.1).to(torch.bfloat16)
    d = ops.ConvDims.infer(x, w, st, pad)
    dy = torch.randn(N, d.Ho, d.Wo, K, device="cuda").to(torch.bfloat16)
    dw = torch.zeros(K, R, R, C, device="cuda", dtype=torch.float32)
    tiles = (K // 64) * (R * R * C // 64)
    out = [name + f" tiles={tiles}:"]
    for target in (256, 512, 1024, 2048):
        ns = max(1, min(target // tiles if tiles else 1, 128))
        ns = max(1, min(ns, d.N * d.Ho * d.Wo // 32 or 1))
        for _ in range(3):
            ops.conv_wgrad(x, dy, dw, d, n_splits=ns)
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        s.record()
        for _ in range(20):
            ops.conv_wgrad(x, dy, dw, d, n_splits=ns)
        e.record(); torch.cuda.synchronize()
        out.append(f"{target}->z{ns}: {s.elapsed_time(e)/20*1000:.0f}us")
    print("  ".join(out))
