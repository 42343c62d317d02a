# Per-shape microbenchmark of the native conv kernels (ResNet-18/50 CIFAR
# shapes at batch 64).  Prints us/call and effective TFLOP/s for
# fwd / dgrad / wgrad on each shape.
import sys
import torch

sys.path.insert(0, ".")
from flashy_amd import ops  # noqa: E402

SHAPES = [
    # (name, H, W, C, K, R, stride)   batch 64, pad = R//2
    ("l1conv", 32, 32, 64, 64, 3, 1),
    ("l2down", 32, 32, 64, 128, 3, 2),
    ("l2conv", 16, 16, 128, 128, 3, 1),
    ("l2skip", 32, 32, 64, 128, 1, 2),
    ("l3conv", 8, 8, 256, 256, 3, 1),
    ("l4conv", 4, 4, 512, 512, 3, 1),
]

# ResNet-50 bottleneck shapes at 224px input (BASELINE config 5), batch 64
SHAPES_R50 = [
    ("r1_1x1a", 56, 56, 64, 64, 1, 1),
    ("r1_3x3", 56, 56, 64, 64, 3, 1),
    ("r1_1x1b", 56, 56, 64, 256, 1, 1),
    ("r1_1x1c", 56, 56, 256, 64, 1, 1),
    ("r2_down", 56, 56, 256, 512, 1, 2),
    ("r2_1x1a", 28, 28, 512, 128, 1, 1),
    ("r2_3x3", 28, 28, 128, 128, 3, 1),
    ("r2_1x1b", 28, 28, 128, 512, 1, 1),
    ("r3_1x1a", 14, 14, 1024, 256, 1, 1),
    ("r3_3x3", 14, 14, 256, 256, 3, 1),
    ("r3_1x1b", 14, 14, 256, 1024, 1, 1),
    ("r4_1x1a", 7, 7, 2048, 512, 1, 1),
    ("r4_3x3", 7, 7, 512, 512, 3, 1),
    ("r4_1x1b", 7, 7, 512, 2048, 1, 1),
]
if "--r50" in sys.argv:
    SHAPES = SHAPES_R50
N = 64
REPS = 30


def time_fn(fn, reps=REPS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / reps * 1000  # us


def main():
    torch.manual_seed(0)
    print(f"{'shape':8} {'fwd us':>8} {'fwdTF':>6} {'dgrad':>8} {'dgTF':>6} "
          f"{'wgrad':>8} {'wgTF':>6}")
    for name, H, W, C, K, R, stride in SHAPES:
        pad = R // 2
        x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
        w = (torch.randn(K, R, R, C, device="cuda") * 0.05).to(torch.bfloat16)
        d = ops.ConvDims.infer(x, w, stride, pad)
        y = x.new_empty((d.N, d.Ho, d.Wo, d.K))
        dy = torch.randn_like(y)
        wt = w.new_empty((d.R, d.S, d.C, d.K))
        ops.weight_transpose(w, wt)
        dx = x.new_empty(x.shape)
        dw = torch.zeros(K, R, R, C, device="cuda", dtype=torch.float32)
        flops = 2.0 * d.N * d.Ho * d.Wo * d.K * d.R * d.S * d.C

        t_f = time_fn(lambda: ops.conv_fwd(x, w, y, d))
        t_d = time_fn(lambda: ops.conv_dgrad(dy, wt, dx, d))
        t_w = time_fn(lambda: ops.conv_wgrad(x, dy, dw, d))
        print(f"{name:8} {t_f:8.1f} {flops / t_f / 1e6:6.0f} "
              f"{t_d:8.1f} {flops / t_d / 1e6:6.0f} "
              f"{t_w:8.1f} {flops / t_w / 1e6:6.0f}")


if __name__ == "__main__":
    main()
