# Minimal single-shape conv driver for PMC counter collection.
import sys

import torch

sys.path.insert(0, ".")
from flashy_amd import ops  # noqa: E402

which = sys.argv[1] if len(sys.argv) > 1 else "fwd"
name = sys.argv[2] if len(sys.argv) > 2 else "l1conv"
SHAPES = {
    "l1conv": (32, 32, 64, 64, 3, 1),
    "l2conv": (16, 16, 128, 128, 3, 1),
    "l4conv": (4, 4, 512, 512, 3, 1),
    "r2_3x3": (28, 28, 128, 128, 3, 1),
    "r1_3x3": (56, 56, 64, 64, 3, 1),
    "r1_1x1b": (56, 56, 64, 256, 1, 1),
    "r4_3x3": (7, 7, 512, 512, 3, 1),      # 8-wave split-K path
    "r4_1x1a": (7, 7, 2048, 512, 1, 1),    # under-fill forced split-K
}
H, W, C, K, R, stride = SHAPES[name]
N, pad = 64, R // 2
torch.manual_seed(0)
x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
w = (torch.randn(K, R, R, C, device="cuda") * 0.05).to(torch.bfloat16)
d = ops.ConvDims.infer(x, w, stride, pad)
y = x.new_empty((d.N, d.Ho, d.Wo, d.K))
dy = torch.randn_like(y)
wt = w.new_empty((d.R, d.S, d.C, d.K))
ops.weight_transpose(w, wt)
dx = x.new_empty(x.shape)
dw = torch.zeros(K, R, R, C, device="cuda", dtype=torch.float32)
for _ in range(3):
    if which == "fwd":
        ops.conv_fwd(x, w, y, d)
    elif which == "dgrad":
        ops.conv_dgrad(dy, wt, dx, d)
    else:
        ops.conv_wgrad(x, dy, dw, d)
torch.cuda.synchronize()
print("done", which, name)
