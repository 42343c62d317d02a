# Sweep bn_msplit for the BN kernel family (fwd stats+finalize+apply,
# bwd reduce+grads+apply) at ResNet layer shapes.
import sys
import torch
sys.path.insert(0, ".")
from flashy_amd import ops

SHAPES = {"l1": (65536, 64), "l2": (16384, 128), "l3": (4096, 256),
          "l4": (1024, 512)}
for name, (M, C) in SHAPES.items():
    x = torch.randn(M * C, device="cuda").to(torch.bfloat16)
    dy = torch.randn_like(x)
    y = torch.empty_like(x)
    dz = torch.empty_like(x)
    dx = torch.empty_like(x)
    gamma = torch.ones(C, device="cuda"); beta = torch.zeros(C, device="cuda")
    rm = torch.zeros(C, device="cuda"); rv = torch.ones(C, device="cuda")
    work = torch.empty(4 * C, device="cuda")
    bsums = torch.empty(2 * C, device="cuda")
    dg = torch.zeros(C, device="cuda"); db = torch.zeros(C, device="cuda")
    cur = ops.bn_msplit(M, C)
    line = [name + f" (cur z{cur}):"]
    for ms in (64, 128, 256, 512, 1024):
        ms = min(ms, max(1, (M + 31) // 32)) & ~3 or 4
        partials = torch.empty(2 * C * ms, device="cuda")
        def seq(ms=ms, partials=partials):
            ops.bn_stats(x, partials, M, C, ms)
            ops.bn_finalize(partials, ms, gamma, beta, rm, rv, work, M, C,
                            1e-5, 0.1, True)
            ops.bn_apply(x, None, y, work, M, C, True)
            ops.bn_bwd_reduce(dy, y, x, work, dz, partials, M, C, ms, True)
            ops.bn_bwd_grads(partials, ms, bsums, dg, db, C)
            ops.bn_bwd_apply(dz, x, work, bsums, dx, M, C)
        for _ in range(3): seq()
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        s.record()
        for _ in range(20): seq()
        e.record(); torch.cuda.synchronize()
        line.append(f"z{ms}: {s.elapsed_time(e)/20*1000:.0f}us")
    print("  ".join(line))
