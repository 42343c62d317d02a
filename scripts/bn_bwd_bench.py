"""Sweep bn_bwd_reduce msplit (block-count) policy over the ResNet-50/224
BN shapes.  The round-2 R50 profile (profiles/r02h) showed small-M shapes
running at 2-2.5 TB/s effective (vs ~7 TB/s for large-M) under the fixed
~512-block budget of ops.bn_msplit — each wave gets only 2-3 loop
iterations and the chip sits at 8 waves/CU.  This measures reduce+combine
total per shape for several block budgets.

Run: gpurun -- 'PYTHONPATH=. python scripts/bn_bwd_bench.py'
"""
import torch

import flashy_amd.ops as ops


def msplit_for(budget: int, M: int, C: int) -> int:
    cols = max(1, C // 64)
    msplit = max(1, min(budget // cols, (M + 31) // 32))
    if msplit >= 4:
        msplit &= ~3
    return msplit


def bench(fn, iters=200):
    s = torch.cuda.Event(True)
    e = torch.cuda.Event(True)
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) * 1e3 / iters  # us


def main():
    torch.manual_seed(0)
    dev = "cuda"
    # (M, C, count/step) for ResNet-50 b64 224 (+ stem)
    shapes = [
        (802816, 64, 1), (200704, 64, 6), (200704, 256, 4),
        (50176, 128, 8), (50176, 512, 5), (12544, 256, 12),
        (12544, 1024, 7), (3136, 512, 6), (3136, 2048, 4),
    ]
    budgets = [512, 1024, 2048, 4096]
    print(f"{'M':>8} {'C':>5} | " + " | ".join(f"b={b:>4}" for b in budgets)
          + "   (reduce+combine us; * = current policy)")
    tot = {b: 0.0 for b in budgets}
    for M, C, n in shapes:
        dy = torch.randn(M, C, device=dev, dtype=torch.bfloat16)
        y = torch.randn(M, C, device=dev, dtype=torch.bfloat16)
        x = torch.randn(M, C, device=dev, dtype=torch.bfloat16)
        work = torch.rand(3 * C, device=dev) + 0.5
        dz = torch.empty_like(dy)
        bsums = torch.zeros(2 * C, device=dev)
        dgamma = torch.zeros(C, device=dev)
        dbeta = torch.zeros(C, device=dev)
        row = []
        for b in budgets:
            ms = msplit_for(b, M, C)
            partials = torch.empty(2 * C * ms, device=dev)

            def step(ms=ms, partials=partials):
                ops.bn_bwd_reduce(dy, y, x, work, dz, partials, M, C, ms,
                                  True, 0.0)
                ops.bn_bwd_grads(partials, ms, bsums, dgamma, dbeta, C)

            t = bench(step)
            cur = ops.bn_msplit(M, C) == ms
            row.append(f"{t:6.1f}{'*' if cur else ' '}")
            tot[b] += t * n
        gb = 4 * M * C * 2 / 1e9
        print(f"{M:>8} {C:>5} | " + " | ".join(row)
              + f"   [{gb*1e3:.0f} MB moved]")
    print("\nper-step totals (us): "
          + "  ".join(f"b={b}: {tot[b]:.0f}" for b in budgets))


if __name__ == "__main__":
    main()
