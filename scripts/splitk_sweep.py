# Sweep split-K zn for the deep-layer fwd/dgrad shapes.
import sys
import torch
sys.path.insert(0, ".")
from flashy_amd import ops

SHAPES = {  # name: (N,H,W,C,K,R,stride)
    "c10-l3": (64, 8, 8, 256, 256, 3, 1),
    "c10-l4": (64, 4, 4, 512, 512, 3, 1),
    "r50-l4c2": (64, 7, 7, 512, 512, 3, 1),
}
ext = ops.require()
for name, (N, H, W, C, K, R, st) in SHAPES.items():
    pad = R // 2
    x = torch.randn(N, H, W, C, device="cuda").to(torch.bfloat16)
    w = (torch.randn(K, R, R, C, device="cuda") * 0.1).to(torch.bfloat16)
    d = ops.ConvDims.infer(x, w, st, pad)
    dy = torch.randn(N, d.Ho, d.Wo, K, device="cuda").to(torch.bfloat16)
    wt = w.new_empty((d.R, d.S, d.C, d.K))
    ops.weight_transpose(w, wt)
    dx = x.new_empty(x.shape)
    M = d.N * d.H * d.W
    rsk = d.R * d.S * d.K
    stages = (rsk + 63) // 64
    line = [name + f" (M={M}, stages={stages}):"]
    for zn in (0, 2, 4, 6, 8):
        if zn == 0:
            fn = lambda: ext.conv_dgrad(dy.data_ptr(), wt.data_ptr(),
                                        dx.data_ptr(), *d, ops._stream())
        else:
            spz = (stages + zn - 1) // zn
            zeff = (stages + spz - 1) // spz
            ws = torch.empty(zeff * M * d.C, dtype=torch.float32, device="cuda")
            def fn(spz=spz, zeff=zeff, ws=ws):
                ext.conv_dgrad_splitk(dy.data_ptr(), wt.data_ptr(),
                                      ws.data_ptr(), *d, spz, ops._stream())
                ext.splitk_combine(ws.data_ptr(), dx.data_ptr(), M * d.C,
                                   zeff, False, ops._stream())
        for _ in range(3): fn()
        torch.cuda.synchronize()
        s, e = torch.cuda.Event(True), torch.cuda.Event(True)
        s.record()
        for _ in range(20): fn()
        e.record(); torch.cuda.synchronize()
        line.append(f"z{zn}: {s.elapsed_time(e)/20*1000:.0f}us")
    print("  ".join(line))
