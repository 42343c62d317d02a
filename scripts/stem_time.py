import sys, torch
sys.path.insert(0, ".")
from flashy_amd import ops
# CIFAR stem: N64 32x32 C3->K64 3x3 s1  +  DCGAN edge 64x64
for (N,H,W,C,K,R,st,pad) in [(64,32,32,3,64,3,1,1),(64,64,64,3,64,4,2,1)]:
    x = torch.randn(N,H,W,C,device="cuda").to(torch.bfloat16)
    w = (torch.randn(K,R,R,C,device="cuda")*0.1).to(torch.bfloat16)
    d = ops.ConvDims.infer(x,w,st,pad)
    y = x.new_empty((d.N,d.Ho,d.Wo,d.K)); dy = torch.randn_like(y)
    dw = torch.zeros(K,R,R,C,device="cuda",dtype=torch.float32)
    for tag,fn in [("fwd",lambda: ops.conv_fwd(x,w,y,d)),
                   ("wgrad",lambda: ops.conv_wgrad(x,dy,dw,d))]:
        for _ in range(5): fn()
        torch.cuda.synchronize()
        s,e = torch.cuda.Event(True),torch.cuda.Event(True); s.record()
        for _ in range(50): fn()
        e.record(); torch.cuda.synchronize()
        print(f"stem {H}x{W} R{R}s{st} {tag}: {s.elapsed_time(e)/50*1000:.1f} us")
