import sys, torch
sys.path.insert(0, ".")
from flashy_amd import ops
for (N,H,W,C,K,R) in [(64,16,16,64,128,3),(64,32,32,64,64,3),(16,16,16,64,128,1)]:
    g = torch.Generator(device="cuda").manual_seed(0)
    x = torch.randn(N,H,W,C,device="cuda",generator=g).to(torch.bfloat16)
    w = (torch.randn(K,R,R,C,device="cuda",generator=g)*0.1).to(torch.bfloat16)
    pad = R//2
    d = ops.ConvDims.infer(x,w,1,pad)
    y = x.new_empty((d.N,d.Ho,d.Wo,d.K))
    st = ops.conv_fwd(x,w,y,d,want_stats=True)
    torch.cuda.synchronize()
    if st is None: print((N,H,W,C,K,R), "splitk - skipped"); continue
    p, ms = st
    s = p[:K*ms].view(K,ms).sum(1); s2 = p[K*ms:].view(K,ms).sum(1)
    yf = y.float().reshape(-1,K)
    rs, rs2 = yf.sum(0), (yf*yf).sum(0)
    es = (s-rs).abs(); es2 = (s2-rs2).abs()
    print((N,H,W,C,K,R), "ms",ms, "err_s", es.max().item(), "err_s2", es2.max().item(),
          "bad_ch_s", (es > 1 + rs.abs()*1e-3).nonzero().flatten()[:8].tolist())
