"""Split-K via chunked bmm for the TN wgrad, plus fp32-accum variants."""
import torch, time, json

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

B = 8192
for (k,n) in [(479,1024),(1024,1024),(1024,512),(512,256),(128,479)]:
    g = torch.randn(B,n,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(B,k,device="cuda",dtype=torch.bfloat16)
    flop = 2*B*k*n
    out = {}
    for s in (4, 8, 16, 32):
        gv = g.view(s, B//s, n)
        xv = x.view(s, B//s, k)
        fn = lambda: torch.bmm(gv.transpose(1,2), xv).sum(0)
        out[f"bmm{s}"] = round(t(fn),1)
    base = t(lambda: g.t() @ x)
    best = min(out.values())
    print(json.dumps({"k":k,"n":n,"base_us":round(base,1),"bmm_us":out,
                      "best_tf": round(flop/best/1e6)}))
