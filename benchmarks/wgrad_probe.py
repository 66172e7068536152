"""Probe wgrad (dW = dY^T @ X) formulations + TunableOp on the DLRM shapes."""
import torch, time, json, os

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

B = 8192
shapes = [(479,1024),(1024,1024),(1024,512),(512,256),(13,512),(512,256),(128,479)]
for (k,n) in shapes:
    g = torch.randn(B,n,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(B,k,device="cuda",dtype=torch.bfloat16)
    flop = 2*B*k*n
    r = {}
    r["gt_x"] = t(lambda: g.t() @ x)                      # dW directly [n,k]
    r["xt_g"] = t(lambda: x.t() @ g)                      # dW^T [k,n]
    r["mm_out"] = None
    o = torch.empty(n,k,device="cuda",dtype=torch.bfloat16)
    r["mm_out"] = t(lambda: torch.mm(g.t(), x, out=o))
    r32 = torch.empty(n,k,device="cuda",dtype=torch.float32)
    try:
        r["f32out"] = t(lambda: torch.mm(g.t().float(), x.float()))
    except Exception:
        r["f32out"] = -1
    print(json.dumps({"k":k,"n":n,"tf_at": {m: round(flop/us/1e6) for m,us in r.items() if us and us>0}, "us": {m: round(us,1) for m,us in r.items() if us}}))

# TunableOp tuning pass on the same shapes
tun = torch.cuda.tunable
tun.enable(True); tun.tuning_enable(True)
for (k,n) in shapes:
    g = torch.randn(B,n,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(B,k,device="cuda",dtype=torch.bfloat16)
    for _ in range(3): (g.t() @ x)
torch.cuda.synchronize()
tun.tuning_enable(False)
for (k,n) in shapes:
    g = torch.randn(B,n,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(B,k,device="cuda",dtype=torch.bfloat16)
    us = t(lambda: g.t() @ x)
    print(json.dumps({"tuned_wgrad":[k,n],"us":round(us,1),"tflops":round(2*B*k*n/us/1e6)}))
tun.write_file("gpurun_out/tunableop_wgrad.csv")
