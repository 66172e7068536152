"""Probe: odd-K (479) vs padded-K GEMM shapes of the DLRM over-arch."""
import torch, time, json

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

B = 8192
for (m,k,n) in [(B,479,1024),(B,480,1024),(B,512,1024),(B,1024,1024),(B,1024,512),
                (B,512,256),(B,13,512),(B,512,256)]:
    a = torch.randn(m,k,device="cuda",dtype=torch.bfloat16)
    w = torch.randn(n,k,device="cuda",dtype=torch.bfloat16)
    us = t(lambda: torch.nn.functional.linear(a,w))
    tf = 2*m*k*n/us/1e6
    print(json.dumps({"shape":[m,k,n],"us":round(us,1),"tflops":round(tf)}))
# wgrad shapes: g^T @ x  -> [n, k]
for (m,k,n) in [(B,479,1024),(B,1024,1024),(B,1024,512)]:
    g = torch.randn(m,n,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(m,k,device="cuda",dtype=torch.bfloat16)
    us = t(lambda: g.t() @ x)
    print(json.dumps({"wgrad":[m,k,n],"us":round(us,1),"tflops":round(2*m*k*n/us/1e6)}))
# dgrad: g @ W
for (m,k,n) in [(B,479,1024),(B,1024,1024)]:
    g = torch.randn(m,n,device="cuda",dtype=torch.bfloat16)
    w = torch.randn(n,k,device="cuda",dtype=torch.bfloat16)
    us = t(lambda: g @ w)
    print(json.dumps({"dgrad":[m,k,n],"us":round(us,1),"tflops":round(2*m*k*n/us/1e6)}))
