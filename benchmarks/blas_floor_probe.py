import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time, json

def t(fn, iters=100):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

shapes = [(256,256,256),(8192,13,512),(8192,512,256),(8192,479,1024),(8192,1024,1024)]
for lib in ("cublaslt", "cublas"):
    try:
        torch.backends.cuda.preferred_blas_library(lib)
    except Exception as e:
        print("skip", lib, e); continue
    for (m,k,n) in shapes:
        a = torch.randn(m,k,device="cuda",dtype=torch.bfloat16)
        b = torch.randn(k,n,device="cuda",dtype=torch.bfloat16)
        us = t(lambda: a @ b)
        print(json.dumps({"lib":lib,"shape":[m,k,n],"us":round(us,1),"tf":round(2*m*k*n/us/1e6)}))
    # wgrad TN
    g = torch.randn(8192,1024,device="cuda",dtype=torch.bfloat16)
    x = torch.randn(8192,1024,device="cuda",dtype=torch.bfloat16)
    print(json.dumps({"lib":lib,"wgrad":[8192,1024,1024],"us":round(t(lambda: g.t() @ x),1)}))
