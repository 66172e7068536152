import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time, json
from torchrec_amd import ops
ops.hip_ops()

def t(fn, iters=50):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters*1e6

for N in (1024, 512, 256):
    y = torch.randn(8192, N, device="cuda", dtype=torch.bfloat16).relu()
    dy = torch.randn(8192, N, device="cuda", dtype=torch.bfloat16)
    fused = t(lambda: torch.ops.trec_amd.relu_bwd_col_sum(dy, y))
    def eager():
        g = dy * (y > 0)
        return g, g.sum(0)
    eag = t(eager)
    print(json.dumps({"N": N, "fused_us": round(fused,1), "eager_us": round(eag,1)}))
