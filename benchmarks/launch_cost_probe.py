import os, sys, time, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

a = torch.randn(8192, 512, device="cuda", dtype=torch.bfloat16)
w = torch.randn(256, 512, device="cuda", dtype=torch.bfloat16)
b = torch.randn(256, device="cuda", dtype=torch.bfloat16)
x = torch.randn(8192, 256, device="cuda", dtype=torch.bfloat16)

def enqueue_cost(fn, n=300):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    t1 = time.perf_counter()          # enqueue-only (no sync)
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    return (t1-t0)/n*1e6, (t2-t0)/n*1e6

for name, fn in [
    ("addmm", lambda: torch.addmm(b, a, w.t())),
    ("mm", lambda: a @ w.t()),
    ("relu_", lambda: x.relu_()),
    ("add_", lambda: x.add_(1.0)),
]:
    host, total = enqueue_cost(fn)
    print(json.dumps({"op": name, "host_us": round(host,1), "wall_us": round(total,1)}))
