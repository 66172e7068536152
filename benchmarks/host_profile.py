"""cProfile the eager-pipeline bench step (host-bound path used at N>1)."""
import cProfile, pstats, io, os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["TREC_NO_HIPGRAPH"] = "1"
sys.argv = ["bench.py", "--steps", "120", "--warmup", "20"]
import bench
pr = cProfile.Profile()
import torch

orig = bench.run_bench
def patched(*a, **k):
    pr.enable()
    try:
        orig(*a, **k)
    finally:
        pr.disable()
patched.__wrapped__ = orig
bench.run_bench = patched
bench.run_bench(1, 120, 20, 8192, 1.0)
s = io.StringIO()
ps = pstats.Stats(pr, stream=s).sort_stats("cumulative")
ps.print_stats(45)
print(s.getvalue())
