import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
for nt in (1, 0):
    for wgs in (1024, 2048, 4096, 8192):
        g = core.bench_membw(nt=nt, mb=2048, wgs=wgs, iters=8)
        print(f"nt={nt} wgs={wgs}: {g:8.0f} GB/s")
