import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
DQ4K, DQ6K = 3, 4
shapes = [("qkv 6144x4096", DQ4K, 6144, 4096), ("o 4096x4096", DQ4K, 4096, 4096),
          ("gateup 28672x4096", DQ4K, 28672, 4096), ("down 4096x14336", DQ4K, 4096, 14336),
          ("head q6k 128256x4096", DQ6K, 128256, 4096)]
print(f"{'shape':24s} {'lds+pre(us)':>12s} {'global-x(us)':>12s}")
for name, dt, N, K in shapes:
    a = core.bench_gemv(dt, N, K, 1, 0, 50) * 1000
    b = core.bench_gemv_g(dt, N, K, 1, 50) * 1000
    print(f"{name:24s} {a:12.1f} {b:12.1f}   {a/b:.2f}x")
