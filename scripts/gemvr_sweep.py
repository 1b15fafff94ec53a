import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
cases = [
    ("qkv  rms  6144x4096", 3, 6144, 4096, 1),
    ("gateup none 28672x4096", 3, 28672, 4096, 0),
    ("o    none 4096x4096", 3, 4096, 4096, 0),
    ("down none 4096x14336", 3, 4096, 14336, 0),
    ("head q6k none 128256x4096", 4, 128256, 4096, 0),
]
wgs = sys.argv[1] if len(sys.argv) > 1 else ""
from crowdllama_amd.ops import get_core
core = get_core()
print(f"CLA_GEMVR_WGS={os.environ.get('CLA_GEMVR_WGS','auto')}")
for name, dt, N, K, pre in cases:
    ms = core.bench_gemv(dt, N, K, 1, pre, 50)
    bpr = K//256*128 + K//256*16 if dt == 3 else K + K//256*32
    gbs = N * bpr / ms / 1e6
    print(f"{name:28s} {ms*1000:8.1f}us {gbs:8.0f} GB/s")
