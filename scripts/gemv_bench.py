import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
DQ4K, DQ6K = 3, 4
NONE, RMS, SILU = 0, 1, 2
cases = [
    ("qkv    q4k rms  6144x4096", DQ4K, 6144, 4096, RMS),
    ("qkv    q4k none 6144x4096", DQ4K, 6144, 4096, NONE),
    ("gateup q4k rms  28672x4096", DQ4K, 28672, 4096, RMS),
    ("gateup q4k none 28672x4096", DQ4K, 28672, 4096, NONE),
    ("down   q4k silu 4096x14336", DQ4K, 4096, 14336, SILU),
    ("down   q4k none 4096x14336", DQ4K, 4096, 14336, NONE),
    ("o      q4k none 4096x4096", DQ4K, 4096, 4096, NONE),
    ("head   q6k rms  128256x4096", DQ6K, 128256, 4096, RMS),
    ("head   q6k none 128256x4096", DQ6K, 128256, 4096, NONE),
]
print(f"{'case':32s} {'ms':>8s} {'GB/s':>8s}")
for name, dt, N, K, pre in cases:
    ms = core.bench_gemv(dt, N, K, 1, pre, 50)
    bpr = K//256*128 + K//256*16 if dt == 3 else K + K//256*32
    gbs = N * bpr / ms / 1e6
    print(f"{name:32s} {ms*1000:8.1f}us {gbs:8.0f}")
