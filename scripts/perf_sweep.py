"""GPU perf sweep: gemv A/B, gemm shapes, engine bench at several batches."""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
DQ4K, DQ6K, DQ8, BF16 = 3, 4, 5, 2
NONE, RMS, SILU = 0, 1, 2
print("== gemv (isolated; L3-resident for small tensors) ==")
for name, dt, N, K, pre in [
    ("qkv q4k rms 6144x4096", DQ4K, 6144, 4096, RMS),
    ("gateup q4k rms 28672x4096", DQ4K, 28672, 4096, RMS),
    ("down q4k silu 4096x14336", DQ4K, 4096, 14336, SILU),
    ("o q4k none 4096x4096", DQ4K, 4096, 4096, NONE),
    ("head q6k rms 128256x4096", DQ6K, 128256, 4096, RMS),
]:
    ms = core.bench_gemv(dt, N, K, 1, pre, 50)
    bpr = K//256*128 + K//256*16 if dt == DQ4K else (K + K//256*32 if dt == DQ6K else K+K//16)
    print(f"  {name:30s} {ms*1000:8.1f}us {N*bpr/ms/1e6:7.0f} GB/s")
print("== gemv q8 (int8-activation dot4 path) ==")
for name, dt, N, K in [("qkv q4k", DQ4K, 6144, 4096), ("gateup q4k", DQ4K, 28672, 4096),
                       ("down q4k", DQ4K, 4096, 14336), ("head q6k", DQ6K, 128256, 4096)]:
    ms = core.bench_gemv_q8(dt, N, K, 1, 50)
    bpr = K//256*128 + K//256*32 if dt == DQ4K else K + K//256*32
    print(f"  {name:30s} {ms*1000:8.1f}us {N*bpr/ms/1e6:7.0f} GB/s")
print("== gemm (M=512 prefill, M=16 batched decode) ==")
for name, dt, M, N, K in [
    ("gateup q4k M512", DQ4K, 512, 28672, 4096),
    ("qkv q4k M512", DQ4K, 512, 6144, 4096),
    ("down q4k M512", DQ4K, 512, 4096, 14336),
    ("gateup q4k M16", DQ4K, 16, 28672, 4096),
    ("head q6k M16", DQ6K, 16, 128256, 4096),
    ("bf16 4096x4096 M512", BF16, 512, 4096, 4096),
]:
    ms = core.bench_gemm(dt, M, N, K, 30)
    tf = 2.0*M*N*K/ms/1e9
    print(f"  {name:30s} {ms*1000:8.1f}us {tf:7.1f} TFLOP/s")
