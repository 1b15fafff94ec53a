import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
core.bench_gemm(3, 16, 6144, 4096, 20)   # decode qkv tile, Q4_K, B=16
