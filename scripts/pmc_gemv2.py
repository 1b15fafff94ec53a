import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
core.bench_gemv(3, 28672, 4096, 1, 1, 10)   # gateup DQ4K PRE_RMS
