import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from crowdllama_amd.ops import get_core
core = get_core()
# a few iterations of the two hot shapes only (keep the PMC file small)
core.bench_gemv(3, 4096, 4096, 1, 0, 10)    # DQ4K o-proj, PRE_NONE
core.bench_gemv(4, 128256, 4096, 1, 0, 3)   # DQ6K head
