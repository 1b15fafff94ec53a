import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from crowdllama_amd.quant import quantize_q6_k, dequantize_q6_k
from crowdllama_amd.ops import get_core
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
from test_gpu_kernels import _repack_q6k, DT_DQ6K

core = get_core()
rng = np.random.default_rng(3)
N, K, M = 192, 512, 1
for blk in [0, 1, 2, 3]:
    w = np.zeros((N, K), dtype=np.float32)
    w[:, blk*16:(blk+1)*16] = rng.standard_normal((N, 16)).astype(np.float32) * 0.1
    raw = quantize_q6_k(w)
    wref = dequantize_q6_k(raw, K).reshape(N, K)
    x = rng.standard_normal((M, K)).astype(np.float32)
    qs, hdr = _repack_q6k(raw.reshape(N, -1), N, K)
    y = core.test_gemm_i8(np.ascontiguousarray(qs), np.ascontiguousarray(hdr), x, DT_DQ6K, N, K)
    b = x.reshape(M, -1, 32)
    amax = np.abs(b).max(axis=2, keepdims=True)
    xq = (np.rint(b * np.where(amax>0, 127.0/np.where(amax==0,1,amax), 0.0)) * (amax/127.0)).reshape(M, K)
    yref = xq @ wref.T
    r = y[0][:8] / np.where(np.abs(yref[0][:8]) > 1e-9, yref[0][:8], 1)
    print("blk", blk, "ratio y/yref first8:", np.round(r, 4))
