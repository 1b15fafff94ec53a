"""Diagnose the k_gemv_r PRE_RMS mismatch: structured inputs that separate
a wrong inv scale (uniform ratio) from a wrong x/gw mapping (per-row or
per-lane pattern)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
from crowdllama_amd.quant import quantize_q4_k, dequantize_q4_k
from crowdllama_amd.ops import get_core

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))
from test_gpu_kernels import _repack_q4k, DT_DQ4K, PRE_RMS, PRE_NONE

core = get_core()
rng = np.random.default_rng(0)
N, K = 192, 2048
w = rng.standard_normal((N, K)).astype(np.float32) * 0.1
raw = quantize_q4_k(w)
wref = dequantize_q4_k(raw, K).reshape(N, K)
qs, hdr = _repack_q4k(raw.reshape(N, -1), N, K)
qs = np.ascontiguousarray(qs); hdr = np.ascontiguousarray(hdr)

for tag, gw, x in [
    ("gw=1 x=rand", np.ones(K, np.float32),
     rng.standard_normal((1, K)).astype(np.float32)),
    ("gw=rand x=1", rng.standard_normal(K).astype(np.float32),
     np.ones((1, K), np.float32)),
    ("gw=rand x=rand", rng.standard_normal(K).astype(np.float32),
     rng.standard_normal((1, K)).astype(np.float32)),
]:
    y = core.test_gemv(qs, hdr, x, DT_DQ4K, N, K, PRE_RMS, gw)
    xe = x / np.sqrt((x * x).mean(axis=1, keepdims=True) + 1e-5) * gw
    yref = xe @ wref.T
    # control: feed the pre-normalized vector through PRE_NONE
    y0 = core.test_gemv(qs, hdr, xe.astype(np.float32), DT_DQ4K, N, K,
                        PRE_NONE, np.zeros(0, np.float32))
    d = np.abs(y - yref).max(); d0 = np.abs(y0 - yref).max()
    ratio = (y / (yref + 1e-30))[0, :8]
    print(f"{tag}: rms-path maxdiff {d:.5f}  none-control {d0:.6f}  "
          f"ratio[0:8]={np.round(ratio, 4)}")
    bad = np.where(np.abs(y - yref)[0] > 5e-3)[0]
    print(f"   bad rows ({len(bad)}): {bad[:16]}")
