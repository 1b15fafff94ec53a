"""K-quant codec tests: round-trip accuracy and byte-layout invariants.

These are the CPU ground truth for the HIP dequant kernels (which are
compared against these codecs in tests/test_gpu_kernels.py).
"""

import numpy as np
import pytest

from crowdllama_amd.quant import (
    GGMLType, dequantize, quantize,
    quantize_q4_k, dequantize_q4_k,
    quantize_q6_k, dequantize_q6_k,
    quantize_q8_0, dequantize_q8_0,
    row_bytes,
)


@pytest.mark.parametrize("n", [32, 256, 4096])
def test_q8_0_roundtrip(n):
    rng = np.random.default_rng(0)
    x = rng.standard_normal(n).astype(np.float32)
    raw = quantize_q8_0(x)
    assert raw.nbytes == n // 32 * 34
    y = dequantize_q8_0(raw, n)
    err = np.abs(x - y).max()
    scale = np.abs(x).max() / 127
    assert err <= scale * 0.51 + 1e-6


@pytest.mark.parametrize("n", [256, 1024, 4096])
def test_q4_k_roundtrip(n):
    rng = np.random.default_rng(1)
    x = rng.standard_normal(n).astype(np.float32)
    raw = quantize_q4_k(x)
    assert raw.nbytes == n // 256 * 144
    y = dequantize_q4_k(raw, n)
    # 4-bit asymmetric: error bounded by ~ (range/15)/2 plus 6-bit scale error
    rel = np.abs(x - y).max() / np.abs(x).max()
    assert rel < 0.10
    # correlation should be very high
    c = np.corrcoef(x, y)[0, 1]
    assert c > 0.99


@pytest.mark.parametrize("n", [256, 1024, 4096])
def test_q6_k_roundtrip(n):
    rng = np.random.default_rng(2)
    x = rng.standard_normal(n).astype(np.float32)
    raw = quantize_q6_k(x)
    assert raw.nbytes == n // 256 * 210
    y = dequantize_q6_k(raw, n)
    rel = np.abs(x - y).max() / np.abs(x).max()
    assert rel < 0.05
    c = np.corrcoef(x, y)[0, 1]
    assert c > 0.999


def test_q4_k_batch_rows():
    rng = np.random.default_rng(3)
    x = rng.standard_normal((8, 512)).astype(np.float32)
    raw = quantize_q4_k(x)
    y = dequantize_q4_k(raw, 512)
    assert y.shape == (8, 512)
    for i in range(8):
        row = dequantize_q4_k(quantize_q4_k(x[i]), 512)
        np.testing.assert_array_equal(y[i], row)


def test_bf16_roundtrip():
    x = np.array([1.0, -2.5, 3.14159, 1e-8, 65504.0], dtype=np.float32)
    raw = quantize(x, GGMLType.BF16)
    y = dequantize(raw, GGMLType.BF16, len(x))
    assert np.abs((y - x) / np.maximum(np.abs(x), 1e-30)).max() < 1 / 128


def test_row_bytes():
    assert row_bytes(GGMLType.Q4_K, 4096) == 4096 // 256 * 144
    assert row_bytes(GGMLType.Q8_0, 4096) == 4096 // 32 * 34
    assert row_bytes(GGMLType.F32, 10) == 40
    with pytest.raises(AssertionError):
        row_bytes(GGMLType.Q4_K, 100)


def test_q4_k_scale_pack_unpack():
    from crowdllama_amd.quant.kquants import _pack_q4k_scales, _unpack_q4k_scales
    rng = np.random.default_rng(4)
    sc = rng.integers(0, 64, size=(16, 8)).astype(np.uint8)
    mn = rng.integers(0, 64, size=(16, 8)).astype(np.uint8)
    s = _pack_q4k_scales(sc, mn)
    sc2, mn2 = _unpack_q4k_scales(s)
    np.testing.assert_array_equal(sc, sc2)
    np.testing.assert_array_equal(mn, mn2)


def test_f16_decode_exhaustive():
    """Host f16->f32 decoder (ops/csrc/common.h) vs numpy float16 over all
    65536 bit patterns. Subnormals matter: Q6_K super-scales of
    small-magnitude weight blocks land subnormal, and an off-by-one in
    the normalization HALVED them (caught via the i8 GEMM Q6_K path)."""
    import numpy as np
    import pytest
    try:
        from crowdllama_amd.ops import get_core
        core = get_core()
    except Exception:
        pytest.skip("ops extension unavailable")
    bits = np.arange(65536, dtype=np.uint16)
    got = np.asarray(core.test_f16_decode(bits))
    want = bits.view(np.float16).astype(np.float32)
    fin = np.isfinite(want)
    np.testing.assert_array_equal(got[fin], want[fin])
    assert np.all(np.isnan(got[np.isnan(want)]))
    inf = np.isinf(want)
    np.testing.assert_array_equal(got[inf], want[inf])
