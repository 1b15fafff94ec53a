"""CPU reference codecs for the GGUF quantization formats the engine serves.

The byte layouts are the public GGML/GGUF on-disk formats (so checkpoints
interoperate with the wider GGUF ecosystem, per the north-star requirement of
keeping the GGUF checkpoint format). The quantizer here is a simple
absmax/minmax fit — any valid encoding dequantizes correctly everywhere; we
do not replicate llama.cpp's iterative scale search.

These numpy implementations are the numerics ground truth the HIP kernels
are tested against (tests/test_kquants.py, tests/test_gpu_kernels.py).

Formats (256-weight super-blocks for K-quants, 32 for Q8_0):

Q4_K (144 B / 256 weights = 4.5 bpw):
    fp16 d, fp16 dmin, u8 scales[12] (8 6-bit scale/min pairs), u8 qs[128]
    w[j*32+l] = d*sc[j]*q - dmin*m[j],  q in [0,15]
Q6_K (210 B / 256 = 6.5625 bpw):
    u8 ql[128], u8 qh[64], i8 scales[16], fp16 d
    w = d * scales[i/16] * (q - 32), q 6-bit from ql(4) | qh(2)
Q8_0 (34 B / 32 = 8.5 bpw):
    fp16 d, i8 qs[32];  w = d * q
"""

from __future__ import annotations

import enum

import numpy as np

QK_K = 256
Q4_K_BLOCK_BYTES = 144
Q6_K_BLOCK_BYTES = 210
Q8_0_BLOCK = 32
Q8_0_BLOCK_BYTES = 34


class GGMLType(enum.IntEnum):
    """GGML tensor dtype enum (subset we support) — values are the GGUF
    on-disk codes."""
    F32 = 0
    F16 = 1
    Q8_0 = 8
    Q4_K = 12
    Q6_K = 14
    BF16 = 30


_BLOCK = {
    GGMLType.F32: (1, 4),
    GGMLType.F16: (1, 2),
    GGMLType.BF16: (1, 2),
    GGMLType.Q8_0: (Q8_0_BLOCK, Q8_0_BLOCK_BYTES),
    GGMLType.Q4_K: (QK_K, Q4_K_BLOCK_BYTES),
    GGMLType.Q6_K: (QK_K, Q6_K_BLOCK_BYTES),
}


def type_block_bytes(t: GGMLType) -> tuple[int, int]:
    """-> (elements per block, bytes per block)."""
    return _BLOCK[GGMLType(t)]


def row_bytes(t: GGMLType, n_cols: int) -> int:
    elems, nbytes = type_block_bytes(t)
    assert n_cols % elems == 0, f"row of {n_cols} not divisible by block {elems}"
    return n_cols // elems * nbytes


def _f16(x: np.ndarray) -> np.ndarray:
    return x.astype(np.float16)


# ---------------------------------------------------------------- Q8_0 ----

def quantize_q8_0(x: np.ndarray) -> np.ndarray:
    """x: (..., K) float, K % 32 == 0 -> uint8 bytes (..., K/32*34)."""
    x = np.asarray(x, dtype=np.float32)
    shape = x.shape
    assert shape[-1] % Q8_0_BLOCK == 0
    b = x.reshape(-1, Q8_0_BLOCK)
    amax = np.abs(b).max(axis=1, keepdims=True)
    d = (amax / 127.0).astype(np.float32)
    inv = np.where(d > 0, 1.0 / np.where(d == 0, 1, d), 0.0)
    q = np.clip(np.round(b * inv), -127, 127).astype(np.int8)
    out = np.empty((b.shape[0], Q8_0_BLOCK_BYTES), dtype=np.uint8)
    out[:, 0:2] = _f16(d[:, 0]).view(np.uint8).reshape(-1, 2)
    out[:, 2:] = q.view(np.uint8)
    return out.reshape(*shape[:-1], -1)


def dequantize_q8_0(raw: np.ndarray, n: int) -> np.ndarray:
    """raw: uint8 (..., nb*34) -> float32 (..., n)."""
    raw = np.asarray(raw, dtype=np.uint8)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, Q8_0_BLOCK_BYTES)
    d = b[:, 0:2].copy().view(np.float16).astype(np.float32)
    q = b[:, 2:].view(np.int8).astype(np.float32)
    y = (d * q).reshape(*lead, -1)
    return y[..., :n]


# ---------------------------------------------------------------- Q4_K ----

def _pack_q4k_scales(sc: np.ndarray, mn: np.ndarray) -> np.ndarray:
    """sc, mn: (nb, 8) uint8 6-bit values -> (nb, 12) packed bytes.

    Layout (ggml): j<4: scales[j] = sc[j] | (sc[j+4] high 2 bits << 6)... the
    canonical decode is:
      j < 4:  sc[j] = s[j] & 63;          mn[j] = s[j+4] & 63
      j >= 4: sc[j] = (s[j+4] & 0xF) | ((s[j-4] >> 6) << 4)
              mn[j] = (s[j+4] >> 4)  | ((s[j]   >> 6) << 4)
    """
    nb = sc.shape[0]
    s = np.zeros((nb, 12), dtype=np.uint8)
    for j in range(4):
        s[:, j] = (sc[:, j] & 63) | ((sc[:, j + 4] >> 4) << 6)
        s[:, j + 4] = (mn[:, j] & 63) | ((mn[:, j + 4] >> 4) << 6)
        s[:, j + 8] = (sc[:, j + 4] & 0xF) | ((mn[:, j + 4] & 0xF) << 4)
    return s


def _unpack_q4k_scales(s: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
    nb = s.shape[0]
    sc = np.zeros((nb, 8), dtype=np.uint8)
    mn = np.zeros((nb, 8), dtype=np.uint8)
    for j in range(4):
        sc[:, j] = s[:, j] & 63
        mn[:, j] = s[:, j + 4] & 63
        sc[:, j + 4] = (s[:, j + 8] & 0xF) | ((s[:, j] >> 6) << 4)
        mn[:, j + 4] = (s[:, j + 8] >> 4) | ((s[:, j + 4] >> 6) << 4)
    return sc, mn


def quantize_q4_k(x: np.ndarray) -> np.ndarray:
    """x: (..., K) float, K % 256 == 0 -> uint8 (..., K/256*144)."""
    x = np.asarray(x, dtype=np.float32)
    shape = x.shape
    assert shape[-1] % QK_K == 0
    b = x.reshape(-1, 8, 32)                      # (nb, sub, 32)
    nb = b.shape[0]
    bmin = np.minimum(b.min(axis=2), 0.0)         # (nb, 8)  m >= 0 in w=d*sc*q - dmin*m
    bmax = np.maximum(b.max(axis=2), 0.0)
    scale = (bmax - bmin) / 15.0                  # per-sub-block scale
    mins = -bmin                                  # >= 0
    # super-block 6-bit quantization of scale and min
    d = scale.max(axis=1) / 63.0                  # (nb,)
    dmin = mins.max(axis=1) / 63.0
    d16 = _f16(d); dmin16 = _f16(dmin)
    d_r = d16.astype(np.float32); dmin_r = dmin16.astype(np.float32)
    with np.errstate(divide="ignore", invalid="ignore"):
        sc = np.where(d_r[:, None] > 0, np.round(scale / d_r[:, None]), 0)
        mn = np.where(dmin_r[:, None] > 0, np.round(mins / dmin_r[:, None]), 0)
    sc = np.clip(sc, 0, 63).astype(np.uint8)
    mn = np.clip(mn, 0, 63).astype(np.uint8)
    # quantize weights with the RECONSTRUCTED scales (d*sc)
    eff_scale = d_r[:, None] * sc                 # (nb, 8)
    eff_min = dmin_r[:, None] * mn
    with np.errstate(divide="ignore", invalid="ignore"):
        q = np.round((b + eff_min[:, :, None]) / eff_scale[:, :, None])
    q = np.where(eff_scale[:, :, None] > 0, q, 0)
    q = np.clip(q, 0, 15).astype(np.uint8)        # (nb, 8, 32)
    # pack nibbles: bytes 32*j..32*j+31 hold sub-block 2j (lo) | 2j+1 (hi)
    lo = q[:, 0::2, :]                            # (nb, 4, 32)
    hi = q[:, 1::2, :]
    qs = (lo | (hi << 4)).reshape(nb, 128)
    out = np.empty((nb, Q4_K_BLOCK_BYTES), dtype=np.uint8)
    out[:, 0:2] = d16.view(np.uint8).reshape(-1, 2)
    out[:, 2:4] = dmin16.view(np.uint8).reshape(-1, 2)
    out[:, 4:16] = _pack_q4k_scales(sc, mn)
    out[:, 16:] = qs
    return out.reshape(*shape[:-1], -1)


def dequantize_q4_k(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.asarray(raw, dtype=np.uint8)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, Q4_K_BLOCK_BYTES)
    nb = b.shape[0]
    d = b[:, 0:2].copy().view(np.float16).astype(np.float32)[:, 0]
    dmin = b[:, 2:4].copy().view(np.float16).astype(np.float32)[:, 0]
    sc, mn = _unpack_q4k_scales(b[:, 4:16])
    qs = b[:, 16:]
    q = np.zeros((nb, 8, 32), dtype=np.float32)
    qb = qs.reshape(nb, 4, 32)
    q[:, 0::2, :] = (qb & 0xF).astype(np.float32)
    q[:, 1::2, :] = (qb >> 4).astype(np.float32)
    w = (d[:, None, None] * sc[:, :, None].astype(np.float32) * q
         - dmin[:, None, None] * mn[:, :, None].astype(np.float32))
    y = w.reshape(*lead, -1)
    return y[..., :n]


# ---------------------------------------------------------------- Q6_K ----

def quantize_q6_k(x: np.ndarray) -> np.ndarray:
    """x: (..., K) float, K % 256 == 0 -> uint8 (..., K/256*210)."""
    x = np.asarray(x, dtype=np.float32)
    shape = x.shape
    assert shape[-1] % QK_K == 0
    b = x.reshape(-1, 16, 16)                     # 16 sub-blocks of 16
    nb = b.shape[0]
    amax = np.abs(b).max(axis=2)                  # (nb, 16)
    scale = amax / 31.0                           # q in [-32, 31] -> use 31
    d = scale.max(axis=1) / 127.0                 # int8 scales
    d16 = _f16(d); d_r = d16.astype(np.float32)
    with np.errstate(divide="ignore", invalid="ignore"):
        sc = np.where(d_r[:, None] > 0, np.round(scale / d_r[:, None]), 0)
    sc = np.clip(sc, -128, 127).astype(np.int8)   # (nb, 16)
    eff = d_r[:, None] * sc.astype(np.float32)
    with np.errstate(divide="ignore", invalid="ignore"):
        q = np.round(b / eff[:, :, None])
    q = np.where(eff[:, :, None] != 0, q, 0)
    q = np.clip(q, -32, 31).astype(np.int32) + 32  # [0, 63]
    q = q.reshape(nb, QK_K).astype(np.uint8)
    # pack: two 128-halves; within each: l in 0..31:
    #  q1=q[l], q2=q[l+32], q3=q[l+64], q4=q[l+96]
    #  ql[l]    = (q1&0xF) | ((q3&0xF)<<4)
    #  ql[l+32] = (q2&0xF) | ((q4&0xF)<<4)
    #  qh[l]    = (q1>>4) | ((q2>>4)<<2) | ((q3>>4)<<4) | ((q4>>4)<<6)
    qh_ = q.reshape(nb, 2, 4, 32)                 # (nb, half, quarter, 32)
    ql = np.empty((nb, 2, 64), dtype=np.uint8)
    ql[:, :, 0:32] = (qh_[:, :, 0] & 0xF) | ((qh_[:, :, 2] & 0xF) << 4)
    ql[:, :, 32:64] = (qh_[:, :, 1] & 0xF) | ((qh_[:, :, 3] & 0xF) << 4)
    qh = ((qh_[:, :, 0] >> 4) | ((qh_[:, :, 1] >> 4) << 2)
          | ((qh_[:, :, 2] >> 4) << 4) | ((qh_[:, :, 3] >> 4) << 6))
    out = np.empty((nb, Q6_K_BLOCK_BYTES), dtype=np.uint8)
    out[:, 0:128] = ql.reshape(nb, 128)
    out[:, 128:192] = qh.reshape(nb, 64)
    out[:, 192:208] = sc.view(np.uint8)
    out[:, 208:210] = d16.view(np.uint8).reshape(-1, 2)
    return out.reshape(*shape[:-1], -1)


def dequantize_q6_k(raw: np.ndarray, n: int) -> np.ndarray:
    raw = np.asarray(raw, dtype=np.uint8)
    lead = raw.shape[:-1]
    b = raw.reshape(-1, Q6_K_BLOCK_BYTES)
    nb = b.shape[0]
    ql = b[:, 0:128].reshape(nb, 2, 64)
    qh = b[:, 128:192].reshape(nb, 2, 32)
    sc = b[:, 192:208].view(np.int8).astype(np.float32)
    d = b[:, 208:210].copy().view(np.float16).astype(np.float32)[:, 0]
    q = np.empty((nb, 2, 4, 32), dtype=np.int32)
    q[:, :, 0] = (ql[:, :, 0:32] & 0xF) | (((qh >> 0) & 3) << 4)
    q[:, :, 1] = (ql[:, :, 32:64] & 0xF) | (((qh >> 2) & 3) << 4)
    q[:, :, 2] = (ql[:, :, 0:32] >> 4) | (((qh >> 4) & 3) << 4)
    q[:, :, 3] = (ql[:, :, 32:64] >> 4) | (((qh >> 6) & 3) << 4)
    q = q.reshape(nb, QK_K).astype(np.float32) - 32.0
    # scale index: element e (0..255) -> sub-block e//16
    scl = np.repeat(sc, 16, axis=1)               # (nb, 256)
    y = (d[:, None] * scl * q).reshape(*lead, -1)
    return y[..., :n]


# ------------------------------------------------------------- dispatch ----

def quantize(x: np.ndarray, t: GGMLType) -> np.ndarray:
    t = GGMLType(t)
    if t == GGMLType.F32:
        return np.ascontiguousarray(x, dtype=np.float32).view(np.uint8)
    if t == GGMLType.F16:
        return np.ascontiguousarray(x, dtype=np.float16).view(np.uint8)
    if t == GGMLType.BF16:
        f = np.ascontiguousarray(x, dtype=np.float32).view(np.uint32)
        # round-to-nearest-even bf16
        rounded = ((f + 0x7FFF + ((f >> 16) & 1)) >> 16).astype(np.uint16)
        return rounded.view(np.uint8)
    if t == GGMLType.Q8_0:
        return quantize_q8_0(x)
    if t == GGMLType.Q4_K:
        return quantize_q4_k(x)
    if t == GGMLType.Q6_K:
        return quantize_q6_k(x)
    raise ValueError(f"unsupported type {t}")


def dequantize(raw: np.ndarray, t: GGMLType, n: int) -> np.ndarray:
    t = GGMLType(t)
    if t == GGMLType.F32:
        return raw.view(np.float32)[..., :n].astype(np.float32)
    if t == GGMLType.F16:
        return raw.view(np.float16)[..., :n].astype(np.float32)
    if t == GGMLType.BF16:
        u = raw.view(np.uint16)[..., :n].astype(np.uint32) << 16
        return u.view(np.float32)
    if t == GGMLType.Q8_0:
        return dequantize_q8_0(raw, n)
    if t == GGMLType.Q4_K:
        return dequantize_q4_k(raw, n)
    if t == GGMLType.Q6_K:
        return dequantize_q6_k(raw, n)
    raise ValueError(f"unsupported type {t}")
