// Common host/device helpers for the crowdllama-amd HIP engine (gfx950).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <cstdio>
#include <stdexcept>
#include <string>

#define HIP_CHECK(expr)                                                       \
    do {                                                                      \
        hipError_t _e = (expr);                                               \
        if (_e != hipSuccess) {                                               \
            throw std::runtime_error(std::string("HIP error: ") +             \
                                     hipGetErrorString(_e) + " at " +         \
                                     __FILE__ + ":" + std::to_string(__LINE__)); \
        }                                                                     \
    } while (0)

namespace cla {

// GGML dtype codes (subset; matches quant/kquants.py GGMLType)
enum class GT : int32_t {
    F32 = 0,
    F16 = 1,
    Q8_0 = 8,
    Q4_K = 12,
    Q6_K = 14,
    BF16 = 30,
};

// Device-side weight encodings after upload-time repack (see engine.cpp).
// The disk formats are AoS blocks; on device we split quant payload ("qs")
// from block headers ("hdr") so GEMV/GEMM sweeps issue perfectly coalesced
// 16-B chunk loads while headers ride the L1/L2 broadcast path.
enum class DT : int32_t {
    F32 = 0,
    F16 = 1,
    BF16 = 2,
    DQ4K = 3,   // qs: [nsb][128B] nibbles; hdr: [nsb][4 pairs][8B]
                //   {f16 d, f16 dmin, u8 sc_lo, u8 mn_lo, u8 sc_hi, u8 mn_hi}
                //   (6-bit scales pre-decoded at upload: branchless kernel)
    DQ6K = 4,   // qs: [K] int8 (q-32 applied); hdr: [nsb][32B] {f16 d, i8 sc[16], pad}
    DQ8 = 5,    // qs: [K] int8; hdr: [K/32] f16 d
};

static constexpr int QK_K = 256;

__host__ __device__ inline int64_t dqs_row_bytes(DT t, int64_t k) {
    switch (t) {
        case DT::F32: return k * 4;
        case DT::F16: case DT::BF16: return k * 2;
        case DT::DQ4K: return k / QK_K * 128;
        case DT::DQ6K: return k;
        case DT::DQ8: return k;
    }
    return 0;
}

__host__ __device__ inline int64_t dhdr_row_bytes(DT t, int64_t k) {
    switch (t) {
        case DT::F32: case DT::F16: case DT::BF16: return 0;
        case DT::DQ4K: return k / QK_K * 32;
        case DT::DQ6K: return k / QK_K * 32;
        case DT::DQ8: return k / 32 * 2;
    }
    return 0;
}

// Transposed, PRE-DECODED header copy for the i8 GEMM's LDS-DMA scale
// staging (gemm_i8.hip): one k-window's scales for all rows contiguous,
// already in f32 so the drain does no header math.
// DQ4K: [K/32][N][{f32 d*sc, f32 dmin*mn}]; DQ8: [K/32][N][f32 d];
// DQ6K: [K/16][N][f32 d*sc16] (per-16 scales -> masked-MFMA K=16 drains).
// Bytes per row (total = n * this):
__host__ __device__ inline int64_t dhdr2_row_bytes(DT t, int64_t k) {
    switch (t) {
        case DT::DQ4K: return k / 32 * 8;
        case DT::DQ8: return k / 32 * 4;
        case DT::DQ6K: return k / 16 * 4;
        default: return 0;
    }
}

// host-side f16 bits -> f32 (GGUF block scales)
inline float f16_bits_to_f32_host(uint16_t h) {
    const uint32_t s = (uint32_t)(h >> 15) & 1u;
    const uint32_t e = (uint32_t)(h >> 10) & 31u;
    const uint32_t m = (uint32_t)h & 1023u;
    uint32_t out;
    if (e == 0) {
        if (m == 0) {
            out = s << 31;
        } else {
            // subnormal: value = (mm/1024) * 2^(-14-k) after k shifts
            // normalize; biased exponent 113 - k (an off-by-one here
            // HALVED subnormal scales — caught by Q6_K super-scales,
            // which land subnormal for small-magnitude weights; pinned
            // by test_f16_decode_exhaustive)
            int ex = 0;
            uint32_t mm = m;
            while (!(mm & 1024u)) { mm <<= 1; ex--; }
            out = (s << 31) | ((uint32_t)(127 - 15 + 1 + ex) << 23) |
                  ((mm & 1023u) << 13);
        }
    } else if (e == 31) {
        out = (s << 31) | 0x7F800000u | (m << 13);
    } else {
        out = (s << 31) | ((e - 15 + 127) << 23) | (m << 13);
    }
    union { uint32_t u; float f; } v;
    v.u = out;
    return v.f;
}

// Decode one row-range of hdr into the hdr2 layout above (shared by
// Engine::upload_pack and the gemm_i8 test harness).
inline void build_hdr2_rows(DT t, const uint8_t* hdr, int64_t hrb,
                            int64_t rows_total, int64_t k, int64_t r_lo,
                            int64_t r_hi, uint8_t* out) {
    const int64_t nb = k / 32;
    if (t == DT::DQ4K) {
        for (int64_t r = r_lo; r < r_hi; r++) {
            const uint8_t* h = hdr + r * hrb;
            for (int64_t kg = 0; kg < nb; kg++) {
                const uint8_t* e = h + (kg >> 3) * 32 + ((kg & 7) >> 1) * 8;
                const float d = f16_bits_to_f32_host(
                    *reinterpret_cast<const uint16_t*>(e));
                const float dmin = f16_bits_to_f32_host(
                    *reinterpret_cast<const uint16_t*>(e + 2));
                const int half = (int)(kg & 1);
                float* dst = reinterpret_cast<float*>(
                    out + ((size_t)kg * rows_total + r) * 8);
                dst[0] = d * (float)e[4 + 2 * half];
                dst[1] = dmin * (float)e[5 + 2 * half];
            }
        }
    } else if (t == DT::DQ8) {
        for (int64_t r = r_lo; r < r_hi; r++) {
            const uint16_t* h = reinterpret_cast<const uint16_t*>(
                hdr + r * hrb);
            for (int64_t b = 0; b < nb; b++)
                *reinterpret_cast<float*>(
                    out + ((size_t)b * rows_total + r) * 4) =
                    f16_bits_to_f32_host(h[b]);
        }
    } else if (t == DT::DQ6K) {
        // device hdr per row: [nsb][32B]{f16 d, pad2, i8 sc[16], pad};
        // hdr2 entry per k16 block s: f32 d*sc[s%16]
        const int64_t n16 = k / 16;
        for (int64_t r = r_lo; r < r_hi; r++) {
            const uint8_t* h = hdr + r * hrb;
            for (int64_t s = 0; s < n16; s++) {
                const uint8_t* e = h + (s >> 4) * 32;
                const float d = f16_bits_to_f32_host(
                    *reinterpret_cast<const uint16_t*>(e));
                const int8_t sc =
                    reinterpret_cast<const int8_t*>(e)[4 + (s & 15)];
                *reinterpret_cast<float*>(
                    out + ((size_t)s * rows_total + r) * 4) = d * (float)sc;
            }
        }
    }
}

// i8-GEMM tile geometry (gemm_i8.hip) — shared with the upload-time
// tiled repack below.
constexpr int I8G_BN = 128;   // rows (output features) per tile
constexpr int I8G_BK = 64;    // k per tile

// Raw quant bytes per row per BK window.
__host__ __device__ inline int64_t i8g_rawb(DT t) {
    return t == DT::DQ4K ? I8G_BK / 2 : I8G_BK;
}

// GEMM-tiled weight copy: [ceil(N/128)][K/64][128 rows][RAWB] — each
// (n-block, k-window) tile is contiguous (= the kernel's LDS image), so
// the weight DMA reads whole cachelines. The row-major qs layout reads
// only 32 B per 128 B line at BK=64 windows (4x HBM over-fetch — the
// round-2 ~1.5 TB/s wall on both GEMM kernels). Rows past N are zeros.
__host__ __device__ inline int64_t dqs2_bytes(DT t, int64_t n, int64_t k) {
    if (t != DT::DQ4K && t != DT::DQ8 && t != DT::DQ6K) return 0;
    const int64_t nb = (n + I8G_BN - 1) / I8G_BN;
    return nb * (k / I8G_BK) * I8G_BN * i8g_rawb(t);
}

// Build the tiled copy for a row range of the (already repacked,
// row-major) qs buffer.
inline void build_qs2_rows(DT t, const uint8_t* qs, int64_t qs_rb,
                           int64_t n, int64_t k, int64_t r_lo, int64_t r_hi,
                           uint8_t* out) {
    const int64_t rawb = i8g_rawb(t);
    const int64_t ktiles = k / I8G_BK;
    const int64_t tile_bytes = I8G_BN * rawb;
    for (int64_t r = r_lo; r < r_hi; r++) {
        const int64_t nb = r / I8G_BN, rl = r % I8G_BN;
        for (int64_t kt = 0; kt < ktiles; kt++) {
            uint8_t* dst = out + (nb * ktiles + kt) * tile_bytes + rl * rawb;
            if (r >= n) {
                for (int64_t b = 0; b < rawb; b++) dst[b] = 0;
                continue;
            }
            const int64_t kb = kt * I8G_BK;
            const uint8_t* src;
            if (t == DT::DQ4K) {
                // qs row layout: [nsb][128B]; BK window = q-group of 64 =
                // 32 contiguous bytes at sb*128 + q*32
                src = qs + r * qs_rb + (kb >> 8) * 128 + ((kb & 255) >> 6) * 32;
            } else {  // DQ8 / DQ6K: row-major int8
                src = qs + r * qs_rb + kb;
            }
            for (int64_t b = 0; b < rawb; b++) dst[b] = src[b];
        }
    }
}

#ifdef __HIPCC__
// Fused activation block-quantizer (8 lanes per 32-block): emits the int8
// row and the interleaved transposed scales ([K/32][2][M4]) the i8 GEMM's
// DMA staging expects (gemm_i8.hip). Call with jl = tid&7 and the block's
// 4 values in v; all 8 lanes of the group must call together.
__device__ __forceinline__ void quant_block_emit(
    float4 v, int jl, int blk, int m, int K, int M4,
    int8_t* __restrict__ xq, float* __restrict__ xsc) {
    float amax = fmaxf(fmaxf(fabsf(v.x), fabsf(v.y)),
                       fmaxf(fabsf(v.z), fabsf(v.w)));
    #pragma unroll
    for (int off = 1; off < 8; off <<= 1)
        amax = fmaxf(amax, __shfl_xor(amax, off, 64));
    const float scale = amax / 127.f;
    const float rinv = amax > 0.f ? 127.f / amax : 0.f;
    const int q0 = (int)rintf(v.x * rinv);
    const int q1 = (int)rintf(v.y * rinv);
    const int q2 = (int)rintf(v.z * rinv);
    const int q3 = (int)rintf(v.w * rinv);
    const uint32_t packed =
        (uint32_t)(q0 & 0xFF) | ((uint32_t)(q1 & 0xFF) << 8) |
        ((uint32_t)(q2 & 0xFF) << 16) | ((uint32_t)(q3 & 0xFF) << 24);
    int sq = q0 + q1 + q2 + q3;
    #pragma unroll
    for (int off = 1; off < 8; off <<= 1) sq += __shfl_xor(sq, off, 64);
    *reinterpret_cast<uint32_t*>(xq + (size_t)m * K + blk * 32 + jl * 4) =
        packed;
    if (jl == 0) {
        xsc[((size_t)blk * 2 + 0) * M4 + m] = scale;
        xsc[((size_t)blk * 2 + 1) * M4 + m] = scale * (float)sq;
    }
}

#endif  // __HIPCC__

// A weight matrix on device: N rows of K quantized columns.
struct WTensor {
    DT dtype = DT::F32;
    int64_t n = 0;   // rows (output features)
    int64_t k = 0;   // cols (input features)
    const void* qs = nullptr;
    const void* hdr = nullptr;
    const void* hdr2 = nullptr;  // transposed pre-decoded headers (i8 GEMM)
    const void* qs2 = nullptr;   // GEMM-tiled weight copy (i8 GEMM)
};

}  // namespace cla
