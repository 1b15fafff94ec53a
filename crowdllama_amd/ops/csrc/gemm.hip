// MFMA dequant-GEMM for gfx950: C[M,N] = X[M,K] @ W[N,K]^T (+res), where W
// rows are quantized (Q4_K/Q6_K/Q8_0) or bf16/f16/f32. Used for prompt
// prefill and batched decode (the GEMV kernels cover decode B=1).
//
// Structure (guide §5 canonical anatomy, correctness-first tier):
// 128x128 tile, BK=64, 256 threads = 4 waves as 2x2, each wave a 64x64
// sub-tile = 4x4 fragments of v_mfma_f32_16x16x32_bf16. Weight tiles are
// dequantized on the fly into LDS bf16 during staging (16-B padded rows
// against ds_read_b128 bank conflicts). Fragment lane maps verified on
// hardware by mfma_probe.hip / tests/test_gpu_kernels.py.
#include "common.h"

namespace cla {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int PAD = 8;                       // bf16 elems -> 16B row pad
constexpr int LDW = BK + PAD;                // LDS row stride (elems)

__device__ __forceinline__ uint16_t f32_to_bf16b(float f) {
    union { uint32_t u; float f; } v;
    v.f = f;
    return (uint16_t)((v.u + 0x7FFF + ((v.u >> 16) & 1)) >> 16);
}

__device__ __forceinline__ float f16b_to_f32(uint32_t h) {
    __half_raw r;
    r.x = (uint16_t)h;
    return __half2float(*reinterpret_cast<__half*>(&r));
}

typedef unsigned int u32x4g __attribute__((ext_vector_type(4)));

// Raw staged bytes for one thread's 32-weight W-tile slice, loaded one
// K-step early so HBM latency hides under the previous tile's MFMA phase
// (T14 split). Quant formats only; float formats load synchronously.
struct WRaw {
    u32x4g q0, q1;   // DQ4K uses q0 only
    uint2 hd;        // DQ4K pre-decoded pair header
    float sc0, sc1;  // DQ6K/DQ8 effective scales
};

template <DT W>
__device__ __forceinline__ void load_w_raw(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    int64_t row, int K, int kb, int h, WRaw* r) {
    if constexpr (W == DT::DQ4K) {
        const int sb = kb >> 8, q = (kb & 255) >> 6, p = 2 * q + h;
        const uint8_t* qrow = qs + row * (K / 256) * 128;
        r->q0 = *(
            reinterpret_cast<const u32x4g*>(qrow) + sb * 8 + p);
        r->hd = reinterpret_cast<const uint2*>(
            hdr + row * (K / 256) * 32)[sb * 4 + p / 2];
    } else if constexpr (W == DT::DQ6K || W == DT::DQ8) {
        const int k0 = kb + h * 32;
        const int8_t* qrow = reinterpret_cast<const int8_t*>(qs + row * K);
        r->q0 = *(
            reinterpret_cast<const u32x4g*>(qrow + k0));
        r->q1 = *(
            reinterpret_cast<const u32x4g*>(qrow + k0) + 1);
        if constexpr (W == DT::DQ6K) {
            const uint8_t* hb = hdr + row * (K / 256) * 32 + (k0 >> 8) * 32;
            const float d = f16b_to_f32(*reinterpret_cast<const uint16_t*>(hb));
            const int s16 = (k0 & 255) >> 4;
            r->sc0 = d * (float)(reinterpret_cast<const int8_t*>(hb)[4 + s16]);
            r->sc1 = d * (float)(reinterpret_cast<const int8_t*>(hb)[4 + s16 + 1]);
        } else {
            const uint16_t* drow = reinterpret_cast<const uint16_t*>(
                hdr + row * (K / 32) * 2);
            r->sc0 = f16b_to_f32(drow[k0 >> 5]);
            r->sc1 = r->sc0;
        }
    } else {
        (void)qs; (void)hdr; (void)row; (void)K; (void)kb; (void)h; (void)r;
    }
}

// Dequantize this thread's 32-weight slice of the W tile into `out` bf16.
// Thread t covers W row (tile_row = t>>1), k-halves h = t&1 within [kb,kb+BK).
template <DT W>
__device__ __forceinline__ void stage_w_slice(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const WRaw& rr, int64_t row, int K, int kb, int h, uint16_t out[32]) {
    if constexpr (W == DT::DQ4K) {
        const int q = (kb & 255) >> 6;
        const int p = 2 * q + h;
        (void)p;
        const uint32_t dw[4] = {rr.q0.x, rr.q0.y, rr.q0.z, rr.q0.w};
        const uint2 hd = rr.hd;
        const float d = f16b_to_f32(hd.x & 0xFFFF);
        const float dmin = f16b_to_f32(hd.x >> 16);
        const float dl = d * (float)(hd.y & 0xFF);
        const float ml = dmin * (float)((hd.y >> 8) & 0xFF);
        const float dh2 = d * (float)((hd.y >> 16) & 0xFF);
        const float mh = dmin * (float)(hd.y >> 24);
        // lo nibbles -> klocal h*16+t (sub-block 2q), hi -> 32+h*16+t
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            const uint32_t lo = dw[j] & 0x0F0F0F0Fu;
            const uint32_t hi = (dw[j] >> 4) & 0x0F0F0F0Fu;
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                out[j * 4 + t] =
                    f32_to_bf16b(dl * (float)((lo >> (8 * t)) & 0xFF) - ml);
                out[16 + j * 4 + t] =
                    f32_to_bf16b(dh2 * (float)((hi >> (8 * t)) & 0xFF) - mh);
            }
        }
    } else if constexpr (W == DT::DQ6K || W == DT::DQ8) {
        const uint32_t dw[8] = {rr.q0.x, rr.q0.y, rr.q0.z, rr.q0.w,
                                rr.q1.x, rr.q1.y, rr.q1.z, rr.q1.w};
        const float sc[2] = {rr.sc0, rr.sc1};
        #pragma unroll
        for (int j = 0; j < 8; j++) {
            const float s = sc[j >> 2];
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                const int8_t v = (int8_t)((dw[j] >> (8 * t)) & 0xFF);
                out[j * 4 + t] = f32_to_bf16b(s * (float)v);
            }
        }
    } else if constexpr (W == DT::BF16) {
        const uint16_t* wrow = reinterpret_cast<const uint16_t*>(qs) + row * K
                               + kb + h * 32;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            const uint4 v = reinterpret_cast<const uint4*>(wrow)[j];
            const uint32_t dw[4] = {v.x, v.y, v.z, v.w};
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                out[j * 8 + 2 * t] = (uint16_t)(dw[t] & 0xFFFF);
                out[j * 8 + 2 * t + 1] = (uint16_t)(dw[t] >> 16);
            }
        }
    } else if constexpr (W == DT::F16) {
        const uint16_t* wrow = reinterpret_cast<const uint16_t*>(qs) + row * K
                               + kb + h * 32;
        #pragma unroll
        for (int t = 0; t < 32; t++)
            out[t] = f32_to_bf16b(f16b_to_f32(wrow[t]));
    } else {  // F32
        const float* wrow = reinterpret_cast<const float*>(qs) + row * K
                            + kb + h * 32;
        #pragma unroll
        for (int t = 0; t < 32; t++) out[t] = f32_to_bf16b(wrow[t]);
    }
    (void)qs; (void)hdr; (void)row; (void)K; (void)kb; (void)h;
}

}  // namespace

template <DT W, int BM_, bool XSILU>
__global__ __launch_bounds__(256) void k_gemm(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ X,     // [M][ldx] f32 (gate half when XSILU)
    const float* __restrict__ X2,    // [M][ldx] up half (XSILU only)
    const float* __restrict__ res,   // [M][ldc] or null (C col-offset applied)
    float* __restrict__ C,           // [M][ldc]
    int M, int N, int K, int ldc, int ldx, int k_chunk) {
    // k_chunk: this block's K-range is [z*k_chunk, min((z+1)*k_chunk, K));
    // splitk > 1 => partial results accumulated with atomicAdd (C pre-zeroed,
    // residual folded in by the z==0 block).
    // Wave grid: BM>=32 uses 2x2 (each wave BM/2 rows x 64 cols); BM=16 uses
    // 1x4 (each wave all 16 rows x 32 cols) so no MFMA row is padding when
    // the decode batch is <=16.
    constexpr int WMW = (BM_ == 16) ? 1 : 2;     // waves tiling M
    constexpr int WNW = 4 / WMW;                 // waves tiling N
    constexpr int JF = BN / WNW / 16;            // b-fragments per wave
    constexpr int FM = (BM_ / WMW) / 16;         // a-fragments per wave
    // Decode tiles (BM<=32) double-buffer: stage tile t+1 into buf[1-p]
    // while the MFMA phase reads buf[p] — one barrier per tile, LDS writes
    // overlap MFMA. The BM=128 prefill tile stays single-buffered: its
    // double LDS footprint (74 KB) would halve occupancy, which costs more
    // than the extra barrier on MFMA-dense prefill tiles.
    constexpr int NBUF = (BM_ <= 32) ? 2 : 1;
    __shared__ __attribute__((aligned(16))) uint16_t Xl[NBUF][BM_ * LDW];
    __shared__ __attribute__((aligned(16))) uint16_t Wl[NBUF][BN * LDW];

    const int tid = threadIdx.x;
    const int bn = blockIdx.x, bm = blockIdx.y, bz = blockIdx.z;
    const bool splitk = gridDim.z > 1;
    const int m0 = bm * BM_, n0 = bn * BN;
    const int kb_lo = bz * k_chunk;
    const int kb_hi = min(kb_lo + k_chunk, K);
    // ceil-rounded split-K chunking can leave trailing z-blocks with an
    // empty K-range; they contribute nothing and must not touch memory
    if (kb_lo >= kb_hi) return;
    const int wid = tid >> 6, lane = tid & 63;
    const int wm = (WMW == 1) ? 0 : (wid >> 1);
    const int wn = (WMW == 1) ? wid : (wid & 1);
    const int lrow = lane & 15, lk = lane >> 4;  // fragment lane coords

    f32x4 acc[FM][JF];
    #pragma unroll
    for (int i = 0; i < FM; i++)
        #pragma unroll
        for (int j = 0; j < JF; j++) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    const int srow0 = tid >> 1, sh0 = tid & 1;   // W staging coords
    const int64_t gn_s = (int64_t)n0 + srow0;
    const int64_t gn_c = gn_s < N ? gn_s : N - 1;  // clamped (branchless)
    WRaw wr, wr_next;
    load_w_raw<W>(qs, hdr, gn_c, K, kb_lo, sh0, &wr);
    // BM=32: the X tile is 8 floats/thread — prefetch it alongside W so the
    // whole staging phase runs from registers (M<=32 decode batches have
    // too few workgroups to hide latency with occupancy alone).
    constexpr bool SMALLM = (BM_ <= 32);
    float4 xr0, xr1, xr0n, xr1n, ur0, ur1, ur0n, ur1n;
    const int xrow = SMALLM ? (tid >> 3) : 0;    // 0..31 (BM=16: >=16 idle)
    const int xseg = SMALLM ? (tid & 7) : 0;
    const int xgm = m0 + xrow < M ? m0 + xrow : (M > 0 ? M - 1 : 0);
    const bool xvalid0 = (m0 + xrow) < M && xrow < BM_;
    if (SMALLM) {
        const float4* src = reinterpret_cast<const float4*>(
            X + (size_t)xgm * ldx + kb_lo + xseg * 8);
        xr0 = src[0];
        xr1 = src[1];
        if constexpr (XSILU) {
            const float4* up = reinterpret_cast<const float4*>(
                X2 + (size_t)xgm * ldx + kb_lo + xseg * 8);
            ur0 = up[0];
            ur1 = up[1];
        }
    }

    auto stage_tile = [&](int kb_s, int pb) {
        // ---- stage X tile (f32 -> bf16), 8-elem units ----
        if (SMALLM) {
            float4 v[2] = {xr0, xr1};
            if constexpr (XSILU) {
                const float4 u[2] = {ur0, ur1};
                #pragma unroll
                for (int j = 0; j < 2; j++) {
                    v[j].x = (v[j].x / (1.f + __expf(-v[j].x))) * u[j].x;
                    v[j].y = (v[j].y / (1.f + __expf(-v[j].y))) * u[j].y;
                    v[j].z = (v[j].z / (1.f + __expf(-v[j].z))) * u[j].z;
                    v[j].w = (v[j].w / (1.f + __expf(-v[j].w))) * u[j].w;
                }
            }
            uint16_t tmp8[8];
            #pragma unroll
            for (int j = 0; j < 2; j++) {
                tmp8[j * 4 + 0] = f32_to_bf16b(v[j].x);
                tmp8[j * 4 + 1] = f32_to_bf16b(v[j].y);
                tmp8[j * 4 + 2] = f32_to_bf16b(v[j].z);
                tmp8[j * 4 + 3] = f32_to_bf16b(v[j].w);
            }
            if (!xvalid0) {
                #pragma unroll
                for (int j = 0; j < 8; j++) tmp8[j] = 0;
            }
            if (xrow < BM_)
                *reinterpret_cast<uint4*>(Xl[pb] + xrow * LDW + xseg * 8) =
                    *reinterpret_cast<const uint4*>(tmp8);
        } else {
            constexpr int UNITS = BM_ * BK / 8;   // 8 bf16 per unit
            #pragma unroll
            for (int ui = 0; ui < (UNITS + 255) / 256; ui++) {
                const int unit = tid + ui * 256;
                if (UNITS < 256 && unit >= UNITS) break;
                const int row = unit >> 3, seg = unit & 7;
                const int gm = m0 + row;
                uint16_t tmp[8];
                if (gm < M) {
                    const float4* src = reinterpret_cast<const float4*>(
                        X + (size_t)gm * ldx + kb_s + seg * 8);
                    #pragma unroll
                    for (int j = 0; j < 2; j++) {
                        float4 v = src[j];
                        if constexpr (XSILU) {
                            const float4 u = reinterpret_cast<const float4*>(
                                X2 + (size_t)gm * ldx + kb_s + seg * 8)[j];
                            v.x = (v.x / (1.f + __expf(-v.x))) * u.x;
                            v.y = (v.y / (1.f + __expf(-v.y))) * u.y;
                            v.z = (v.z / (1.f + __expf(-v.z))) * u.z;
                            v.w = (v.w / (1.f + __expf(-v.w))) * u.w;
                        }
                        tmp[j * 4 + 0] = f32_to_bf16b(v.x);
                        tmp[j * 4 + 1] = f32_to_bf16b(v.y);
                        tmp[j * 4 + 2] = f32_to_bf16b(v.z);
                        tmp[j * 4 + 3] = f32_to_bf16b(v.w);
                    }
                } else {
                    #pragma unroll
                    for (int j = 0; j < 8; j++) tmp[j] = 0;
                }
                *reinterpret_cast<uint4*>(Xl[pb] + row * LDW + seg * 8) =
                    *reinterpret_cast<const uint4*>(tmp);
            }
        }
        // ---- stage + dequant W tile (quant raw bytes preloaded) ----
        {
            const int srow = srow0, sh = sh0;
            uint16_t tmp[32];
            if (gn_s < N) {
                stage_w_slice<W>(qs, hdr, wr, gn_c, K, kb_s, sh, tmp);
            } else {
                #pragma unroll
                for (int j = 0; j < 32; j++) tmp[j] = 0;
            }
            if constexpr (W == DT::DQ4K) {
                uint4* d0 = reinterpret_cast<uint4*>(
                    Wl[pb] + srow * LDW + sh * 16);
                uint4* d1 = reinterpret_cast<uint4*>(
                    Wl[pb] + srow * LDW + 32 + sh * 16);
                d0[0] = reinterpret_cast<const uint4*>(tmp)[0];
                d0[1] = reinterpret_cast<const uint4*>(tmp)[1];
                d1[0] = reinterpret_cast<const uint4*>(tmp)[2];
                d1[1] = reinterpret_cast<const uint4*>(tmp)[3];
            } else {
                uint4* dst = reinterpret_cast<uint4*>(
                    Wl[pb] + srow * LDW + sh * 32);
                #pragma unroll
                for (int j = 0; j < 4; j++)
                    dst[j] = reinterpret_cast<const uint4*>(tmp)[j];
            }
        }
    };

    auto load_next = [&](int kb_n) {
        load_w_raw<W>(qs, hdr, gn_c, K, kb_n, sh0, &wr_next);
        if (SMALLM) {
            const float4* src = reinterpret_cast<const float4*>(
                X + (size_t)xgm * ldx + kb_n + xseg * 8);
            xr0n = src[0];
            xr1n = src[1];
            if constexpr (XSILU) {
                const float4* up = reinterpret_cast<const float4*>(
                    X2 + (size_t)xgm * ldx + kb_n + xseg * 8);
                ur0n = up[0];
                ur1n = up[1];
            }
        }
    };

    auto mfma_tile = [&](int pb) {
        #pragma unroll
        for (int ks = 0; ks < BK; ks += 32) {
            bf16x8 a[FM], b[JF];
            #pragma unroll
            for (int i = 0; i < FM; i++) {
                const int xr = wm * (BM_ / WMW) + i * 16 + lrow;
                a[i] = *reinterpret_cast<const bf16x8*>(
                    Xl[pb] + xr * LDW + ks + lk * 8);
            }
            #pragma unroll
            for (int j = 0; j < JF; j++) {
                const int wrr = wn * (BN / WNW) + j * 16 + lrow;
                b[j] = *reinterpret_cast<const bf16x8*>(
                    Wl[pb] + wrr * LDW + ks + lk * 8);
            }
            #pragma unroll
            for (int i = 0; i < FM; i++)
                #pragma unroll
                for (int j = 0; j < JF; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a[i], b[j], acc[i][j], 0, 0, 0);
        }
    };

    if constexpr (NBUF == 1) {
        // single-buffer: stage -> barrier -> (prefetch t+1 raw) -> MFMA ->
        // barrier, raw W/X registers one tile ahead (T14)
        for (int kb = kb_lo; kb < kb_hi; kb += BK) {
            stage_tile(kb, 0);
            __syncthreads();
            const int kbn = (kb + BK < kb_hi) ? kb + BK : kb;
            load_next(kbn);
            mfma_tile(0);
            __syncthreads();
            wr = wr_next;
            if (SMALLM) {
                xr0 = xr0n; xr1 = xr1n;
                if constexpr (XSILU) { ur0 = ur0n; ur1 = ur1n; }
            }
        }
    } else {
        // prologue: tile 0 into buf 0, then prefetch tile 1's raw bytes
        stage_tile(kb_lo, 0);
        __syncthreads();
        {
            const int kb1 = (kb_lo + BK < kb_hi) ? kb_lo + BK : kb_lo;
            load_next(kb1);
            wr = wr_next;
            if (SMALLM) {
                xr0 = xr0n; xr1 = xr1n;
                if constexpr (XSILU) { ur0 = ur0n; ur1 = ur1n; }
            }
        }
        int pb = 0;
        for (int kb = kb_lo; kb < kb_hi; kb += BK) {
            const bool has_next = (kb + BK) < kb_hi;
            if (has_next) {
                // raw loads for tile t+2 issue first: their HBM latency
                // hides under this iteration's ds_writes + MFMA (T14)
                const int kb2 = (kb + 2 * BK < kb_hi) ? kb + 2 * BK : kb;
                load_next(kb2);
                stage_tile(kb + BK, 1 - pb);   // overlaps MFMA (other buf)
            }
            mfma_tile(pb);
            __syncthreads();
            wr = wr_next;
            if (SMALLM) {
                xr0 = xr0n; xr1 = xr1n;
                if constexpr (XSILU) { ur0 = ur0n; ur1 = ur1n; }
            }
            pb ^= 1;
        }
    }

    // ---- epilogue ----
    #pragma unroll
    for (int i = 0; i < FM; i++) {
        #pragma unroll
        for (int r = 0; r < 4; r++) {
            const int m = m0 + wm * (BM_ / WMW) + i * 16 + lk * 4 + r;
            if (m >= M) continue;
            #pragma unroll
            for (int j = 0; j < JF; j++) {
                const int n = n0 + wn * (BN / WNW) + j * 16 + lrow;
                if (n >= N) continue;
                const size_t idx = (size_t)m * ldc + n;
                const float rv = (res && (!splitk || bz == 0)) ? res[idx] : 0.f;
                if (splitk) {
#ifdef CLA_PERF_PROBE_NOATOMIC
                    C[idx] = acc[i][j][r] + rv;  // WRONG, perf probe only
#else
                    atomicAdd(&C[idx], acc[i][j][r] + rv);
#endif
                } else {
                    C[idx] = acc[i][j][r] + rv;
                }
            }
        }
    }
}

// ---- row-wise elementwise kernels for the GEMM (prefill/batched) path ----

// grid M; block 256. out[m] = rmsnorm(x[m]) * gw; optionally emits the
// fused i8-GEMM activation quantization (xq non-null — common.h helper).
__global__ __launch_bounds__(256) void k_rmsnorm_rows(
    const float* __restrict__ X, const float* __restrict__ gw,
    float* __restrict__ out, int K, float eps,
    int8_t* __restrict__ xq, float* __restrict__ xsc, int M4) {
    const int m = blockIdx.x;
    const float4* x4 = reinterpret_cast<const float4*>(X + (size_t)m * K);
    float4* o4 = reinterpret_cast<float4*>(out + (size_t)m * K);
    const float4* g4 = reinterpret_cast<const float4*>(gw);
    const int K4 = K >> 2;
    float ss = 0.f;
    for (int k = threadIdx.x; k < K4; k += 256) {
        const float4 v = x4[k];
        ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
    }
    __shared__ float red[256];
    red[threadIdx.x] = ss;
    __syncthreads();
    #pragma unroll
    for (int off = 128; off > 0; off >>= 1) {
        if ((int)threadIdx.x < off) red[threadIdx.x] += red[threadIdx.x + off];
        __syncthreads();
    }
    const float inv = rsqrtf(red[0] / (float)K + eps);
    if (xq) {
        const int jl = threadIdx.x & 7;
        for (int b32 = threadIdx.x >> 3; b32 < K / 32; b32 += 32) {
            const int k = b32 * 8 + jl;
            const float4 v = x4[k];
            const float4 g = g4[k];
            float4 o;
            o.x = v.x * inv * g.x; o.y = v.y * inv * g.y;
            o.z = v.z * inv * g.z; o.w = v.w * inv * g.w;
            o4[k] = o;
            quant_block_emit(o, jl, b32, m, K, M4, xq, xsc);
        }
    } else {
        for (int k = threadIdx.x; k < K4; k += 256) {
            const float4 v = x4[k];
            const float4 g = g4[k];
            float4 o;
            o.x = v.x * inv * g.x; o.y = v.y * inv * g.y;
            o.z = v.z * inv * g.z; o.w = v.w * inv * g.w;
            o4[k] = o;
        }
    }
}


// Prefill RoPE + KV append over M prompt rows of ONE slot.
// grid (M, KVH); block 128. Position of row m is pos0 + m.
__global__ __launch_bounds__(128) void k_rope_prefill(
    float* __restrict__ qkv,          // [M][(NH+2*NKV)*D]
    const float* __restrict__ inv_freq,
    const int32_t* __restrict__ page_table, uint16_t* __restrict__ kv_pool,
    int slot, int pos0, int NH, int NKV, int D, int G, int page_size,
    int max_pages, int64_t page_stride) {
    const int m = blockIdx.x, kvh = blockIdx.y;
    const int pos = pos0 + m;
    float* row = qkv + (size_t)m * (NH + 2 * NKV) * D;
    float* kh = row + (size_t)(NH + kvh) * D;
    const float* vh = row + (size_t)(NH + NKV + kvh) * D;
    const int half = D / 2;
    for (int idx = threadIdx.x; idx < (G + 1) * half; idx += 128) {
        const int hsel = idx / half, i = idx % half;
        float* p = (hsel < G) ? (row + (size_t)(kvh * G + hsel) * D) : kh;
        float sn, cs;
        __sincosf((float)pos * inv_freq[i], &sn, &cs);
        const float x0 = p[2 * i], x1 = p[2 * i + 1];
        p[2 * i] = x0 * cs - x1 * sn;
        p[2 * i + 1] = x0 * sn + x1 * cs;
    }
    __syncthreads();
    const int page = page_table[(size_t)slot * max_pages + pos / page_size];
    uint16_t* kdst = kv_pool + (int64_t)page * page_stride
                     + ((int64_t)kvh * 2 + 0) * page_size * D
                     + (int64_t)(pos % page_size) * D;
    uint16_t* vdst = kv_pool + (int64_t)page * page_stride
                     + ((int64_t)kvh * 2 + 1) * page_size * D
                     + (int64_t)(pos % page_size) * D;
    for (int d = threadIdx.x; d < D; d += 128) {
        union { uint32_t u; float f; } a, b;
        a.f = kh[d];
        b.f = vh[d];
        kdst[d] = (uint16_t)((a.u + 0x7FFF + ((a.u >> 16) & 1)) >> 16);
        vdst[d] = (uint16_t)((b.u + 0x7FFF + ((b.u >> 16) & 1)) >> 16);
    }
}

// Causal prefill attention for ONE slot reading the paged cache.
// grid (ceil(M/16), NH); block 256 = 4 waves x 4 quarters = 16 q-rows.
// Each 16-lane quarter owns one q row's full (m,l,o[D]) accumulator and
// sweeps cached positions 0..pos0+row (causal).
template <int D>
__global__ __launch_bounds__(256) void k_attn_prefill(
    const float* __restrict__ qkv,    // [M][(NH+2*NKV)*D], q already roped
    const int32_t* __restrict__ page_table,
    const uint16_t* __restrict__ kv_pool,
    float* __restrict__ attn_out,     // [M][NH*D]
    int slot, int pos0, int M, int NH, int NKV, int page_size, int max_pages,
    int64_t page_stride, float scale) {
    constexpr int DPL = D / 16;
    const int qt = blockIdx.x, head = blockIdx.y;
    const int kvh = head / (NH / NKV);
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int quarter = (tid >> 6) * 4 + (lane >> 4);   // 0..15
    const int qlane = lane & 15;
    const int d0 = qlane * DPL;
    const int m = qt * 16 + quarter;                    // q row in [0,M)
    const bool active = m < M;
    const int causal_len = active ? (pos0 + m + 1) : 0;

    float qf[DPL];
    #pragma unroll
    for (int j = 0; j < DPL; j++) qf[j] = 0.f;
    if (active) {
        const float* qh = qkv + (size_t)m * (NH + 2 * NKV) * D
                          + (size_t)head * D + d0;
        #pragma unroll
        for (int j = 0; j < DPL; j++) qf[j] = qh[j] * scale;
    }

    float mx = -1e30f, l = 0.f, o[DPL];
    #pragma unroll
    for (int j = 0; j < DPL; j++) o[j] = 0.f;

    for (int p = 0; p < causal_len; p++) {
        const int page = page_table[(size_t)slot * max_pages + p / page_size];
        const uint16_t* kp = kv_pool + (int64_t)page * page_stride
                             + ((int64_t)kvh * 2 + 0) * page_size * D
                             + (int64_t)(p % page_size) * D + d0;
        const uint16_t* vp = kv_pool + (int64_t)page * page_stride
                             + ((int64_t)kvh * 2 + 1) * page_size * D
                             + (int64_t)(p % page_size) * D + d0;
        float dot = 0.f;
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            const uint32_t kw = reinterpret_cast<const uint32_t*>(kp)[j];
            union { uint32_t u; float f; } lo, hi;
            lo.u = (kw & 0xFFFF) << 16;
            hi.u = (kw >> 16) << 16;
            dot += qf[2 * j] * lo.f + qf[2 * j + 1] * hi.f;
        }
        #pragma unroll
        for (int off = 1; off < 16; off <<= 1) dot += __shfl_xor(dot, off, 64);
        const float mn = fmaxf(mx, dot);
        const float alpha = __expf(mx - mn);
        const float w = __expf(dot - mn);
        l = l * alpha + w;
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            const uint32_t vw = reinterpret_cast<const uint32_t*>(vp)[j];
            union { uint32_t u; float f; } lo, hi;
            lo.u = (vw & 0xFFFF) << 16;
            hi.u = (vw >> 16) << 16;
            o[2 * j] = o[2 * j] * alpha + w * lo.f;
            o[2 * j + 1] = o[2 * j + 1] * alpha + w * hi.f;
        }
        mx = mn;
    }
    if (active) {
        float* dst = attn_out + (size_t)m * NH * D + (size_t)head * D + d0;
        #pragma unroll
        for (int j = 0; j < DPL; j++) dst[j] = o[j] / l;
    }
}

// --------------------------------------------------------- launch stubs

#define DISPATCH_DT_GEMM(DTV, FN)                                     \
    switch (DTV) {                                                    \
        case DT::DQ4K: FN(DT::DQ4K); break;                           \
        case DT::DQ6K: FN(DT::DQ6K); break;                           \
        case DT::DQ8:  FN(DT::DQ8);  break;                           \
        case DT::BF16: FN(DT::BF16); break;                           \
        case DT::F16:  FN(DT::F16);  break;                           \
        case DT::F32:  FN(DT::F32);  break;                           \
        default: throw std::runtime_error("bad dtype");               \
    }

// split-K workgroup target: >=2 WGs per CU keeps all 8 XCDs fed when M is
// small; overridable for on-hardware sweeps (CLA_SPLITK_TARGET).
// BM=16 decode tiles (M<=16) like deeper splits (1024 once the k-loop
// went barrier-free: extra WGs no longer cost arrival skew — B=8 2822
// vs 2778 tok/s, B=16 tied; 1536 regresses); BM=32 tiles regressed deep
// (B=32: 5549 vs 6260 tok/s) and keep 512.
static int splitk_target(int M) {
    static int env = [] {
        const char* e = getenv("CLA_SPLITK_TARGET");
        return e ? atoi(e) : 0;
    }();
    if (env > 0) return env;
    return M <= 16 ? 1024 : 512;
}

// Single source of truth for the split-K factor: launch_gemm_ex and
// gemm_uses_splitk (which tells the caller whether C must be pre-zeroed
// for the atomicAdd accumulation) MUST agree exactly — a mismatch for any
// M range means accumulating into stale scratch (round-1 advisor finding:
// M=33..128 prompts silently corrupted activations).
int gemm_splitk_factor(int N, int K, int M) {
    // split-K applies to the decode/short-prefill range; beyond it there
    // are enough (bm x n) tiles to fill the chip without it
    const bool small_m = M <= 128;
    if (!small_m) return 1;
    // BM tiling matches the launchers: 16/32 decode tiles
    const int bm_tiles = (M + 31) / 32;
    const int n_tiles = (N + BN - 1) / BN;
    const int tgt = splitk_target(M);
    const int wgs = n_tiles * bm_tiles;
    int splitk = K / BK < tgt / (wgs ? wgs : 1) ? K / BK : tgt / (wgs ? wgs : 1);
    if (splitk < 1) splitk = 1;
    // re-derive so ceil-rounding leaves no empty z-blocks
    const int steps = K / BK;
    const int chunks = (steps + splitk - 1) / splitk;
    splitk = (steps + chunks - 1) / chunks;
    return splitk;
}

bool gemm_uses_splitk(int N, int K, int M) {
    return gemm_splitk_factor(N, K, M) > 1;
}

void launch_gemm_ex(const WTensor& w, const float* X, const float* X2,
                    int ldx, bool xsilu, const float* res, float* C, int M,
                    int ldc, hipStream_t stream) {
    const int N = (int)w.n, K = (int)w.k;
    if (K % BK != 0) throw std::runtime_error("gemm: K must be /64");
    // decode-batch sizing: BM=32 tiles (BM=16 for B<=16) with split-K up
    // to M=128 — a single 128-row tile grid is only n_tiles workgroups and
    // leaves the chip mostly idle (B=64 measured 5x slower without this).
    const bool small_m = M <= 128;
    const int bm_tiles = small_m ? (M + 31) / 32 : (M + BM - 1) / BM;
    const int n_tiles = (N + BN - 1) / BN;
    const int splitk = gemm_splitk_factor(N, K, M);
    const int k_chunk = ((K / BK + splitk - 1) / splitk) * BK;
    dim3 grid(n_tiles, bm_tiles, splitk), block(256);
    // splitk > 1 accumulates with atomicAdd: caller must pre-zero C
    // (gemm_uses_splitk tells it whether that is needed).
    #define GEMM_ONE(WT, BMV, XSV)                                             \
        hipLaunchKernelGGL((k_gemm<WT, BMV, XSV>), grid, block, 0, stream,     \
            (const uint8_t*)w.qs, (const uint8_t*)w.hdr, X, X2, res, C,        \
            M, N, K, ldc, ldx, k_chunk)
    #define GEMM_CASE(WT)                                                      \
        do {                                                                   \
            if (M <= 16 && xsilu) GEMM_ONE(WT, 16, true);                      \
            else if (M <= 16) GEMM_ONE(WT, 16, false);                         \
            else if (small_m && xsilu) GEMM_ONE(WT, 32, true);                 \
            else if (small_m) GEMM_ONE(WT, 32, false);                         \
            else if (xsilu) GEMM_ONE(WT, 128, true);                           \
            else GEMM_ONE(WT, 128, false);                                     \
        } while (0)
    DISPATCH_DT_GEMM(w.dtype, GEMM_CASE);
    #undef GEMM_CASE
    #undef GEMM_ONE
}

void launch_gemm(const WTensor& w, const float* X, const float* res, float* C,
                 int M, int ldc, hipStream_t stream) {
    launch_gemm_ex(w, X, nullptr, (int)w.k, false, res, C, M, ldc, stream);
}

void launch_rmsnorm_rows(const float* X, const float* gw, float* out, int M,
                         int K, float eps, hipStream_t stream) {
    hipLaunchKernelGGL(k_rmsnorm_rows, dim3(M), dim3(256), 0, stream,
                       X, gw, out, K, eps, (int8_t*)nullptr,
                       (float*)nullptr, 0);
}

// fused-quant variant: also emits xq/xsc for the i8 GEMM consumer
void launch_rmsnorm_rows_q(const float* X, const float* gw, float* out,
                           int M, int K, float eps, int8_t* xq, float* xsc,
                           hipStream_t stream) {
    const int M4 = (M + 3) & ~3;
    hipLaunchKernelGGL(k_rmsnorm_rows, dim3(M), dim3(256), 0, stream,
                       X, gw, out, K, eps, xq, xsc, M4);
}


void launch_rope_prefill(float* qkv, const float* inv_freq,
                         const int32_t* page_table, uint16_t* kv_pool,
                         int slot, int pos0, int M, int NH, int NKV, int D,
                         int page_size, int max_pages, int64_t page_stride,
                         hipStream_t stream) {
    hipLaunchKernelGGL(k_rope_prefill, dim3(M, NKV), dim3(128), 0, stream,
                       qkv, inv_freq, page_table, kv_pool, slot, pos0,
                       NH, NKV, D, NH / NKV, page_size, max_pages,
                       page_stride);
}

void launch_attn_prefill(const float* qkv, const int32_t* page_table,
                         const uint16_t* kv_pool, float* attn_out, int slot,
                         int pos0, int M, int NH, int NKV, int D,
                         int page_size, int max_pages, int64_t page_stride,
                         float scale, hipStream_t stream) {
    dim3 grid((M + 15) / 16, NH), block(256);
    if (D == 128) {
        hipLaunchKernelGGL(k_attn_prefill<128>, grid, block, 0, stream,
                           qkv, page_table, kv_pool, attn_out, slot, pos0, M,
                           NH, NKV, page_size, max_pages, page_stride, scale);
    } else if (D == 64) {
        hipLaunchKernelGGL(k_attn_prefill<64>, grid, block, 0, stream,
                           qkv, page_table, kv_pool, attn_out, slot, pos0, M,
                           NH, NKV, page_size, max_pages, page_stride, scale);
    } else {
        throw std::runtime_error("head_dim must be 64 or 128");
    }
}

}  // namespace cla
