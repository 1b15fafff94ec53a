// int8-activation MFMA dequant-GEMM for gfx950: the batched-decode hot path.
//
// C[M,N] = Xq[M,K] @ W[N,K]^T (+res) where W is Q4_K or Q8_0 and X was
// pre-quantized to int8 per 32-block (k_quant_rows below; same semantics as
// the act_q8 GEMV / ref_numpy(act_q8=True): x ~ rint(x/xd)*xd).
//
// Why int8 MFMA instead of the bf16-staging GEMM (gemm.hip) for decode
// batches: round-1 PMC showed the bf16 tile 34.8% active-issue / 52.9% wait
// with the weight stream at ~1.1 TB/s — the per-weight float dequant into
// LDS bf16 (~5 VALU/weight) dominated. Here weights stage as RAW int4/int8
// values (nibble unpack only, ~0.25 VALU/weight), LDS traffic halves
// (1 B/weight), and v_mfma_i32_16x16x32_i8 does the dot exactly; per-32
// block scales are applied at a per-K=32 drain:
//   y += d_w*d_x*dot_i32 - m_w*(d_x*sum qx)     (Q4_K; Q8_0 has m_w=0)
// The i32 dot is exact (|q|<=15, |qx|<=127, 32 terms), so numerics match
// the act_q8 GEMV path bit-for-ish (f32 accumulation order aside).
//
// Replaces (functionally) llama.cpp's MMQ path for batched decode — the
// compute the reference delegates to Ollama; designed for
// CDNA4 wave64/XCD geometry, not ported.
#include "common.h"

namespace cla {

typedef int v4i __attribute__((ext_vector_type(4)));
typedef unsigned int u32x4v __attribute__((ext_vector_type(4)));

int gemm_splitk_factor(int N, int K, int M);  // gemm.hip (shared contract)

namespace {

constexpr int BN = 128, BK = 64;
constexpr int LDQ = BK + 16;    // int8 row stride: conflict-free b64 reads

__device__ __forceinline__ float f16b2f(uint32_t h) {
    __half_raw r;
    r.x = (uint16_t)h;
    return __half2float(*reinterpret_cast<__half*>(&r));
}

// Raw bytes one thread stages per tile (loaded a tile ahead, T14 split).
template <DT W>
struct WRawI8 {
    u32x4v q0, q1;  // Q4K uses q0 only (16 B = 32 nibbles)
    uint2 hd;       // Q4K pair header
    float d0;       // Q8: block scale
};

template <DT W>
__device__ __forceinline__ void load_w_raw_i8(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    int64_t row, int K, int kb, int h, WRawI8<W>* r) {
    if constexpr (W == DT::DQ4K) {
        const int sb = kb >> 8, q = (kb & 255) >> 6, p = 2 * q + h;
        const uint8_t* qrow = qs + row * (K / 256) * 128;
        r->q0 = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4v*>(qrow) + sb * 8 + p);
        r->hd = reinterpret_cast<const uint2*>(
            hdr + row * (K / 256) * 32)[sb * 4 + q];
    } else {  // DQ8
        const int k0 = kb + h * 32;
        const uint8_t* qrow = qs + row * K;
        r->q0 = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4v*>(qrow + k0));
        r->q1 = __builtin_nontemporal_load(
            reinterpret_cast<const u32x4v*>(qrow + k0) + 1);
        const uint16_t* drow = reinterpret_cast<const uint16_t*>(
            hdr + row * (K / 32) * 2);
        r->d0 = f16b2f(drow[k0 >> 5]);  // both 16B chunks share this block
    }
}

}  // namespace

// BM_ in {16, 32}. 256 threads = 4 waves; BM=16 -> 1x4 wave grid (each wave
// all 16 M-rows x 32 cols), BM=32 -> 2x2 (16 rows x 64 cols per wave).
// Double-buffered LDS; split-K accumulates into pre-zeroed C via atomicAdd
// (identical contract to gemm.hip's k_gemm: gemm_uses_splitk tells the
// caller to pre-zero).
template <DT W, int BM_>
__global__ __launch_bounds__(256) void k_gemm_i8(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const int8_t* __restrict__ xq,    // [M][ldxq] int8 (pre-quantized)
    const float* __restrict__ xs,     // [M][ldxq/32] block scales
    const float* __restrict__ xsum,   // [M][ldxq/32] dx*sum(qx) per block
    const float* __restrict__ res,    // [M][ldc] or null
    float* __restrict__ C,            // [M][ldc]
    int M, int N, int K, int ldc, int ldxq, int k_chunk) {
    constexpr int WMW = (BM_ == 16) ? 1 : 2;
    constexpr int WNW = 4 / WMW;
    constexpr int JF = BN / WNW / 16;       // b fragments per wave (2 or 4)
    constexpr int NBUF = 2;
    __shared__ __attribute__((aligned(16))) int8_t Wq[NBUF][BN * LDQ];
    __shared__ __attribute__((aligned(16))) int8_t Xq[NBUF][BM_ * LDQ];
    __shared__ __attribute__((aligned(16))) float2 Wsc[NBUF][2][BN];
    __shared__ __attribute__((aligned(16))) float Xdx[NBUF][2][BM_];
    __shared__ __attribute__((aligned(16))) float Xsm[NBUF][2][BM_];

    const int tid = threadIdx.x;
    const int bn = blockIdx.x, bm = blockIdx.y, bz = blockIdx.z;
    const bool splitk = gridDim.z > 1;
    const int m0 = bm * BM_, n0 = bn * BN;
    const int kb_lo = bz * k_chunk;
    const int kb_hi = min(kb_lo + k_chunk, K);
    if (kb_lo >= kb_hi) return;             // empty trailing z-block
    const int wid = tid >> 6, lane = tid & 63;
    const int wm = (WMW == 1) ? 0 : (wid >> 1);
    const int wn = (WMW == 1) ? wid : (wid & 1);
    const int lrow = lane & 15, lk = lane >> 4;

    float facc[JF][4];
    #pragma unroll
    for (int j = 0; j < JF; j++)
        #pragma unroll
        for (int r = 0; r < 4; r++) facc[j][r] = 0.f;

    // ---- staging coordinates ----
    const int srow = tid >> 1, sh = tid & 1;       // W: row, half
    const int64_t gn_s = (int64_t)n0 + srow;
    const int64_t gn_c = gn_s < N ? gn_s : N - 1;  // clamped address
    // X: BM_*4 threads copy one uint4 each; 2*BM_ threads copy scales
    const int xrow = tid >> 2, xseg = tid & 3;
    const int xgm = (m0 + xrow < M) ? m0 + xrow : (M > 0 ? M - 1 : 0);
    const bool xvalid = (m0 + xrow) < M && xrow < BM_;
    const int scrow = tid >> 1, sckb = tid & 1;
    const bool scvalid = scrow < BM_ && (m0 + scrow) < M;
    const int scgm = (m0 + scrow < M) ? m0 + scrow : (M > 0 ? M - 1 : 0);

    WRawI8<W> wr, wr_next;
    uint4 xr_raw, xr_next;
    float2 xsc_raw, xsc_next;
    const int nb32 = ldxq / 32;

    auto load_raw = [&](int kb) {
        load_w_raw_i8<W>(qs, hdr, gn_c, K, kb, sh, &wr_next);
        if (xrow < BM_)
            xr_next = *reinterpret_cast<const uint4*>(
                xq + (size_t)xgm * ldxq + kb + xseg * 16);
        if (scrow < BM_) {
            xsc_next.x = xs[(size_t)scgm * nb32 + (kb >> 5) + sckb];
            xsc_next.y = xsum[(size_t)scgm * nb32 + (kb >> 5) + sckb];
        }
    };

    auto stage_tile = [&](int pb) {
        // ---- W tile: unpack nibbles (Q4K) / copy (Q8) + scales ----
        if constexpr (W == DT::DQ4K) {
            const uint32_t dw[4] = {wr.q0.x, wr.q0.y, wr.q0.z, wr.q0.w};
            uint32_t lo[4], hi[4];
            #pragma unroll
            for (int j = 0; j < 4; j++) {
                lo[j] = dw[j] & 0x0F0F0F0Fu;
                hi[j] = (dw[j] >> 4) & 0x0F0F0F0Fu;
            }
            int8_t* wrow = Wq[pb] + srow * LDQ;
            *reinterpret_cast<uint4*>(wrow + sh * 16) =
                *reinterpret_cast<const uint4*>(lo);
            *reinterpret_cast<uint4*>(wrow + 32 + sh * 16) =
                *reinterpret_cast<const uint4*>(hi);
            if (sh == 0) {
                const float d = f16b2f(wr.hd.x & 0xFFFF);
                const float dmin = f16b2f(wr.hd.x >> 16);
                Wsc[pb][0][srow] = {d * (float)(wr.hd.y & 0xFF),
                                    dmin * (float)((wr.hd.y >> 8) & 0xFF)};
                Wsc[pb][1][srow] = {d * (float)((wr.hd.y >> 16) & 0xFF),
                                    dmin * (float)(wr.hd.y >> 24)};
            }
        } else {
            int8_t* wrow = Wq[pb] + srow * LDQ;
            *reinterpret_cast<u32x4v*>(wrow + sh * 32) = wr.q0;
            *reinterpret_cast<u32x4v*>(wrow + sh * 32 + 16) = wr.q1;
            // h selects one 32-weight half = exactly one 32-block (k0 =
            // kb+h*32); d0 is that block's scale, the min term is 0
            Wsc[pb][sh][srow] = {wr.d0, 0.f};
        }
        // ---- X tile ----
        if (xrow < BM_) {
            uint4 v = xr_raw;
            if (!xvalid) v = {0, 0, 0, 0};
            *reinterpret_cast<uint4*>(Xq[pb] + xrow * LDQ + xseg * 16) = v;
        }
        if (scrow < BM_) {
            Xdx[pb][sckb][scrow] = scvalid ? xsc_raw.x : 0.f;
            Xsm[pb][sckb][scrow] = scvalid ? xsc_raw.y : 0.f;
        }
    };

    auto mfma_tile = [&](int pb) {
        #pragma unroll
        for (int kb2 = 0; kb2 < 2; kb2++) {
            const long a = *reinterpret_cast<const long*>(
                Xq[pb] + (wm * 16 + lrow) * LDQ + kb2 * 32 + lk * 8);
            const float4 dx4 = *reinterpret_cast<const float4*>(
                &Xdx[pb][kb2][wm * 16 + lk * 4]);
            const float4 sm4 = *reinterpret_cast<const float4*>(
                &Xsm[pb][kb2][wm * 16 + lk * 4]);
            #pragma unroll
            for (int j = 0; j < JF; j++) {
                const int col = wn * (BN / WNW) + j * 16 + lrow;
                const long b = *reinterpret_cast<const long*>(
                    Wq[pb] + col * LDQ + kb2 * 32 + lk * 8);
                const float2 dm = Wsc[pb][kb2][col];
                v4i c = {0, 0, 0, 0};
                c = __builtin_amdgcn_mfma_i32_16x16x32_i8(a, b, c, 0, 0, 0);
                facc[j][0] += dm.x * dx4.x * (float)c[0] - dm.y * sm4.x;
                facc[j][1] += dm.x * dx4.y * (float)c[1] - dm.y * sm4.y;
                facc[j][2] += dm.x * dx4.z * (float)c[2] - dm.y * sm4.z;
                facc[j][3] += dm.x * dx4.w * (float)c[3] - dm.y * sm4.w;
            }
        }
    };

    // prologue: raw tile 0, stage into buf 0, prefetch tile 1
    load_raw(kb_lo);
    wr = wr_next; xr_raw = xr_next; xsc_raw = xsc_next;
    stage_tile(0);
    __syncthreads();
    {
        const int kb1 = (kb_lo + BK < kb_hi) ? kb_lo + BK : kb_lo;
        load_raw(kb1);
        wr = wr_next; xr_raw = xr_next; xsc_raw = xsc_next;
    }
    int pb = 0;
    for (int kb = kb_lo; kb < kb_hi; kb += BK) {
        const bool has_next = (kb + BK) < kb_hi;
        if (has_next) {
            const int kb2 = (kb + 2 * BK < kb_hi) ? kb + 2 * BK : kb;
            load_raw(kb2);               // tile t+2 raw: hides under MFMA
            stage_tile(1 - pb);          // tile t+1 into the other buffer
        }
        mfma_tile(pb);
        __syncthreads();
        wr = wr_next; xr_raw = xr_next; xsc_raw = xsc_next;
        pb ^= 1;
    }

    // ---- epilogue (same contract as gemm.hip k_gemm) ----
    #pragma unroll
    for (int r = 0; r < 4; r++) {
        const int m = m0 + wm * 16 + lk * 4 + r;
        if (m >= M) continue;
        #pragma unroll
        for (int j = 0; j < JF; j++) {
            const int n = n0 + wn * (BN / WNW) + j * 16 + lrow;
            if (n >= N) continue;
            const size_t idx = (size_t)m * ldc + n;
            const float rv = (res && (!splitk || bz == 0)) ? res[idx] : 0.f;
            if (splitk) {
                atomicAdd(&C[idx], facc[j][r] + rv);
            } else {
                C[idx] = facc[j][r] + rv;
            }
        }
    }
}

// ---------------------------------------------------- activation quantizer
// Per-row, per-32-block symmetric int8 (xd = amax/127, rint) — identical
// semantics to the act_q8 GEMV staging / ref_numpy(act_q8=True). mode 1
// applies silu(gate)*up first (X is [M][2K]: gate | up halves).
// 8 lanes per block (lane-parallel; see k_gemv_q8 staging).
__global__ __launch_bounds__(256) void k_quant_rows(
    const float* __restrict__ X, int8_t* __restrict__ xq,
    float* __restrict__ xs, float* __restrict__ xsum,
    int K, int ldx, int mode) {
    const int m = blockIdx.x;
    const int tid = threadIdx.x;
    const int jl = tid & 7;
    const int NB = K / 32;
    const float* xrow = X + (size_t)m * ldx;
    for (int blk = tid >> 3; blk < NB; blk += 32) {
        float4 v;
        if (mode == 1) {
            const float4 g = reinterpret_cast<const float4*>(
                xrow + blk * 32)[jl];
            const float4 u = reinterpret_cast<const float4*>(
                xrow + K + blk * 32)[jl];
            v.x = (g.x / (1.f + __expf(-g.x))) * u.x;
            v.y = (g.y / (1.f + __expf(-g.y))) * u.y;
            v.z = (g.z / (1.f + __expf(-g.z))) * u.z;
            v.w = (g.w / (1.f + __expf(-g.w))) * u.w;
        } else {
            v = reinterpret_cast<const float4*>(xrow + blk * 32)[jl];
        }
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
        int s = q0 + q1 + q2 + q3;
        #pragma unroll
        for (int off = 1; off < 8; off <<= 1) s += __shfl_xor(s, off, 64);
        *reinterpret_cast<uint32_t*>(
            xq + (size_t)m * K + blk * 32 + jl * 4) = packed;
        if (jl == 0) {
            xs[(size_t)m * NB + blk] = scale;
            xsum[(size_t)m * NB + blk] = scale * (float)s;
        }
    }
}

// --------------------------------------------------------- launch stubs

void launch_quant_rows(const float* X, int8_t* xq, float* xs, float* xsum,
                       int M, int K, int ldx, int mode, hipStream_t stream) {
    if (K % 32) throw std::runtime_error("quant_rows: K must be /32");
    hipLaunchKernelGGL(k_quant_rows, dim3(M), dim3(256), 0, stream,
                       X, xq, xs, xsum, K, ldx, mode);
}

bool gemm_i8_supported(DT dtype, int M, int K) {
    return (dtype == DT::DQ4K || dtype == DT::DQ8) && M <= 128 &&
           K % BK == 0 && K % 256 == 0;
}

void launch_gemm_i8(const WTensor& w, const int8_t* xq, const float* xs,
                    const float* xsum, int ldxq, const float* res, float* C,
                    int M, int ldc, hipStream_t stream) {
    const int N = (int)w.n, K = (int)w.k;
    if (!gemm_i8_supported(w.dtype, M, K))
        throw std::runtime_error("gemm_i8: unsupported dtype/shape");
    const bool bm16 = M <= 16;
    const int bm_tiles = bm16 ? 1 : (M + 31) / 32;
    const int n_tiles = (N + BN - 1) / BN;
    const int splitk = gemm_splitk_factor(N, K, M);
    const int k_chunk = ((K / BK + splitk - 1) / splitk) * BK;
    dim3 grid(n_tiles, bm_tiles, splitk), block(256);
    #define GI8_ONE(WT, BMV)                                                   \
        hipLaunchKernelGGL((k_gemm_i8<WT, BMV>), grid, block, 0, stream,       \
            (const uint8_t*)w.qs, (const uint8_t*)w.hdr, xq, xs, xsum,         \
            res, C, M, N, K, ldc, ldxq, k_chunk)
    switch (w.dtype) {
        case DT::DQ4K:
            if (bm16) GI8_ONE(DT::DQ4K, 16); else GI8_ONE(DT::DQ4K, 32);
            break;
        case DT::DQ8:
            if (bm16) GI8_ONE(DT::DQ8, 16); else GI8_ONE(DT::DQ8, 32);
            break;
        default: throw std::runtime_error("gemm_i8: quant dtypes only");
    }
    #undef GI8_ONE
}

// ----------------------------------------------------------- layout probe
// One-wave C[16][16] = A[16][32] x B[32][16] via mfma_i32_16x16x32_i8 under
// the assumed (bf16-analogous) lane maps; tests diff vs numpy int math.
__global__ __launch_bounds__(64) void k_mfma_probe_i8(
    const int8_t* __restrict__ A, const int8_t* __restrict__ B,
    int32_t* __restrict__ C) {
    const int lane = threadIdx.x & 63;
    const int half = lane >> 4, idx = lane & 15;
    int8_t av[8], bv[8];
    #pragma unroll
    for (int i = 0; i < 8; i++) {
        const int k = half * 8 + i;
        av[i] = A[idx * 32 + k];
        bv[i] = B[k * 16 + idx];
    }
    const long a = *reinterpret_cast<const long*>(av);
    const long b = *reinterpret_cast<const long*>(bv);
    v4i acc = {0, 0, 0, 0};
    acc = __builtin_amdgcn_mfma_i32_16x16x32_i8(a, b, acc, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; r++) C[(half * 4 + r) * 16 + idx] = acc[r];
}

void launch_mfma_probe_i8_test(const int8_t* A, const int8_t* B, int32_t* C) {
    void *dA, *dB, *dC;
    HIP_CHECK(hipMalloc(&dA, 16 * 32));
    HIP_CHECK(hipMalloc(&dB, 32 * 16));
    HIP_CHECK(hipMalloc(&dC, 16 * 16 * 4));
    HIP_CHECK(hipMemcpy(dA, A, 16 * 32, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dB, B, 32 * 16, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_mfma_probe_i8, dim3(1), dim3(64), 0, nullptr,
                       (const int8_t*)dA, (const int8_t*)dB, (int32_t*)dC);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(C, dC, 16 * 16 * 4, hipMemcpyDeviceToHost));
    (void)hipFree(dA); (void)hipFree(dB); (void)hipFree(dC);
}

// -------------------------------------------------------- test entry point
// Full path: quantize X rows, run the i8 GEMM (host buffers in/out).
void launch_gemm_i8_test(const void* qs, const void* hdr, const float* x,
                         float* y, int dtype, int M, int N, int K,
                         size_t qs_bytes, size_t hdr_bytes) {
    const DT dt = static_cast<DT>(dtype);
    void *d_qs = nullptr, *d_hdr = nullptr;
    HIP_CHECK(hipMalloc(&d_qs, qs_bytes));
    HIP_CHECK(hipMemcpy(d_qs, qs, qs_bytes, hipMemcpyHostToDevice));
    HIP_CHECK(hipMalloc(&d_hdr, hdr_bytes));
    HIP_CHECK(hipMemcpy(d_hdr, hdr, hdr_bytes, hipMemcpyHostToDevice));
    float* d_x = nullptr;
    HIP_CHECK(hipMalloc((void**)&d_x, (size_t)M * K * 4));
    HIP_CHECK(hipMemcpy(d_x, x, (size_t)M * K * 4, hipMemcpyHostToDevice));
    int8_t* d_xq = nullptr;
    float *d_xs = nullptr, *d_xsum = nullptr, *d_y = nullptr;
    HIP_CHECK(hipMalloc((void**)&d_xq, (size_t)M * K));
    HIP_CHECK(hipMalloc((void**)&d_xs, (size_t)M * (K / 32) * 4));
    HIP_CHECK(hipMalloc((void**)&d_xsum, (size_t)M * (K / 32) * 4));
    HIP_CHECK(hipMalloc((void**)&d_y, (size_t)M * N * 4));
    launch_quant_rows(d_x, d_xq, d_xs, d_xsum, M, K, K, 0, nullptr);
    WTensor w;
    w.dtype = dt; w.n = N; w.k = K; w.qs = d_qs; w.hdr = d_hdr;
    if (gemm_splitk_factor(N, K, M) > 1)
        HIP_CHECK(hipMemset(d_y, 0, (size_t)M * N * 4));
    launch_gemm_i8(w, d_xq, d_xs, d_xsum, K, nullptr, d_y, M, N, nullptr);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(y, d_y, (size_t)M * N * 4, hipMemcpyDeviceToHost));
    (void)hipFree(d_qs); (void)hipFree(d_hdr); (void)hipFree(d_x);
    (void)hipFree(d_xq); (void)hipFree(d_xs); (void)hipFree(d_xsum);
    (void)hipFree(d_y);
}

}  // namespace cla
