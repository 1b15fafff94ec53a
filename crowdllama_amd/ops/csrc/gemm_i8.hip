// int8-activation MFMA dequant-GEMM for gfx950: the batched-decode hot path.
//
// C[M,N] = Xq[M,K] @ W[N,K]^T (+res) where W is Q4_K or Q8_0 and X was
// pre-quantized to int8 per 32-block (k_quant_rows below; same semantics as
// the act_q8 GEMV / ref_numpy(act_q8=True): x ~ rint(x/xd)*xd).
//
// Design (v2, after the round-2 register-staged version measured the same
// ~1.1 TB/s wall as the bf16 kernel — the bottleneck was staging latency,
// not dequant VALU):
// - W stages as RAW quant bytes via `global_load_lds` DMA (nt policy): no
//   VGPR round-trip, no unpack in staging, 0.5 B/weight of LDS for Q4_K.
//   Nibble unpack happens at fragment read (2-4 VALU per 8 weights), and
//   one 8-byte LDS read serves both K=32 halves of a BK=64 tile.
// - raw `s_barrier` + counted `s_waitcnt vmcnt(N)` keep the DMA pipeline
//   across barriers (the guide's 8-phase discipline; __syncthreads would
//   drain vmcnt(0) while glds is in flight).
// - v_mfma_i32_16x16x32_i8 computes the exact i32 dot; per-32 scales are
//   applied at a K=32 drain:
//     y += d_w*d_x*dot_i32 - m_w*(d_x*sum qx)   (Q4_K; Q8_0 has m_w=0)
//   Block headers and activation scales are read from global (L1/L2
//   broadcast path), prefetched one tile ahead into registers.
// - activation scales live TRANSPOSED (xsT/xsumT [K/32][M4], M4 = M
//   rounded to 4) so a drain reads its 4 slots as one float4.
//
// Replaces (functionally) llama.cpp's MMQ path for batched decode — the
// compute the reference delegates to Ollama; designed for
// CDNA4 wave64/XCD/LDS-DMA geometry, not ported.
#include "common.h"

namespace cla {

typedef int v4i __attribute__((ext_vector_type(4)));

int gemm_splitk_factor(int N, int K, int M);  // gemm.hip (shared contract)

namespace {

constexpr int BN = 128, BK = 64;
constexpr int LDXQ = BK + 16;   // X LDS row stride (conflict-free b64)

__device__ __forceinline__ float f16b2f(uint32_t h) {
    __half_raw r;
    r.x = (uint16_t)h;
    return __half2float(*reinterpret_cast<__half*>(&r));
}

}  // namespace

// BM_ in {16, 32}. 256 threads = 4 waves; BM=16 -> 1x4 wave grid (each wave
// all 16 M-rows x 32 cols), BM=32 -> 2x2 (16 rows x 64 cols per wave).
// Split-K accumulates into pre-zeroed C via atomicAdd (identical contract
// to gemm.hip's k_gemm: gemm_uses_splitk tells the caller to pre-zero).
template <DT W, int BM_>
__global__ __launch_bounds__(256) void k_gemm_i8(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const int8_t* __restrict__ xq,    // [M][ldxq] int8 (pre-quantized)
    const float* __restrict__ xsT,    // [ldxq/32][M4] block scales
    const float* __restrict__ xsumT,  // [ldxq/32][M4] dx*sum(qx) per block
    const float* __restrict__ res,    // [M][ldc] or null
    float* __restrict__ C,            // [M][ldc]
    int M, int N, int K, int ldc, int ldxq, int k_chunk) {
    constexpr int WMW = (BM_ == 16) ? 1 : 2;
    constexpr int WNW = 4 / WMW;
    constexpr int JF = BN / WNW / 16;        // b fragments per wave (2 or 4)
    constexpr int RAWB = (W == DT::DQ4K) ? BK / 2 : BK;  // raw bytes/row
    constexpr int NGL = (BN / 4) * RAWB / 1024;          // glds per wave
    __shared__ __attribute__((aligned(16))) int8_t Wr[2][BN * RAWB];
    __shared__ __attribute__((aligned(16))) int8_t Xl[2][BM_ * LDXQ];

    const int tid = threadIdx.x;
    const int bn = blockIdx.x, bm = blockIdx.y, bz = blockIdx.z;
    const bool splitk = gridDim.z > 1;
    const int m0 = bm * BM_, n0 = bn * BN;
    const int kb_lo = bz * k_chunk;
    const int kb_hi = min(kb_lo + k_chunk, K);
    if (kb_lo >= kb_hi) return;              // empty trailing z-block
    const int wid = tid >> 6, lane = tid & 63;
    const int wm = (WMW == 1) ? 0 : (wid >> 1);
    const int wn = (WMW == 1) ? wid : (wid & 1);
    const int lrow = lane & 15, lk = lane >> 4;
    const int M4 = (M + 3) & ~3;
    const int nb32 = ldxq / 32;

    float facc[JF][4];
    #pragma unroll
    for (int j = 0; j < JF; j++)
        #pragma unroll
        for (int r = 0; r < 4; r++) facc[j][r] = 0.f;

    // ---- W DMA addressing: wave wid owns rows [wid*32, wid*32+32) ----
    // Q4K: 32 B/row -> lane covers row wid*32+(l>>1), 16B half (l&1).
    // Q8:  64 B/row, two glds: glds g covers rows wid*32+g*16+(l>>2),
    //      16B quarter (l&3).
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);
    int64_t wrow_g[NGL];     // this lane's global W row per glds
    int wboff[NGL];          // byte offset of this lane's 16B within the row
    #pragma unroll
    for (int g = 0; g < NGL; g++) {
        int rl;
        if constexpr (W == DT::DQ4K) {
            rl = wid * 32 + (lane >> 1);
            wboff[g] = (lane & 1) * 16;
        } else {
            rl = wid * 32 + g * 16 + (lane >> 2);
            wboff[g] = (lane & 3) * 16;
        }
        const int64_t gn = (int64_t)n0 + rl;
        wrow_g[g] = gn < N ? gn : N - 1;
    }

    auto issue_w_glds = [&](int kb, int pb) {
        // per-row byte offset of the BK window's raw bytes
        #pragma unroll
        for (int g = 0; g < NGL; g++) {
            int64_t off;
            if constexpr (W == DT::DQ4K) {
                const int sb = kb >> 8, q = (kb & 255) >> 6;
                off = wrow_g[g] * qs_rb + sb * 128 + q * 32 + wboff[g];
            } else {
                off = wrow_g[g] * qs_rb + kb + wboff[g];
            }
            __builtin_amdgcn_global_load_lds(
                reinterpret_cast<const uint32_t*>(qs + off),
                reinterpret_cast<uint32_t*>(
                    Wr[pb] + wid * (NGL * 1024) + g * 1024),
                16, 0, 2 /* nt: streamed once */);
        }
    };

    // ---- X register staging (tiny: BM_*64 B/tile) ----
    const int xrow = tid >> 2, xseg = tid & 3;
    const bool xwave = xrow < BM_;           // wave-uniform (wave 0 / 0..1)
    const int xgm = (m0 + xrow < M) ? m0 + xrow : (M > 0 ? M - 1 : 0);
    const bool xvalid = (m0 + xrow) < M && xrow < BM_;
    uint4 xregs = {0, 0, 0, 0};
    auto load_x = [&](int kb) {
        if (xwave)
            xregs = *reinterpret_cast<const uint4*>(
                xq + (size_t)xgm * ldxq + kb + xseg * 16);
    };
    auto write_x = [&](int pb) {
        if (xwave) {
            uint4 v = xvalid ? xregs : uint4{0, 0, 0, 0};
            *reinterpret_cast<uint4*>(Xl[pb] + xrow * LDXQ + xseg * 16) = v;
        }
    };

    // ---- scale prefetch (global -> regs, two tiles ahead) ----
    // W headers: per fragment j, the (col, q-group) pair header. Q4K: uint2
    // {d,dmin | sc/mn x2} covers BOTH K=32 halves. Q8: one u32 = two f16 d.
    // Parity-indexed slots: tile t uses slot t&1; the slot is refilled with
    // tile t+2's scales right after t consumes it (never overwrites t+1's).
    uint2 hd_s[2][JF];
    float4 dx_s[2][2], sm_s[2][2];
    int64_t hcol[JF];
    #pragma unroll
    for (int j = 0; j < JF; j++) {
        const int64_t gn = (int64_t)n0 + wn * (BN / WNW) + j * 16 + lrow;
        hcol[j] = gn < N ? gn : N - 1;
    }
    // Exactly this many VMEM instructions per wave per load_scales call
    // (every wave issues the same count — vmcnt literals depend on it).
    constexpr int RSC = JF + 4;
    auto load_scales = [&](int kb, int slot) {
        #pragma unroll
        for (int j = 0; j < JF; j++) {
            if constexpr (W == DT::DQ4K) {
                const int sb = kb >> 8, q = (kb & 255) >> 6;
                hd_s[slot][j] = *reinterpret_cast<const uint2*>(
                    hdr + hcol[j] * hdr_rb + sb * 32 + q * 8);
            } else {
                hd_s[slot][j].x = *reinterpret_cast<const uint32_t*>(
                    hdr + hcol[j] * hdr_rb + (kb >> 5) * 2);
                hd_s[slot][j].y = 0;
            }
        }
        const int kg = kb >> 5;
        const int mrow = m0 + wm * 16 + lk * 4;
        #pragma unroll
        for (int k2 = 0; k2 < 2; k2++) {
            dx_s[slot][k2] = *reinterpret_cast<const float4*>(
                xsT + (size_t)(kg + k2) * M4 + mrow);
            sm_s[slot][k2] = *reinterpret_cast<const float4*>(
                xsumT + (size_t)(kg + k2) * M4 + mrow);
        }
    };

    auto mfma_tile = [&](int pb, int slot) {
        // raw W fragment bytes: one b64 per j serves both K=32 halves (Q4K)
        long rawj[JF][(W == DT::DQ4K) ? 1 : 2];
        #pragma unroll
        for (int j = 0; j < JF; j++) {
            const int r = wn * (BN / WNW) + j * 16 + lrow;
            if constexpr (W == DT::DQ4K) {
                rawj[j][0] = *reinterpret_cast<const long*>(
                    Wr[pb] + r * RAWB + lk * 8);
            } else {
                rawj[j][0] = *reinterpret_cast<const long*>(
                    Wr[pb] + r * RAWB + lk * 8);
                rawj[j][1] = *reinterpret_cast<const long*>(
                    Wr[pb] + r * RAWB + 32 + lk * 8);
            }
        }
        #pragma unroll
        for (int kb2 = 0; kb2 < 2; kb2++) {
            const long a = *reinterpret_cast<const long*>(
                Xl[pb] + (wm * 16 + lrow) * LDXQ + kb2 * 32 + lk * 8);
            const float4 dx4 = dx_s[slot][kb2], sm4 = sm_s[slot][kb2];
            #pragma unroll
            for (int j = 0; j < JF; j++) {
                long b;
                float d, m;
                if constexpr (W == DT::DQ4K) {
                    b = (kb2 == 0)
                            ? (rawj[j][0] & 0x0F0F0F0F0F0F0F0FLL)
                            : ((rawj[j][0] >> 4) & 0x0F0F0F0F0F0F0F0FLL);
                    const float dd = f16b2f(hd_s[slot][j].x & 0xFFFF);
                    const float dmin = f16b2f(hd_s[slot][j].x >> 16);
                    const uint32_t y = hd_s[slot][j].y;
                    d = dd * (float)((y >> (16 * kb2)) & 0xFF);
                    m = dmin * (float)((y >> (16 * kb2 + 8)) & 0xFF);
                } else {
                    b = rawj[j][kb2];
                    d = f16b2f((hd_s[slot][j].x >> (16 * kb2)) & 0xFFFF);
                    m = 0.f;
                }
                v4i c = {0, 0, 0, 0};
                c = __builtin_amdgcn_mfma_i32_16x16x32_i8(a, b, c, 0, 0, 0);
                facc[j][0] += d * dx4.x * (float)c[0] - m * sm4.x;
                facc[j][1] += d * dx4.y * (float)c[1] - m * sm4.y;
                facc[j][2] += d * dx4.z * (float)c[2] - m * sm4.z;
                facc[j][3] += d * dx4.w * (float)c[3] - m * sm4.w;
            }
        }
    };

    // ---- prologue: tile 0 staged, tile 1 in flight ----
    const int kb_last = kb_hi - BK;          // all tiles full (K%BK==0)
    auto clamp_kb = [&](int kb) { return kb <= kb_last ? kb : kb_last; };
    load_x(kb_lo);
    load_scales(kb_lo, 0);
    issue_w_glds(kb_lo, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    write_x(0);
    load_x(clamp_kb(kb_lo + BK));
    issue_w_glds(clamp_kb(kb_lo + BK), 1);
    load_scales(clamp_kb(kb_lo + BK), 1);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    // steady state at iteration t (buffer pb, scale slot t&1):
    //   buf[pb] landed for every wave; buf[1-pb] DMA in flight;
    //   xregs hold X(t+1); scale slots hold t (t&1) and t+1 (1-(t&1)).
    // Tail iterations clamp their t+2 prefetches to the last tile (the
    // re-staged bytes are never read) so the wait/barrier pattern stays
    // uniform with no divergent branches around loads.
    int pb = 0, slot = 0;
    for (int kb = kb_lo; kb < kb_hi; kb += BK) {
        write_x(1 - pb);                     // X of tile t+1 (regs ready)
        mfma_tile(pb, slot);
        if (kb + BK >= kb_hi) break;         // last tile: no more staging
        const int kb2 = clamp_kb(kb + 2 * BK);
        load_x(kb2);
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();        // buf[pb] fully consumed
        issue_w_glds(kb2, pb);
        load_scales(kb2, slot);              // slot t&1 just consumed
        // retire everything older than [t+2 glds + t+2 scale loads]:
        // exactly the t+1 glds (and the older t+2 xregs loads)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(NGL + RSC) : "memory");
        __builtin_amdgcn_s_barrier();        // everyone's t+1 landed
        pb ^= 1;
        slot ^= 1;
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
// 8 lanes per block; scales/sums written TRANSPOSED ([K/32][M4]) so the
// GEMM drain reads 4 slots as one float4. Grid (ceil(K/32/32), M).
__global__ __launch_bounds__(256) void k_quant_rows(
    const float* __restrict__ X, int8_t* __restrict__ xq,
    float* __restrict__ xsT, float* __restrict__ xsumT,
    int K, int ldx, int mode, int M4) {
    const int m = blockIdx.y;
    const int tid = threadIdx.x;
    const int jl = tid & 7;
    const int NB = K / 32;
    const int blk = blockIdx.x * 32 + (tid >> 3);
    if (blk >= NB) return;
    const float* xrow = X + (size_t)m * ldx;
    float4 v;
    if (mode == 1) {
        const float4 g = reinterpret_cast<const float4*>(xrow + blk * 32)[jl];
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
    *reinterpret_cast<uint32_t*>(xq + (size_t)m * K + blk * 32 + jl * 4) =
        packed;
    if (jl == 0) {
        xsT[(size_t)blk * M4 + m] = scale;
        xsumT[(size_t)blk * M4 + m] = scale * (float)s;
    }
}

// --------------------------------------------------------- launch stubs

void launch_quant_rows(const float* X, int8_t* xq, float* xs, float* xsum,
                       int M, int K, int ldx, int mode, hipStream_t stream) {
    if (K % 32) throw std::runtime_error("quant_rows: K must be /32");
    const int M4 = (M + 3) & ~3;
    dim3 grid((K / 32 + 31) / 32, M), block(256);
    hipLaunchKernelGGL(k_quant_rows, grid, block, 0, stream,
                       X, xq, xs, xsum, K, ldx, mode, M4);
}

bool gemm_i8_supported(DT dtype, int M, int K) {
    return (dtype == DT::DQ4K || dtype == DT::DQ8) && M <= 128 &&
           K % BK == 0 && K % 256 == 0;
}

void launch_gemm_i8(const WTensor& w, const int8_t* xq, const float* xs,
                    const float* xsum, int ldxq, const float* res, float* C,
                    int M, int ldc, hipStream_t stream, int force_splitk) {
    const int N = (int)w.n, K = (int)w.k;
    if (!gemm_i8_supported(w.dtype, M, K))
        throw std::runtime_error("gemm_i8: unsupported dtype/shape");
    const bool bm16 = M <= 16;
    const int bm_tiles = bm16 ? 1 : (M + 31) / 32;
    const int n_tiles = (N + BN - 1) / BN;
    const int splitk =
        force_splitk > 0 ? force_splitk : gemm_splitk_factor(N, K, M);
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
                         size_t qs_bytes, size_t hdr_bytes,
                         int force_splitk) {
    const DT dt = static_cast<DT>(dtype);
    const int M4 = (M + 3) & ~3;
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
    HIP_CHECK(hipMalloc((void**)&d_xs, (size_t)M4 * (K / 32) * 4));
    HIP_CHECK(hipMalloc((void**)&d_xsum, (size_t)M4 * (K / 32) * 4));
    HIP_CHECK(hipMalloc((void**)&d_y, (size_t)M * N * 4));
    HIP_CHECK(hipMemset(d_xs, 0, (size_t)M4 * (K / 32) * 4));
    HIP_CHECK(hipMemset(d_xsum, 0, (size_t)M4 * (K / 32) * 4));
    launch_quant_rows(d_x, d_xq, d_xs, d_xsum, M, K, K, 0, nullptr);
    WTensor w;
    w.dtype = dt; w.n = N; w.k = K; w.qs = d_qs; w.hdr = d_hdr;
    const int sk = force_splitk > 0 ? force_splitk : gemm_splitk_factor(N, K, M);
    if (sk > 1)
        HIP_CHECK(hipMemset(d_y, 0, (size_t)M * N * 4));
    launch_gemm_i8(w, d_xq, d_xs, d_xsum, K, nullptr, d_y, M, N, nullptr,
                   force_splitk);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(y, d_y, (size_t)M * N * 4, hipMemcpyDeviceToHost));
    (void)hipFree(d_qs); (void)hipFree(d_hdr); (void)hipFree(d_x);
    (void)hipFree(d_xq); (void)hipFree(d_xs); (void)hipFree(d_xsum);
    (void)hipFree(d_y);
}

}  // namespace cla
