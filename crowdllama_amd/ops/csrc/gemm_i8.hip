// int8-activation MFMA dequant-GEMM for gfx950: the batched-decode hot path.
//
// C[M,N] = Xq[M,K] @ W[N,K]^T (+res) where W is Q4_K or Q8_0 and X was
// pre-quantized to int8 per 32-block (k_quant_rows below; same semantics as
// the act_q8 GEMV / ref_numpy(act_q8=True): x ~ rint(x/xd)*xd).
//
// Design (v3 — ALL k-loop operands stream by `global_load_lds` DMA):
// Round-2 history: the register-staged v1 and the mixed glds/register v2
// both plateaued at the bf16 kernel's ~1.1 TB/s because of two .s-level
// traps the CDNA guide documents: a second __shared__ object, and any
// ordinary VGPR-destination load inside a glds loop, each make hipcc wait
// vmcnt(0) mid-pipeline (confirmed in this kernel's disassembly). v3:
// - ONE __shared__ array; W raw bytes, the X int8 tile, W block headers
//   (via hdr2, an upload-time transposed header copy — common.h), and the
//   interleaved activation scales ALL arrive by glds (nt on the weight
//   stream), so the k-loop contains no compiler-counted loads at all.
// - raw `s_barrier` + counted `s_waitcnt vmcnt(N)` with an exact per-tile
//   glds count (uniform across waves) keep one full tile of DMA in flight
//   across every barrier.
// - v_mfma_i32_16x16x32_i8 computes the exact i32 dot; per-32 scales are
//   applied at a K=32 drain:
//     y += d_w*d_x*dot_i32 - m_w*(d_x*sum qx)   (Q4_K; Q8_0 has m_w=0)
//   Q4_K weights stay as raw nibbles in LDS (0.5 B/weight): one 8-byte
//   read serves both K=32 halves (lo/hi nibble planes of the same bytes).
//
// Replaces (functionally) llama.cpp's MMQ path for batched decode — the
// compute the reference delegates to Ollama; designed for
// CDNA4 wave64/XCD/LDS-DMA geometry, not ported.
#include "common.h"

#include <cstring>
#include <vector>

namespace cla {

typedef int v4i __attribute__((ext_vector_type(4)));

int gemm_splitk_factor(int N, int K, int M);  // gemm.hip (shared contract)

namespace {

constexpr int BN = 128, BK = 64;

__device__ __forceinline__ float f16b2f(uint32_t h) {
    __half_raw r;
    r.x = (uint16_t)h;
    return __half2float(*reinterpret_cast<__half*>(&r));
}

// LDS-DMA issued through inline asm (guide §5.7 glds16_asm recipe): the
// builtin form is tracked by hipcc's waitcnt pass, which inserts a
// conservative vmcnt(0) before the loop-header ds_reads whenever a glds is
// outstanding across the backedge — draining the prefetched tile every
// iteration (confirmed in this kernel's .s). Inside asm the DMA is
// invisible; ALL vmcnt accounting is done by this kernel's own counted
// waits. M0 (the LDS destination base) is written and restored in the same
// statement; `lds_off` must be wave-uniform (readfirstlane at the caller).
__device__ __forceinline__ void glds16(const void* gsrc, unsigned lds_off) {
    unsigned keep;
    asm volatile(
        "s_mov_b32 %0, m0\n\t"
        "s_mov_b32 m0, %2\n\t"
        "s_nop 0\n\t"
        "global_load_lds_dwordx4 %1, off\n\t"
        "s_mov_b32 m0, %0"
        : "=&s"(keep) : "v"(gsrc), "s"(lds_off) : "memory");
}

__device__ __forceinline__ void glds16_nt(const void* gsrc,
                                          unsigned lds_off) {
    unsigned keep;
    asm volatile(
        "s_mov_b32 %0, m0\n\t"
        "s_mov_b32 m0, %2\n\t"
        "s_nop 0\n\t"
        "global_load_lds_dwordx4 %1, off nt\n\t"
        "s_mov_b32 m0, %0"
        : "=&s"(keep) : "v"(gsrc), "s"(lds_off) : "memory");
}

__device__ __forceinline__ void glds4(const void* gsrc, unsigned lds_off) {
    unsigned keep;
    asm volatile(
        "s_mov_b32 %0, m0\n\t"
        "s_mov_b32 m0, %2\n\t"
        "s_nop 0\n\t"
        "global_load_lds_dword %1, off\n\t"
        "s_mov_b32 m0, %0"
        : "=&s"(keep) : "v"(gsrc), "s"(lds_off) : "memory");
}

}  // namespace

// BM_ in {16, 32, 64, 128}. 256 threads = 4 waves; BM=16 -> 1x4 wave grid
// (each wave all 16 M-rows x 32 cols), BM=32 -> 2x2 (16 rows x 64 cols),
// BM=128 -> 2x2 with FM=4 row fragments per wave (prefill tiles: one pass
// over the W stream covers 128 activation rows instead of re-streaming W
// per 32-row tile). BM<=32 keeps per-wave X copies; BM=128 stages ONE
// shared X image cooperatively.
// Split-K accumulates into pre-zeroed C via atomicAdd (identical contract
// to gemm.hip's k_gemm: gemm_uses_splitk tells the caller to pre-zero).
//
// xsc is the interleaved activation-scale array [K/32][2][M4]:
// [kg][0][m] = block scale dx, [kg][1][m] = dx*sum(qx). M4 = M round-to-4.
template <DT W, int BM_>
__global__ __launch_bounds__(256) void k_gemm_i8(
    const uint8_t* __restrict__ qs2, const uint8_t* __restrict__ hdr2,
    const int8_t* __restrict__ xq,    // [M][ldxq] int8 (pre-quantized)
    const float* __restrict__ xsc,    // [ldxq/32][2][M4]
    const float* __restrict__ res,    // [M][ldc] or null
    float* __restrict__ C,            // [M][ldc]
    int M, int N, int K, int ldc, int ldxq, int k_chunk) {
    constexpr int WMW = (BM_ == 16) ? 1 : 2;
    constexpr int WNW = 4 / WMW;
    constexpr int JF = BN / WNW / 16;        // b fragments per wave (2 or 4)
    constexpr int FM = (BM_ / WMW) / 16;     // a fragments per wave (1 or 4)
    constexpr bool XSHARED = BM_ > 32;       // one X image vs per-wave
    constexpr int RAWB = (W == DT::DQ4K) ? BK / 2 : BK;  // raw bytes/row
    // BM<=32: W staging is wave-PRIVATE — each wave DMAs exactly the
    // (BN/WNW) fragment columns it consumes, so NO cross-wave LDS
    // dependency exists anywhere and the k-loop runs with counted
    // per-wave vmcnt and NO BARRIER AT ALL (the s_barrier arrival skew
    // was the dominant parked share: 52.6% of decode GEMM wave cycles).
    // BM=16 (WMW=1) duplicates nothing (4 waves own 4 disjoint column
    // quarters); BM=32 (WMW=2) stages each column half twice (+WSZ LDS
    // per buffer, L2 absorbs the 2x reads). BM>=64 keeps the shared
    // image + barrier.
    constexpr bool PRIVW = !XSHARED;
    constexpr int WPRIV = (BN / WNW) * RAWB;     // private W bytes/wave
    constexpr int NGW = PRIVW ? WPRIV / 1024
                              : (BN / 4) * RAWB / 1024;  // W glds/wave
    constexpr int NGX = XSHARED ? BM_ * BK / 4096
                                : BM_ * BK / 1024;   // X glds per wave
    // hdr2 glds/wave: Q4K JF (f32 pairs per 32-block), Q8 JF/2 (f32 per
    // 32-block), Q6K JF (f32 per 16-block: 4 k16 x JF*16 cols x 4B)
    constexpr int NGH = (W == DT::DQ8) ? JF / 2 : JF;
    constexpr int NGS = (FM == 2) ? 2 : 1;       // xsc glds per wave
    constexpr int NGLT = NGW + NGX + NGH + NGS;  // per wave per tile
    // --- single LDS array (a second __shared__ object would make hipcc
    // drain vmcnt(0) before every ds_read — guide §5 trap (a)) ---
    // map (per buffer):
    //   [0, WSZ)                      W raw bytes, row-major RAWB/row
    //   [WSZ, +4*XSZ)                 per-wave X int8 copies, [BM_][64]
    //   [.., +4*1024) (Q4K) / +4*512 (Q8)  per-wave hdr copies
    //   [.., +4*256)                  per-wave xsc copies [kg2][2][16] f32
    constexpr int WSZ = PRIVW ? 4 * WPRIV : BN * RAWB;
    constexpr int XSZ = BM_ * BK;
    constexpr int XTOT = XSHARED ? XSZ : 4 * XSZ;
    constexpr int HSZ = (W == DT::DQ8) ? JF * 128 : JF * 256;
    constexpr int SSZ = FM * 256;
    constexpr int BUFSZ = WSZ + XTOT + 4 * (HSZ + SSZ);
    // 3-deep DMA ring: depth 4 re-tested after the barrier removal and
    // still loses (B=32 7028 -> 5607: the extra buffer's LDS costs a
    // WG/CU of residency, which now matters MORE at split-K 1024)
    constexpr int NBUF = 3;
    __shared__ __attribute__((aligned(16))) int8_t lds[NBUF][BUFSZ];

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

    float facc[FM][JF][4];
    #pragma unroll
    for (int i = 0; i < FM; i++)
        #pragma unroll
        for (int j = 0; j < JF; j++)
            #pragma unroll
            for (int r = 0; r < 4; r++) facc[i][j][r] = 0.f;

    // ---- W source: the GEMM-tiled qs2 copy (common.h) — each (n-block,
    // k-window) tile is contiguous and identical to the LDS image, so a
    // wave's DMA covers whole cachelines (the row-major layout read only
    // 32 of every 128 B here: the round-2 bandwidth wall).
    const int64_t ktiles = K / BK;
    const uint8_t* wtile0 = qs2 + (int64_t)bn * ktiles * (BN * RAWB);
    // cooperative (XSHARED) source/dest; private: slice by wn, copy by wid
    const int wlocal = PRIVW ? wn * WPRIV + lane * 16
                             : wid * (NGW * 1024) + lane * 16;
    const int wldst = PRIVW ? wid * WPRIV : wid * (NGW * 1024);
    // X staging rows. BM<=32: each wave DMAs its own full copy (glds gx
    // covers rows gx*16 + lane>>2). BM=128: ONE shared image; wave wid
    // DMAs rows [wid*32, wid*32+32) (glds gx covers wid*32 + gx*16 + ...).
    int64_t xrow_off[NGX];
    #pragma unroll
    for (int g = 0; g < NGX; g++) {
        int xr;
        if constexpr (XSHARED) {
            xr = wid * (BM_ / 4) + g * 16 + (lane >> 2);
        } else {
            xr = (NGX == 1) ? (lane >> 2) : (g * 16 + (lane >> 2));
        }
        const int gm = (m0 + xr < M) ? m0 + xr : (M > 0 ? M - 1 : 0);
        xrow_off[g] = (int64_t)gm * ldxq + (lane & 3) * 16;
    }
    // hdr2: per-wave copy of the k-window's headers for all BN cols.
    // Q4K: 128 cols x 8B = 1KB, lane takes cols {2l, 2l+1} (16B).
    // Q8: 2 x (128 cols x 2B f16) = 2 x 256B dword-glds, lane 4B.
    // xsc: 4 x 64B chunks = 256B dword-glds: chunk c = (kg2, dx|sum),
    //      lane l: chunk l>>4, slot (l&15) of this wave's 16 M-rows.

    // integer LDS offsets only — pointer-typed locals into __shared__
    // decay to GENERIC address space and the reads compile to flat_load
    // (caught in this kernel's v3 disassembly: zero ds_read in the loop)
    const int xo = XSHARED ? WSZ : WSZ + wid * XSZ;      // read base
    const int xow = XSHARED ? WSZ + wid * (XSZ / 4) : xo;  // DMA dest base
    const int ho = WSZ + XTOT + wid * HSZ;
    const int so = WSZ + XTOT + 4 * HSZ + wid * SSZ;

    auto issue_tile = [&](int kb, int pb) {
        // W raw (nt: streamed once per step); source offset == LDS offset
        {
            const uint8_t* wt = wtile0 + (int64_t)(kb / BK) * (BN * RAWB);
            #pragma unroll
            for (int g = 0; g < NGW; g++)
                glds16_nt(wt + wlocal + g * 1024,
                          __builtin_amdgcn_readfirstlane((unsigned)(size_t)
                              &lds[pb][wldst + g * 1024]));
        }
        // X tile (copy per wave, or this wave's quarter of the shared one)
        #pragma unroll
        for (int g = 0; g < NGX; g++) {
            glds16(xq + xrow_off[g] + kb,
                   __builtin_amdgcn_readfirstlane((unsigned)(size_t)
                       &lds[pb][xow + g * 1024]));
        }
        // headers: hdr2 is pre-decoded f32 ({d*sc, dmin*mn} pairs for
        // Q4K, d for Q8) laid out [K/32][N]; each wave stages only its own
        // fragment columns
        {
            const int kg = kb >> 5;
            const int cb = n0 + wn * (BN / WNW);   // this wave's first col
            if constexpr (W == DT::DQ4K) {
                #pragma unroll
                for (int kb2 = 0; kb2 < 2; kb2++)
                    #pragma unroll
                    for (int gg = 0; gg < JF / 2; gg++)
                        glds4(hdr2 + ((int64_t)(kg + kb2) * N + cb +
                                      gg * 32 + (lane >> 1)) * 8 +
                                  (lane & 1) * 4,
                              __builtin_amdgcn_readfirstlane(
                                  (unsigned)(size_t)&lds[pb][
                                      ho + kb2 * (JF * 128) + gg * 256]));
            } else if constexpr (W == DT::DQ6K && JF == 2) {
                // per-16 scales: 4 k16 blocks x 32 cols x f32 = 512B per
                // wave = 2 glds4 (each: 2 k16 blocks x 32 cols)
                const int kg16 = kb >> 4;
                #pragma unroll
                for (int g = 0; g < 2; g++)
                    glds4(hdr2 + ((int64_t)(kg16 + g * 2 + (lane >> 5)) * N
                                  + cb + (lane & 31)) * 4,
                          __builtin_amdgcn_readfirstlane(
                              (unsigned)(size_t)&lds[pb][ho + g * 256]));
            } else if constexpr (W == DT::DQ6K) {  // JF == 4 (BM32/128)
                const int kg16 = kb >> 4;
                #pragma unroll
                for (int g = 0; g < 4; g++)
                    glds4(hdr2 + ((int64_t)(kg16 + g) * N + cb + lane) * 4,
                          __builtin_amdgcn_readfirstlane(
                              (unsigned)(size_t)&lds[pb][ho + g * 256]));
            } else if constexpr (JF == 2) {  // Q8 BM16: one glds, both kb2
                glds4(hdr2 + ((int64_t)(kg + (lane >> 5)) * N + cb +
                              (lane & 31)) * 4,
                      __builtin_amdgcn_readfirstlane(
                          (unsigned)(size_t)&lds[pb][ho]));
            } else {                         // Q8 BM32: one glds per kb2
                #pragma unroll
                for (int kb2 = 0; kb2 < 2; kb2++)
                    glds4(hdr2 + ((int64_t)(kg + kb2) * N + cb + lane) * 4,
                          __builtin_amdgcn_readfirstlane(
                              (unsigned)(size_t)&lds[pb][ho + kb2 * 256]));
            }
        }
        // activation scales: per (kg2, arr) chunks of this wave's M rows.
        // Slot indices clamp into the valid [0, M4) range (partial BM
        // tiles): the duplicated values land in LDS slots whose output
        // rows are discarded anyway.
        if constexpr (FM == 1) {
            const int kg = kb >> 5;
            const int c = lane >> 4;
            int srow = m0 + wm * 16 + (lane & 15);
            if (srow > M4 - 1) srow = M4 - 1;
            const int64_t soff =
                ((int64_t)(kg + (c >> 1)) * 2 + (c & 1)) * M4 + srow;
            glds4(xsc + soff,
                  __builtin_amdgcn_readfirstlane((unsigned)(size_t)
                      &lds[pb][so]));
        } else if constexpr (FM == 2) {
            // BM=64: two glds4 (64 lanes x 4B), each covering 2 of the 4
            // (kg2 x {dx,sum}) chunks for this wave's 32 M rows
            const int kg = kb >> 5;
            const int cw = lane >> 5;
            int srow = m0 + wm * (BM_ / WMW) + (lane & 31);
            if (srow > M4 - 1) srow = M4 - 1;
            #pragma unroll
            for (int g = 0; g < 2; g++) {
                const int c = g * 2 + cw;
                const int64_t soff =
                    ((int64_t)(kg + (c >> 1)) * 2 + (c & 1)) * M4 + srow;
                glds4(xsc + soff,
                      __builtin_amdgcn_readfirstlane((unsigned)(size_t)
                          &lds[pb][so + g * 256]));
            }
        } else {  // BM=128: one glds16 (64 lanes x 16B = 1KB) covers all
            // 4 chunks (kg2 x {dx,sum}) of this wave's 64 M rows: lane l
            // -> chunk c = l>>4, slots ((l&15)*4 .. +4)
            const int kg = kb >> 5;
            const int c = lane >> 4;
            int sbase = m0 + wm * (BM_ / WMW) + (lane & 15) * 4;
            if (sbase > M4 - 4) sbase = M4 - 4;
            if (sbase < 0) sbase = 0;
            const int64_t soff =
                ((int64_t)(kg + (c >> 1)) * 2 + (c & 1)) * M4 + sbase;
            glds16(xsc + soff,
                   __builtin_amdgcn_readfirstlane((unsigned)(size_t)
                       &lds[pb][so]));
        }
    };

    constexpr int SROW = BM_ / WMW;             // xsc slots per wave
    auto mfma_tile = [&](int pb) {
        // raw W fragment bytes: one b64 per j serves both K=32 halves (Q4K)
        long rawj[JF][(W == DT::DQ4K) ? 1 : 2];
        (void)0;
        #pragma unroll
        for (int j = 0; j < JF; j++) {
            const int r = PRIVW ? j * 16 + lrow            // private image
                                : wn * (BN / WNW) + j * 16 + lrow;
            const int wbase = PRIVW ? wid * WPRIV : 0;
            rawj[j][0] = *reinterpret_cast<const long*>(
                &lds[pb][wbase + r * RAWB + lk * 8]);
            if constexpr (W == DT::DQ8 || W == DT::DQ6K)
                rawj[j][1] = *reinterpret_cast<const long*>(
                    &lds[pb][wbase + r * RAWB + 32 + lk * 8]);
        }
        #pragma unroll
        for (int kb2 = 0; kb2 < 2; kb2++) {
            long a[FM];
            float4 dx4[FM], sm4[FM];
            #pragma unroll
            for (int i = 0; i < FM; i++) {
                const int xr = wm * (BM_ / WMW) + i * 16 + lrow;
                a[i] = *reinterpret_cast<const long*>(
                    &lds[pb][xo + xr * BK + kb2 * 32 + lk * 8]);
                const int sl = i * 16 + lk * 4;   // wave-local slot base
                dx4[i] = *reinterpret_cast<const float4*>(
                    &lds[pb][so + ((kb2 * 2 + 0) * SROW + sl) * 4]);
                sm4[i] = *reinterpret_cast<const float4*>(
                    &lds[pb][so + ((kb2 * 2 + 1) * SROW + sl) * 4]);
            }
            #pragma unroll
            for (int j = 0; j < JF; j++) {
                const int cl = j * 16 + lrow;   // wave-local column
                long b;
                float d, m;
                if constexpr (W == DT::DQ6K) {
                    // per-16 scales: split the K=32 MFMA into two
                    // lane-masked halves (lanes lk<2 carry k16 block 0,
                    // lk>=2 block 1 — each lane's 8 values lie entirely
                    // within one k16 block) and scale each i32 result by
                    // its block's d*sc16.
                    const long braw = rawj[j][kb2];
                    const long b_lo = (lk < 2) ? braw : 0;
                    const long b_hi = (lk >= 2) ? braw : 0;
                    const int k16base = kb2 * 2;   // k16 index within tile
                    const float d0 = *reinterpret_cast<const float*>(
                        &lds[pb][ho + (k16base + 0) * (JF * 64) + cl * 4]);
                    const float d1 = *reinterpret_cast<const float*>(
                        &lds[pb][ho + (k16base + 1) * (JF * 64) + cl * 4]);
                    #pragma unroll
                    for (int i = 0; i < FM; i++) {
                        v4i c0 = {0, 0, 0, 0}, c1 = {0, 0, 0, 0};
                        c0 = __builtin_amdgcn_mfma_i32_16x16x32_i8(
                            a[i], b_lo, c0, 0, 0, 0);
                        c1 = __builtin_amdgcn_mfma_i32_16x16x32_i8(
                            a[i], b_hi, c1, 0, 0, 0);
                        facc[i][j][0] += dx4[i].x *
                            (d0 * (float)c0[0] + d1 * (float)c1[0]);
                        facc[i][j][1] += dx4[i].y *
                            (d0 * (float)c0[1] + d1 * (float)c1[1]);
                        facc[i][j][2] += dx4[i].z *
                            (d0 * (float)c0[2] + d1 * (float)c1[2]);
                        facc[i][j][3] += dx4[i].w *
                            (d0 * (float)c0[3] + d1 * (float)c1[3]);
                    }
                    continue;
                }
                if constexpr (W == DT::DQ4K) {
                    b = (kb2 == 0)
                            ? (rawj[j][0] & 0x0F0F0F0F0F0F0F0FLL)
                            : ((rawj[j][0] >> 4) & 0x0F0F0F0F0F0F0F0FLL);
                    const float2 dm = *reinterpret_cast<const float2*>(
                        &lds[pb][ho + kb2 * (JF * 128) + cl * 8]);
                    d = dm.x;
                    m = dm.y;
                } else {
                    b = rawj[j][kb2];
                    d = *reinterpret_cast<const float*>(
                        &lds[pb][ho + kb2 * (JF * 64) + cl * 4]);
                    m = 0.f;
                }
                #pragma unroll
                for (int i = 0; i < FM; i++) {
                    v4i c = {0, 0, 0, 0};
                    c = __builtin_amdgcn_mfma_i32_16x16x32_i8(a[i], b, c,
                                                              0, 0, 0);
                    facc[i][j][0] += d * dx4[i].x * (float)c[0] - m * sm4[i].x;
                    facc[i][j][1] += d * dx4[i].y * (float)c[1] - m * sm4[i].y;
                    facc[i][j][2] += d * dx4[i].z * (float)c[2] - m * sm4[i].z;
                    facc[i][j][3] += d * dx4[i].w * (float)c[3] - m * sm4[i].w;
                }
            }
        }
    };

    // ---- DMA pipeline: ONE barrier per tile, two tiles in flight ----
    // At iteration t: wait own tile t (vmcnt leaves t+1), barrier (now
    // everyone's t landed AND everyone finished reading t-1, so buffer
    // (t+2)%3 is free), issue t+2 into it, compute t. The issue for t+2
    // happens before mfma(t), so its DMA has a full tile of work to hide
    // under, and the single barrier serves both release and acquire.
    const int kb_last = kb_hi - BK;          // all tiles full (K%BK==0)
    #pragma unroll
    for (int b = 0; b < NBUF - 1; b++)
        issue_tile(kb_lo + b * BK <= kb_last ? kb_lo + b * BK : kb_lo, b);
    int pb = 0;
    for (int kb = kb_lo; kb < kb_hi; kb += BK) {
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"((NBUF - 2) * NGLT)
                     : "memory");
        if constexpr (!PRIVW)   // private staging: no cross-wave deps
            __builtin_amdgcn_s_barrier();
        if (kb + (NBUF - 1) * BK <= kb_last) {
            const int nb = pb + NBUF - 1 >= NBUF ? pb - 1 : pb + NBUF - 1;
            issue_tile(kb + (NBUF - 1) * BK, nb);
        }
        mfma_tile(pb);
        pb = (pb == NBUF - 1) ? 0 : pb + 1;
    }

    // ---- epilogue (same contract as gemm.hip k_gemm) ----
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
                const float rv =
                    (res && (!splitk || bz == 0)) ? res[idx] : 0.f;
                if (splitk) {
                    atomicAdd(&C[idx], facc[i][j][r] + rv);
                } else {
                    C[idx] = facc[i][j][r] + rv;
                }
            }
        }
    }
}

// ---------------------------------------------------- activation quantizer
// Per-row, per-32-block symmetric int8 (xd = amax/127, rint) — identical
// semantics to the act_q8 GEMV staging / ref_numpy(act_q8=True). mode 1
// applies silu(gate)*up first (X is [M][2K]: gate | up halves).
// 8 lanes per block; scale and dx*sum(qx) go to the interleaved transposed
// array xsc[K/32][2][M4] the GEMM's DMA staging expects.
// Grid (ceil(K/32/32), M).
__global__ __launch_bounds__(256) void k_quant_rows(
    const float* __restrict__ X, int8_t* __restrict__ xq,
    float* __restrict__ xsc, int K, int ldx, int mode, int M4) {
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
        xsc[((size_t)blk * 2 + 0) * M4 + m] = scale;
        xsc[((size_t)blk * 2 + 1) * M4 + m] = scale * (float)s;
    }
}

// --------------------------------------------------------- launch stubs

void launch_quant_rows(const float* X, int8_t* xq, float* xsc, int M, int K,
                       int ldx, int mode, hipStream_t stream) {
    if (K % 32) throw std::runtime_error("quant_rows: K must be /32");
    const int M4 = (M + 3) & ~3;
    dim3 grid((K / 32 + 31) / 32, M), block(256);
    hipLaunchKernelGGL(k_quant_rows, grid, block, 0, stream,
                       X, xq, xsc, K, ldx, mode, M4);
}

bool gemm_i8_supported(DT dtype, int M, int K) {
    // M <= 128: BM=16/32 decode tiles. The BM=128 FM=4 variant exists
    // (template above) but measured SLOWER than both the BM=32-with-
    // re-read i8 path (B=64: 4458 vs 8338 tok/s) and the bf16 prefill
    // kernel for 1024-row chunks (prefill 1541 vs 1382 ms at 16x1024) —
    // its 20 KB/buffer LDS footprint halves occupancy. Kept unselected;
    // see docs/PERF_NOTES.md round-2 notes.
    return (dtype == DT::DQ4K || dtype == DT::DQ8 || dtype == DT::DQ6K) &&
           M <= 128 && K % BK == 0 && K % 256 == 0;
}

void launch_gemm_i8(const WTensor& w, const int8_t* xq, const float* xsc,
                    int ldxq, const float* res, float* C, int M, int ldc,
                    hipStream_t stream, int force_splitk) {
    const int N = (int)w.n, K = (int)w.k;
    if (!gemm_i8_supported(w.dtype, M, K) || !w.hdr2 || !w.qs2)
        throw std::runtime_error("gemm_i8: unsupported dtype/shape");
    int BMSEL = M <= 16 ? 16 : (M <= 128 ? 32 : 128);
    // prefill tile A/B: CLA_I8_BM=64 selects the FM=2 tile for M>32
    // (halves the per-output weight re-stream vs BM=32 at 14 KB/buffer
    // LDS — BM=128's 20 KB halved occupancy and lost)
    if (M > 32) {
        if (const char* e = getenv("CLA_I8_BM")) {
            const int v = atoi(e);
            if (v == 16 || v == 32 || v == 64 || v == 128) BMSEL = v;
        }
    }
    const int bm_tiles = (M + BMSEL - 1) / BMSEL;
    const int n_tiles = (N + BN - 1) / BN;
    const int splitk =
        force_splitk > 0 ? force_splitk : gemm_splitk_factor(N, K, M);
    const int k_chunk = ((K / BK + splitk - 1) / splitk) * BK;
    dim3 grid(n_tiles, bm_tiles, splitk), block(256);
    #define GI8_ONE(WT, BMV)                                                   \
        hipLaunchKernelGGL((k_gemm_i8<WT, BMV>), grid, block, 0, stream,       \
            (const uint8_t*)w.qs2, (const uint8_t*)w.hdr2, xq, xsc,            \
            res, C, M, N, K, ldc, ldxq, k_chunk)
    #define GI8_BM(WT)                                                         \
        do { if (BMSEL == 16) GI8_ONE(WT, 16);                                 \
             else if (BMSEL == 32) GI8_ONE(WT, 32);                            \
             else if (BMSEL == 64) GI8_ONE(WT, 64);                            \
             else GI8_ONE(WT, 128); } while (0)
    switch (w.dtype) {
        case DT::DQ4K: GI8_BM(DT::DQ4K); break;
        case DT::DQ8: GI8_BM(DT::DQ8); break;
        case DT::DQ6K: GI8_BM(DT::DQ6K); break;
        default: throw std::runtime_error("gemm_i8: quant dtypes only");
    }
    #undef GI8_BM
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
// Full path: build hdr2 host-side, quantize X rows, run the i8 GEMM.
void launch_gemm_i8_test(const void* qs, const void* hdr, const float* x,
                         float* y, int dtype, int M, int N, int K,
                         size_t qs_bytes, size_t hdr_bytes,
                         int force_splitk) {
    const DT dt = static_cast<DT>(dtype);
    const int M4 = (M + 3) & ~3;
    void *d_qs = nullptr, *d_h2 = nullptr, *d_q2 = nullptr;
    HIP_CHECK(hipMalloc(&d_qs, qs_bytes));
    HIP_CHECK(hipMemcpy(d_qs, qs, qs_bytes, hipMemcpyHostToDevice));
    {   // GEMM-tiled weight copy
        std::vector<uint8_t> q2((size_t)dqs2_bytes(dt, N, K));
        build_qs2_rows(dt, reinterpret_cast<const uint8_t*>(qs),
                       dqs_row_bytes(dt, K), N, K, 0,
                       (N + I8G_BN - 1) / I8G_BN * I8G_BN, q2.data());
        HIP_CHECK(hipMalloc(&d_q2, q2.size()));
        HIP_CHECK(hipMemcpy(d_q2, q2.data(), q2.size(),
                            hipMemcpyHostToDevice));
    }
    // transposed pre-decoded headers (same builder Engine::upload_pack uses)
    {
        const int64_t h2_rb = dhdr2_row_bytes(dt, K);
        const int64_t hrb = dhdr_row_bytes(dt, K);
        std::vector<uint8_t> h2((size_t)h2_rb * N + 1024, 0);
        (void)hdr_bytes;
        build_hdr2_rows(dt, reinterpret_cast<const uint8_t*>(hdr), hrb, N,
                        K, 0, N, h2.data());
        HIP_CHECK(hipMalloc(&d_h2, h2.size()));
        HIP_CHECK(hipMemcpy(d_h2, h2.data(), h2.size(),
                            hipMemcpyHostToDevice));
    }
    float* d_x = nullptr;
    HIP_CHECK(hipMalloc((void**)&d_x, (size_t)M * K * 4));
    HIP_CHECK(hipMemcpy(d_x, x, (size_t)M * K * 4, hipMemcpyHostToDevice));
    int8_t* d_xq = nullptr;
    float *d_xsc = nullptr, *d_y = nullptr;
    HIP_CHECK(hipMalloc((void**)&d_xq, (size_t)M * K));
    HIP_CHECK(hipMalloc((void**)&d_xsc, (size_t)M4 * (K / 32) * 2 * 4));
    HIP_CHECK(hipMalloc((void**)&d_y, (size_t)M * N * 4));
    HIP_CHECK(hipMemset(d_xsc, 0, (size_t)M4 * (K / 32) * 2 * 4));
    launch_quant_rows(d_x, d_xq, d_xsc, M, K, K, 0, nullptr);
    WTensor w;
    w.dtype = dt; w.n = N; w.k = K; w.qs = d_qs; w.hdr = nullptr;
    w.hdr2 = d_h2; w.qs2 = d_q2;
    const int sk = force_splitk > 0 ? force_splitk
                                    : gemm_splitk_factor(N, K, M);
    if (sk > 1)
        HIP_CHECK(hipMemset(d_y, 0, (size_t)M * N * 4));
    launch_gemm_i8(w, d_xq, d_xsc, K, nullptr, d_y, M, N, nullptr,
                   force_splitk);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(y, d_y, (size_t)M * N * 4, hipMemcpyDeviceToHost));
    (void)hipFree(d_qs); (void)hipFree(d_h2); (void)hipFree(d_q2);
    (void)hipFree(d_x);
    (void)hipFree(d_xq); (void)hipFree(d_xsc); (void)hipFree(d_y);
}

}  // namespace cla
