// HIP/CDNA4 (gfx950) kernels for the crowdllama-amd decode engine.
//
// MI355X-native replacements for the GGML compute the reference delegates to
// Ollama/llama.cpp: fused dequant-GEMV (Q4_K/Q6_K/Q8_0),
// RMSNorm + SwiGLU fused into GEMV activation staging, RoPE + paged-KV
// append, split-KV online-softmax decode attention, embedding gather,
// on-device greedy sampling.
//
// Design notes (see /opt/skills guides):
// - wave64 everywhere; block = 256 threads = 4 waves.
// - GEMV: one wave per output row sweeping the row's quant payload in 16-B
//   chunks (coalesced dwordx4), activations staged once per workgroup into
//   LDS as f32; block headers ride the same-address broadcast path through
//   L1/L2 (no cross-lane shuffles needed).
// - Weights are repacked at upload into split qs/hdr arrays (common.h DT).
// - Attention: each 16-lane quarter-wave owns a full online-softmax
//   accumulator (D 64/128); 4 positions in flight per wave with the next
//   position's K/V prefetched; split-KV partials combined in-kernel by the
//   last-arriving split (agent-scope ticket + release/acquire fences).

#include "common.h"

namespace cla {

// ------------------------------------------------------------------ utils

__device__ __forceinline__ float f16_bits_to_f32(uint32_t h) {
    __half_raw r;
    r.x = static_cast<uint16_t>(h);
    return __half2float(*reinterpret_cast<__half*>(&r));
}

__device__ __forceinline__ float bf16_bits_to_f32(uint32_t h) {
    union { uint32_t u; float f; } v;
    v.u = h << 16;
    return v.f;
}

__device__ __forceinline__ uint16_t f32_to_bf16_bits(float f) {
    union { uint32_t u; float f; } v;
    v.f = f;
    uint32_t r = (v.u + 0x7FFF + ((v.u >> 16) & 1)) >> 16;
    return static_cast<uint16_t>(r);
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
    return v;
}

enum Pre : int { PRE_NONE = 0, PRE_RMS = 1, PRE_SILU = 2 };

// LDS address map for staged activations: +16B pad per 64 floats breaks the
// systematic bank collision of 16/64-aligned chunk bases across a wave's
// lanes (ds_read_b128 bank = dword % 64; unpadded, every lane's k0 % 64 is
// one of {0,16,32,48} -> up-to-8-way conflicts per lane group).
__device__ __forceinline__ int xpad(int k) { return k + ((k >> 6) << 2); }
__device__ __forceinline__ constexpr int xpad_size(int k) {
    return k + (k >> 4);
}

// Weights per 16-B qs chunk for each device dtype.
template <DT W> struct ChunkTraits;
template <> struct ChunkTraits<DT::DQ4K> { static constexpr int W_PER_CHUNK = 32; };
template <> struct ChunkTraits<DT::DQ6K> { static constexpr int W_PER_CHUNK = 16; };
template <> struct ChunkTraits<DT::DQ8>  { static constexpr int W_PER_CHUNK = 16; };
template <> struct ChunkTraits<DT::BF16> { static constexpr int W_PER_CHUNK = 8; };
template <> struct ChunkTraits<DT::F16>  { static constexpr int W_PER_CHUNK = 8; };
template <> struct ChunkTraits<DT::F32>  { static constexpr int W_PER_CHUNK = 4; };

typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));

// Raw per-chunk payload (issued early; decode consumes it later).
template <DT W>
struct ChunkRaw {
    u32x4 qv;
    uint2 hd;       // DQ4K pair header {d, dmin, sc/mn x2}
    float d0, d1;   // DQ6K: two eff scales; DQ8: block scale
};

template <DT W>
__device__ __forceinline__ void load_chunk(
    const uint8_t* __restrict__ qs_row, const uint8_t* __restrict__ hdr_row,
    int c, ChunkRaw<W>* r) {
    // plain (cached) loads: the nontemporal hint measured 2.6 TB/s vs
    // 6.4 TB/s for normal loads on the same stream (scripts/membw.py)
    r->qv = *(
        reinterpret_cast<const u32x4*>(qs_row) + c);
    if constexpr (W == DT::DQ4K) {
        r->hd = reinterpret_cast<const uint2*>(hdr_row)[c >> 1];
    } else if constexpr (W == DT::DQ6K) {
        const int sb = c >> 4, s16 = c & 15;
        const uint8_t* hb = hdr_row + sb * 32;
        const float d = f16_bits_to_f32(*reinterpret_cast<const uint16_t*>(hb));
        r->d0 = d * (float)(reinterpret_cast<const int8_t*>(hb)[4 + s16]);
    } else if constexpr (W == DT::DQ8) {
        r->d0 = f16_bits_to_f32(
            reinterpret_cast<const uint16_t*>(hdr_row)[c >> 1]);
    }
}

// Decode one 16-B chunk of a row into `w[]` weights starting at column
// `k0` (and for DQ4K a second run of 16 at k0+32).
template <DT W>
__device__ __forceinline__ void decode_chunk_raw(
    const ChunkRaw<W>& r, int c, float* __restrict__ w, int* k0) {
    uint4 qv;
    qv.x = r.qv.x; qv.y = r.qv.y; qv.z = r.qv.z; qv.w = r.qv.w;
    const uint32_t dw[4] = {qv.x, qv.y, qv.z, qv.w};
    if constexpr (W == DT::DQ4K) {
        const int sb = c >> 3, p = c & 7, q = p >> 1, h = p & 1;
        const float d = f16_bits_to_f32(r.hd.x & 0xFFFF);
        const float dmin = f16_bits_to_f32(r.hd.x >> 16);
        const float dl = d * (float)(r.hd.y & 0xFF);
        const float ml = dmin * (float)((r.hd.y >> 8) & 0xFF);
        const float dh = d * (float)((r.hd.y >> 16) & 0xFF);
        const float mh = dmin * (float)(r.hd.y >> 24);
        *k0 = sb * 256 + q * 64 + h * 16;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            const uint32_t lo = dw[j] & 0x0F0F0F0Fu;
            const uint32_t hi = (dw[j] >> 4) & 0x0F0F0F0Fu;
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                w[j * 4 + t] = dl * (float)((lo >> (8 * t)) & 0xFF) - ml;
                w[16 + j * 4 + t] = dh * (float)((hi >> (8 * t)) & 0xFF) - mh;
            }
        }
    } else if constexpr (W == DT::DQ6K) {
        const float sc = r.d0;
        *k0 = c * 16;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                const int8_t q8 = (int8_t)((dw[j] >> (8 * t)) & 0xFF);
                w[j * 4 + t] = sc * (float)q8;
            }
        }
    } else if constexpr (W == DT::DQ8) {
        const float d = r.d0;
        *k0 = c * 16;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                const int8_t q8 = (int8_t)((dw[j] >> (8 * t)) & 0xFF);
                w[j * 4 + t] = d * (float)q8;
            }
        }
    } else if constexpr (W == DT::BF16) {
        *k0 = c * 8;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            w[j * 2] = bf16_bits_to_f32(dw[j] & 0xFFFF);
            w[j * 2 + 1] = bf16_bits_to_f32(dw[j] >> 16);
        }
    } else if constexpr (W == DT::F16) {
        *k0 = c * 8;
        #pragma unroll
        for (int j = 0; j < 4; j++) {
            w[j * 2] = f16_bits_to_f32(dw[j] & 0xFFFF);
            w[j * 2 + 1] = f16_bits_to_f32(dw[j] >> 16);
        }
    } else {  // F32
        *k0 = c * 4;
        #pragma unroll
        for (int j = 0; j < 4; j++)
            w[j] = __uint_as_float(dw[j]);
    }
}

// ------------------------------------------------------------------ GEMV

// y[b][r] = sum_k W[r][k] * x[b][k] (+ res[b][r])
// PRE_RMS:  x = rmsnorm(xin) * gw   (gamma)
// PRE_SILU: x[k] = silu(xin[b][k]) * xin[b][K+k]   (xin is [B][2K])
// One wave per row; x staged in LDS as f32 with per-16 partial sums.
template <DT W, int P, int RPW, int NT>
__global__ __launch_bounds__(NT) void k_gemv(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ xin, const float* __restrict__ gw,
    const float* __restrict__ res, float* __restrict__ y,
    int N, int K, int B, int ldy, float eps) {
    extern __shared__ __attribute__((aligned(16))) char smem[];
    float* x_lds = reinterpret_cast<float*>(smem);     // [B][xpad_size(K)]
    const int KP = xpad_size(K);
    float* red = x_lds + (size_t)B * KP;               // [8]

    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    constexpr int WAVES = NT / 64;
    // RPW rows per wave; NT=512 for big-N GEMVs (8 waves share one staged
    // x copy: +33% waves/CU and half the staging traffic).
    const int rbase = (int)blockIdx.x * (WAVES * RPW) + wave * RPW;
    int r[RPW];
    #pragma unroll
    for (int i = 0; i < RPW; i++) r[i] = rbase + i;
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);
    const int n_chunks = (int)(qs_rb / 16);
    constexpr int WPC = ChunkTraits<W>::W_PER_CHUNK;

    // issue the first weight chunks BEFORE staging: they have no
    // dependency on x, so their HBM latency hides under the staging phase.
    // Loads are unconditional with clamped indices — a branch around a
    // load makes hipcc drain vmcnt(0) at reconvergence (guide §5 traps),
    // killing the software pipeline.
    ChunkRaw<W> cur[RPW], nxt[RPW];
    int rc[RPW];   // clamped row for addressing; r[] keeps validity
    #pragma unroll
    for (int i = 0; i < RPW; i++)
        rc[i] = r[i] < N ? r[i] : N - 1;
    const int c0 = lane < n_chunks ? lane : 0;
    const int c1 = (lane + 64 < n_chunks) ? lane + 64 : c0;
    #pragma unroll
    for (int i = 0; i < RPW; i++) {
        load_chunk<W>(qs + (int64_t)rc[i] * qs_rb,
                      hdr + (int64_t)rc[i] * hdr_rb, c0, &cur[i]);
        load_chunk<W>(qs + (int64_t)rc[i] * qs_rb,
                      hdr + (int64_t)rc[i] * hdr_rb, c1, &nxt[i]);
    }

    // ---- stage activations (vectorized float4; K always %4==0) ----
    // Two-phase batches: issue up to 4 independent global loads, then the
    // LDS writes — hipcc otherwise emits load->vmcnt(0)->ds_write per
    // iteration, serializing the staging into K/1024 memory round trips.
    const int K4 = K >> 2;
    for (int b = 0; b < B; b++) {
        float ss = 0.f;
        float* xlb = x_lds + (size_t)b * KP;
        if constexpr (P == PRE_SILU) {
            const float4* g4 = reinterpret_cast<const float4*>(
                xin + (size_t)b * 2 * K);
            const float4* u4 = reinterpret_cast<const float4*>(
                xin + (size_t)b * 2 * K + K);
            int k0 = tid;
            for (; k0 + 3 * NT < K4; k0 += 4 * NT) {  // guard hoisted: no
                float4 gs[4], us[4];             // per-element branches
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    gs[j] = g4[k0 + j * NT];
                    us[j] = u4[k0 + j * NT];
                }
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    const float4 g = gs[j], u = us[j];
                    float4 o;
                    o.x = (g.x / (1.f + __expf(-g.x))) * u.x;
                    o.y = (g.y / (1.f + __expf(-g.y))) * u.y;
                    o.z = (g.z / (1.f + __expf(-g.z))) * u.z;
                    o.w = (g.w / (1.f + __expf(-g.w))) * u.w;
                    *reinterpret_cast<float4*>(xlb + xpad(k0 * 4 + j * NT * 4)) = o;
                }
            }
            for (int k = k0; k < K4; k += NT) {
                const float4 g = g4[k], u = u4[k];
                float4 o;
                o.x = (g.x / (1.f + __expf(-g.x))) * u.x;
                o.y = (g.y / (1.f + __expf(-g.y))) * u.y;
                o.z = (g.z / (1.f + __expf(-g.z))) * u.z;
                o.w = (g.w / (1.f + __expf(-g.w))) * u.w;
                *reinterpret_cast<float4*>(xlb + xpad(k * 4)) = o;
            }
        } else {
            const float4* x4 = reinterpret_cast<const float4*>(
                xin + (size_t)b * K);
            int k0 = tid;
            for (; k0 + 3 * NT < K4; k0 += 4 * NT) {  // guard hoisted
                float4 vs[4];
                #pragma unroll
                for (int j = 0; j < 4; j++) vs[j] = x4[k0 + j * NT];
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    const float4 v = vs[j];
                    *reinterpret_cast<float4*>(xlb + xpad(k0 * 4 + j * NT * 4)) = v;
                    if constexpr (P == PRE_RMS)
                        ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
                }
            }
            for (int k = k0; k < K4; k += NT) {
                const float4 v = x4[k];
                *reinterpret_cast<float4*>(xlb + xpad(k * 4)) = v;
                if constexpr (P == PRE_RMS)
                    ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
            }
        }
        if constexpr (P == PRE_RMS) {
            // wave-level reduce + one cross-wave pass (2 barriers total)
            const float ws = wave_reduce_sum(ss);
            if (lane == 0) red[wave] = ws;
            __syncthreads();
            float rsum = 0.f;
            #pragma unroll
            for (int wv = 0; wv < WAVES; wv++) rsum += red[wv];
            const float inv = rsqrtf(rsum / (float)K + eps);
            const float4* gw4 = reinterpret_cast<const float4*>(gw);
            int kg = tid;
            for (; kg + 768 < K4; kg += 1024) {  // batched gw loads
                float4 gs[4];
                #pragma unroll
                for (int j = 0; j < 4; j++) gs[j] = gw4[kg + j * NT];
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    float4* vp = reinterpret_cast<float4*>(
                        xlb + xpad(kg * 4 + j * NT * 4));
                    float4 v = *vp;
                    v.x *= inv * gs[j].x; v.y *= inv * gs[j].y;
                    v.z *= inv * gs[j].z; v.w *= inv * gs[j].w;
                    *vp = v;
                }
            }
            for (int k = kg; k < K4; k += NT) {
                const float4 g = gw4[k];
                float4* vp = reinterpret_cast<float4*>(xlb + xpad(k * 4));
                float4 v = *vp;
                v.x *= inv * g.x; v.y *= inv * g.y;
                v.z *= inv * g.z; v.w *= inv * g.w;
                *vp = v;
            }
        }
    }
    __syncthreads();

    // ---- per-wave sweep over RPW rows (2-buffer, prefetch distance 1;
    // measured best: deeper rotations and pair-unrolls regress on register
    // pressure — see profiles/ notes) ----
    float acc[RPW][2];  // [row][b], B <= 2
    #pragma unroll
    for (int i = 0; i < RPW; i++) { acc[i][0] = 0.f; acc[i][1] = 0.f; }
    for (int c = lane; c < n_chunks; c += 64) {
        #pragma unroll
        for (int i = 0; i < RPW; i++) {
            float w[WPC];
            int k0;
            decode_chunk_raw<W>(cur[i], c, w, &k0);
            // refill the just-consumed buffer with chunk c+128, then swap:
            // both first chunks were issued before staging, so the sweep
            // never sees a cold load.
            const int cn = (c + 128 < n_chunks) ? c + 128 : c0;
            load_chunk<W>(qs + (int64_t)rc[i] * qs_rb,
                          hdr + (int64_t)rc[i] * hdr_rb, cn, &cur[i]);
            #pragma unroll 2
            for (int b = 0; b < B; b++) {
                const float4* xb4 = reinterpret_cast<const float4*>(
                    x_lds + (size_t)b * KP + xpad(k0));
                // 4 independent partials: a single serial accumulator is a
                // 32-deep dependent v_fma chain (~4 cyc each) — PMC showed
                // 34% of cycles in SQ_WAIT_INST_ANY issue stalls.
                float p0 = 0.f, p1 = 0.f, p2 = 0.f, p3 = 0.f;
                if constexpr (W == DT::DQ4K) {
                    #pragma unroll
                    for (int t4 = 0; t4 < 4; t4++) {
                        const float4 xl = xb4[t4];
                        const float4 xh = xb4[8 + t4];
                        p0 += w[t4 * 4 + 0] * xl.x + w[t4 * 4 + 1] * xl.y;
                        p1 += w[t4 * 4 + 2] * xl.z + w[t4 * 4 + 3] * xl.w;
                        p2 += w[16 + t4 * 4 + 0] * xh.x
                            + w[16 + t4 * 4 + 1] * xh.y;
                        p3 += w[16 + t4 * 4 + 2] * xh.z
                            + w[16 + t4 * 4 + 3] * xh.w;
                    }
                } else {
                    #pragma unroll
                    for (int t4 = 0; t4 < WPC / 4; t4++) {
                        const float4 xv = xb4[t4];
                        p0 += w[t4 * 4 + 0] * xv.x;
                        p1 += w[t4 * 4 + 1] * xv.y;
                        p2 += w[t4 * 4 + 2] * xv.z;
                        p3 += w[t4 * 4 + 3] * xv.w;
                    }
                }
                acc[i][b] += (p0 + p1) + (p2 + p3);
            }
            ChunkRaw<W> t = cur[i];
            cur[i] = nxt[i];
            nxt[i] = t;
        }
    }
    #pragma unroll
    for (int i = 0; i < RPW; i++) {
        if (r[i] >= N) continue;
        #pragma unroll 2
        for (int b = 0; b < B; b++) {
            float v = wave_reduce_sum(acc[i][b]);
            if (lane == 0) {
                const int64_t idx = (int64_t)b * ldy + r[i];
                y[idx] = v + (res ? res[idx] : 0.f);
            }
        }
    }
}

// Q8-activation GEMV for quantized weights: activations are quantized to
// int8 per 32-block during staging (symmetric, xd = amax/127, rint), and
// the dot product runs on v_dot4_i32_i8 — 4 weights per instruction and a
// quarter of the LDS read traffic vs the f32 path. PMC showed the f32
// path saturating the VALU issue port (~30% active x 6 waves/SIMD).
// Semantics are exactly y = W . (rint(x/xd)*xd), replicated by the numpy
// reference (ref_numpy act_q8) and the kernel tests.
__device__ __forceinline__ int xpad8(int k) {  // byte index pad per 256B
    return k + ((k >> 8) << 4);
}
__device__ __forceinline__ constexpr int xpad8_size(int k) {
    return k + (k >> 4);
}

template <DT W, int P>
__global__ __launch_bounds__(256) void k_gemv_q8(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ xin, const float* __restrict__ gw,
    const float* __restrict__ res, float* __restrict__ y,
    int N, int K, int B, int ldy, float eps) {
    static_assert(W == DT::DQ4K || W == DT::DQ6K || W == DT::DQ8,
                  "q8 path covers quantized weights only");
    extern __shared__ __attribute__((aligned(16))) char smem[];
    // per b: x8 [xpad8_size(K)] int8, xd [K/32] f32, s16f [K/16] f32
    const int X8B = (xpad8_size(K) + 15) & ~15;
    const int NB32 = K / 32, NB16 = K / 16;
    int8_t* x8 = reinterpret_cast<int8_t*>(smem);               // [B][X8B]
    float* xd = reinterpret_cast<float*>(smem + (size_t)B * X8B);  // [B][NB32]
    float* s16f = xd + (size_t)B * NB32;                        // [B][NB16]
    float* red = s16f + (size_t)B * NB16;                       // [8]

    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int r = (int)blockIdx.x * 4 + wave;
    const int rc = r < N ? r : N - 1;
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);
    const int n_chunks = (int)(qs_rb / 16);

    // issue first weight chunks before staging (as in the f32 path)
    ChunkRaw<W> cur, nxt;
    const int c0 = lane < n_chunks ? lane : 0;
    const int c1 = (lane + 64 < n_chunks) ? lane + 64 : c0;
    load_chunk<W>(qs + (int64_t)rc * qs_rb, hdr + (int64_t)rc * hdr_rb, c0,
                  &cur);
    load_chunk<W>(qs + (int64_t)rc * qs_rb, hdr + (int64_t)rc * hdr_rb, c1,
                  &nxt);

    // ---- stage + quantize activations ----
    for (int b = 0; b < B; b++) {
        float inv = 1.f;
        if constexpr (P == PRE_RMS) {
            const float4* x4 = reinterpret_cast<const float4*>(
                xin + (size_t)b * K);
            float ss = 0.f;
            for (int k = tid; k < K >> 2; k += 256) {
                const float4 v = x4[k];
                ss += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
            }
            const float ws = wave_reduce_sum(ss);
            if (lane == 0) red[wave] = ws;
            __syncthreads();
            inv = rsqrtf((red[0] + red[1] + red[2] + red[3]) / (float)K + eps);
        }
        // lane-parallel block quantization: 8 lanes per 32-block, one
        // float4 each, amax/sum via 8-lane shuffles. The round-1 per-thread
        // version serialized ~120 dependent ops per block on at most half
        // the threads (PERF_NOTES lever 2).
        const int jl = tid & 7;   // lane within the block's 8-lane group
        for (int blk = tid >> 3; blk < NB32; blk += 32) {
            float4 v;
            if constexpr (P == PRE_SILU) {
                const float4 g = reinterpret_cast<const float4*>(
                    xin + (size_t)b * 2 * K + blk * 32)[jl];
                const float4 u = reinterpret_cast<const float4*>(
                    xin + (size_t)b * 2 * K + K + blk * 32)[jl];
                v.x = (g.x / (1.f + __expf(-g.x))) * u.x;
                v.y = (g.y / (1.f + __expf(-g.y))) * u.y;
                v.z = (g.z / (1.f + __expf(-g.z))) * u.z;
                v.w = (g.w / (1.f + __expf(-g.w))) * u.w;
            } else {
                v = reinterpret_cast<const float4*>(
                    xin + (size_t)b * K + blk * 32)[jl];
                if constexpr (P == PRE_RMS) {
                    const float4 g = reinterpret_cast<const float4*>(
                        gw + blk * 32)[jl];
                    v.x *= inv * g.x; v.y *= inv * g.y;
                    v.z *= inv * g.z; v.w *= inv * g.w;
                }
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
            s += __shfl_xor(s, 1, 64);
            s += __shfl_xor(s, 2, 64);  // lanes {0..3}/{4..7} hold 16-sums
            *reinterpret_cast<uint32_t*>(
                x8 + (size_t)b * X8B + xpad8(blk * 32) + jl * 4) = packed;
            if (jl == 0) {
                xd[(size_t)b * NB32 + blk] = scale;
                s16f[(size_t)b * NB16 + blk * 2] = scale * (float)s;
            } else if (jl == 4) {
                s16f[(size_t)b * NB16 + blk * 2 + 1] = scale * (float)s;
            }
        }
        if constexpr (P == PRE_RMS) __syncthreads();  // red[] reuse next b
    }
    __syncthreads();

    // ---- per-wave row sweep (dot4 on int8) ----
    float acc[2] = {0.f, 0.f};
    for (int c = lane; c < n_chunks; c += 64) {
        uint32_t dwv[4] = {cur.qv.x, cur.qv.y, cur.qv.z, cur.qv.w};
        if constexpr (W == DT::DQ4K) {
            const int sb = c >> 3, p = c & 7, q = p >> 1, h = p & 1;
            const float d = f16_bits_to_f32(cur.hd.x & 0xFFFF);
            const float dmin = f16_bits_to_f32(cur.hd.x >> 16);
            const float dl = d * (float)(cur.hd.y & 0xFF);
            const float ml = dmin * (float)((cur.hd.y >> 8) & 0xFF);
            const float dh = d * (float)((cur.hd.y >> 16) & 0xFF);
            const float mh = dmin * (float)(cur.hd.y >> 24);
            const int k0 = sb * 256 + q * 64 + h * 16;
            const int cn = (c + 128 < n_chunks) ? c + 128 : c0;
            load_chunk<W>(qs + (int64_t)rc * qs_rb,
                          hdr + (int64_t)rc * hdr_rb, cn, &cur);
            #pragma unroll 2
            for (int b = 0; b < B; b++) {
                const int8_t* x8b = x8 + (size_t)b * X8B;
                const uint4 xl = *reinterpret_cast<const uint4*>(
                    x8b + xpad8(k0));
                const uint4 xh = *reinterpret_cast<const uint4*>(
                    x8b + xpad8(k0 + 32));
                int s1l = 0, s1h = 0;
                s1l = __builtin_amdgcn_sdot4((int)(dwv[0] & 0x0F0F0F0Fu), (int)xl.x, s1l, false);
                s1l = __builtin_amdgcn_sdot4((int)(dwv[1] & 0x0F0F0F0Fu), (int)xl.y, s1l, false);
                s1l = __builtin_amdgcn_sdot4((int)(dwv[2] & 0x0F0F0F0Fu), (int)xl.z, s1l, false);
                s1l = __builtin_amdgcn_sdot4((int)(dwv[3] & 0x0F0F0F0Fu), (int)xl.w, s1l, false);
                s1h = __builtin_amdgcn_sdot4((int)((dwv[0] >> 4) & 0x0F0F0F0Fu), (int)xh.x, s1h, false);
                s1h = __builtin_amdgcn_sdot4((int)((dwv[1] >> 4) & 0x0F0F0F0Fu), (int)xh.y, s1h, false);
                s1h = __builtin_amdgcn_sdot4((int)((dwv[2] >> 4) & 0x0F0F0F0Fu), (int)xh.z, s1h, false);
                s1h = __builtin_amdgcn_sdot4((int)((dwv[3] >> 4) & 0x0F0F0F0Fu), (int)xh.w, s1h, false);
                const float xdl = xd[(size_t)b * NB32 + (k0 >> 5)];
                const float xdh = xd[(size_t)b * NB32 + ((k0 + 32) >> 5)];
                acc[b] += dl * xdl * (float)s1l - ml * s16f[(size_t)b * NB16 + (k0 >> 4)]
                        + dh * xdh * (float)s1h - mh * s16f[(size_t)b * NB16 + ((k0 + 32) >> 4)];
            }
        } else {
            // DQ6K / DQ8: 16 signed weights per chunk, one scale
            const float sc = cur.d0;
            const int k0 = c * 16;
            const int cn = (c + 128 < n_chunks) ? c + 128 : c0;
            load_chunk<W>(qs + (int64_t)rc * qs_rb,
                          hdr + (int64_t)rc * hdr_rb, cn, &cur);
            #pragma unroll 2
            for (int b = 0; b < B; b++) {
                const int8_t* x8b = x8 + (size_t)b * X8B;
                const uint4 xv = *reinterpret_cast<const uint4*>(
                    x8b + xpad8(k0));
                int s1 = 0;
                s1 = __builtin_amdgcn_sdot4((int)dwv[0], (int)xv.x, s1, false);
                s1 = __builtin_amdgcn_sdot4((int)dwv[1], (int)xv.y, s1, false);
                s1 = __builtin_amdgcn_sdot4((int)dwv[2], (int)xv.z, s1, false);
                s1 = __builtin_amdgcn_sdot4((int)dwv[3], (int)xv.w, s1, false);
                acc[b] += sc * xd[(size_t)b * NB32 + (k0 >> 5)] * (float)s1;
            }
        }
        // rotate: cur was refilled with chunk c+128; nxt holds c+64
        ChunkRaw<W> t = cur;
        cur = nxt;
        nxt = t;
    }
    if (r < N) {
        #pragma unroll 2
        for (int b = 0; b < B; b++) {
            float v = wave_reduce_sum(acc[b]);
            if (lane == 0) {
                const int64_t idx = (int64_t)b * ldy + r;
                y[idx] = v + (res ? res[idx] : 0.f);
            }
        }
    }
}

// ------------------------------------------------------ register-x GEMV
//
// Round-2 B=1 decode path (guide row: "GEMV / M<=16 decode weights: load
// straight to VGPRs, deep unroll, late vmcnt"). Key facts this design
// exploits:
//  - For every quant dtype the chunk->k mapping makes lane l's chunk set
//    (c = l*CPL .. l*CPL+CPL-1) cover EXACTLY x[l*64*(SEGF/64) ...): the
//    whole activation stripe fits in 64 (or 32) VGPRs per lane, loaded
//    ONCE per wave — no LDS staging, no barriers in the sweep, and the
//    per-WG re-staging traffic of the LDS kernel (which at B=1 rivals
//    the weight traffic itself: ~16 KB x N/4 rows) is gone.
//  - Each wave streams ROWS: per row, each lane issues CPL consecutive
//    dwordx4 weight loads (the wave covers the stripe's payload as one
//    perfectly-coalesced clause) into an NB-slot software pipeline, so
//    2-3 rows of weight loads are always in flight per lane.
//  - Dequant uses v_cvt_f32_ubyte0..3 (extract+convert in one VALU op)
//    and, for Q4_K, the regroup sum((d*sc*q - dmin*mn)*x) =
//    d*sc*sum(q*x) - dmin*mn*sum(x) with sum(x) per 16-run precomputed
//    once per lane — ~2.3 VALU/weight, under the HBM-bound time.
//  - Selected for single-stripe shapes (K = 2048 or 4096) only. A
//    gridDim.z-striped atomicAdd variant for K > 4096 was built and
//    measured SLOWER than the legacy LDS kernel on the down projection
//    (~33 vs 23 us: x re-staging there is L2-served and the stripe
//    launches don't amortize), so it was removed; K > 4096 uses
//    k_gemv_rl (K = 14336) or the legacy kernel.
// Reference parity: same mat-vec the reference delegates to llama.cpp
// (SURVEY.md §2.3); numerics = plain f32 dot of dequantized weights.

template <DT W, int P, int SEGF, int BB = 1>
__global__ __launch_bounds__(256) void k_gemv_r(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ xin, const float* __restrict__ gw,
    const float* __restrict__ res, float* __restrict__ y,
    int N, int K, int kbeg0, int ldy, float eps) {
    static_assert(W == DT::DQ4K || W == DT::DQ6K || W == DT::DQ8,
                  "register-x GEMV covers quantized weights only");
    static_assert(BB == 1 || BB == 2, "register-x GEMV: 1 or 2 rows");
    constexpr int CPL = (W == DT::DQ4K) ? SEGF / 32 : SEGF / 16;
    // pipeline depth: NB-1 rows of weight loads in flight per lane. The
    // binding constraint at 8 waves/CU is outstanding bytes (~2 us
    // effective latency wants ~4 rows in flight), so go as deep as the
    // register file allows: 5 slots for DQ4K (11 VGPRs/slot) — except
    // PRE_RMS, where the extra live state tipped occupancy and measured
    // slower (qkv 8.4 -> 10.7 us) — and 3 for DQ8/DQ6K (19 VGPRs/slot;
    // the Q6_K head hit 6.0 TB/s = 93% of the streaming ceiling there).
    // BB=2 (both decode rows share one weight stream) doubles the x
    // registers, so back the depth off a notch.
    constexpr int NB =
        BB == 2 ? ((W == DT::DQ4K) ? 3 : 2)
                : ((W == DT::DQ4K) ? (P == PRE_RMS ? 3 : 5) : 3);
    static_assert(CPL >= 1 && CPL <= 4, "stripe must be 2048 or 4096");

    const int lane = threadIdx.x & 63;
    const int wid = (int)blockIdx.x * 4 + (threadIdx.x >> 6);
    const int nw = (int)gridDim.x * 4;
    const int kbeg = kbeg0 + (int)blockIdx.z * (SEGF * 64);
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);
    int64_t qso, hdro;   // stripe byte offsets within a row
    if constexpr (W == DT::DQ4K) { qso = kbeg / 2; hdro = (int64_t)(kbeg >> 8) * 32; }
    else if constexpr (W == DT::DQ8) { qso = kbeg; hdro = (int64_t)(kbeg >> 5) * 2; }
    else { qso = kbeg; hdro = (int64_t)(kbeg >> 8) * 32; }
    const uint8_t* qsb = qs + qso;
    const uint8_t* hdb = hdr + hdro;
    constexpr int KL = SEGF * 64;            // stripe width in weights

    // ---- cooperative, COALESCED x phase ----
    // The per-lane chunk->k mapping is a 256 B lane-stride gather: loading
    // x straight into the mapped registers issues 64 single-line requests
    // per instruction (measured as a ~10-15 us per-wave ramp that only
    // head-sized rows/wave amortized). Instead the WG loads the stripe
    // lane-LINEARLY (perfectly coalesced), applies SILU / gw and the
    // block-wide RMS reduce in that linear pass, bounces through LDS, and
    // each lane then gathers its mapped segment with ds_reads.
    __shared__ float4 xs4[BB * KL / 4];
    __shared__ float redw[BB * 4];
    {
        const int tid = threadIdx.x;
        #pragma unroll
        for (int b = 0; b < BB; b++) {
            const float* xrow = xin + (int64_t)b * (P == PRE_SILU ? 2 : 1) * K;
            float ssp = 0.f;
            #pragma unroll
            for (int i = 0; i < KL / 4 / 256; i++) {
                const int k4 = i * 256 + tid;      // float4 index in stripe
                const int k = kbeg + k4 * 4;
                float4 v;
                if constexpr (P == PRE_SILU) {
                    const float4 g = *reinterpret_cast<const float4*>(xrow + k);
                    const float4 u =
                        *reinterpret_cast<const float4*>(xrow + K + k);
                    v.x = (g.x / (1.f + __expf(-g.x))) * u.x;
                    v.y = (g.y / (1.f + __expf(-g.y))) * u.y;
                    v.z = (g.z / (1.f + __expf(-g.z))) * u.z;
                    v.w = (g.w / (1.f + __expf(-g.w))) * u.w;
                } else {
                    v = *reinterpret_cast<const float4*>(xrow + k);
                }
                if constexpr (P == PRE_RMS) {
                    ssp += v.x * v.x + v.y * v.y + v.z * v.z + v.w * v.w;
                    const float4 g = *reinterpret_cast<const float4*>(gw + k);
                    v.x *= g.x; v.y *= g.y; v.z *= g.z; v.w *= g.w;
                }
                xs4[b * (KL / 4) + k4] = v;
            }
            if constexpr (P == PRE_RMS) {
                ssp = wave_reduce_sum(ssp);
                if ((tid & 63) == 0) redw[b * 4 + (tid >> 6)] = ssp;
            }
        }
        __syncthreads();
    }
    float inv[BB];
    #pragma unroll
    for (int b = 0; b < BB; b++) {
        inv[b] = 1.f;
        if constexpr (P == PRE_RMS)   // single-stripe only: K == KL
            inv[b] = rsqrtf((redw[b * 4] + redw[b * 4 + 1] + redw[b * 4 + 2] +
                             redw[b * 4 + 3]) / (float)K + eps);
    }

    // gather this lane's mapped segment out of LDS (xr holds x*gw*inv)
    float4 xr[BB * SEGF / 4];
    #pragma unroll
    for (int b = 0; b < BB; b++) {
        constexpr int XB = SEGF / 4;
        #pragma unroll
        for (int j = 0; j < CPL; j++) {
            const int c = lane * CPL + j;
            const int sb4 = b * (KL / 4);
            if constexpr (W == DT::DQ4K) {
                const int sb = c >> 3, p = c & 7, q = p >> 1, h = p & 1;
                const int k0 = sb * 256 + q * 64 + h * 16;   // stripe-local
                #pragma unroll
                for (int t = 0; t < 4; t++) {
                    xr[b * XB + j * 8 + t] = xs4[sb4 + (k0 >> 2) + t];
                    xr[b * XB + j * 8 + 4 + t] =
                        xs4[sb4 + ((k0 + 32) >> 2) + t];
                }
            } else {
                const int k0 = c * 16;
                #pragma unroll
                for (int t = 0; t < 4; t++)
                    xr[b * XB + j * 4 + t] = xs4[sb4 + (k0 >> 2) + t];
            }
        }
        if constexpr (P == PRE_RMS) {
            #pragma unroll
            for (int t = 0; t < XB; t++) {
                xr[b * XB + t].x *= inv[b]; xr[b * XB + t].y *= inv[b];
                xr[b * XB + t].z *= inv[b]; xr[b * XB + t].w *= inv[b];
            }
        }
    }
    // Q4_K: per-16-run x sums for the d*sc*sum(qx) - dmin*mn*sum(x) regroup
    float sxl[BB * CPL], sxh[BB * CPL];
    #pragma unroll
    for (int b = 0; b < BB; b++) {
        constexpr int XB = SEGF / 4;
        #pragma unroll
        for (int j = 0; j < CPL; j++) {
            if constexpr (W == DT::DQ4K) {
                float a = 0.f, bs = 0.f;
                #pragma unroll
                for (int t = 0; t < 4; t++) {
                    const float4 l = xr[b * XB + j * 8 + t];
                    const float4 h = xr[b * XB + j * 8 + 4 + t];
                    a += l.x + l.y + l.z + l.w;
                    bs += h.x + h.y + h.z + h.w;
                }
                sxl[b * CPL + j] = a; sxh[b * CPL + j] = bs;
            } else {
                sxl[b * CPL + j] = 0.f; sxh[b * CPL + j] = 0.f;
            }
        }
    }

    // rows this wave owns: the CONTIGUOUS block [wid*rpw, wid*rpw+rows_my).
    // Contiguous (not nw-strided) so one wave's row stream walks adjacent
    // DRAM pages, and sized >=4 rows by the launcher so the one-time x
    // register-load phase amortizes (both measured: strided single-row
    // waves ran the o-projection at 740 GB/s).
    const int rpw = (N + nw - 1) / nw;
    const int r0 = wid * rpw;
    const int rows_my = (N > r0) ? ((N - r0) < rpw ? (N - r0) : rpw) : 0;

    u32x4 qv[NB][CPL];
    uint2 hd4[NB];      // DQ4K pair header
    uint32_t hs[NB];    // DQ8 packed f16 d's / DQ6K packed sc bytes
    uint32_t hdd[NB];   // DQ6K f16 d bits
    uint32_t rv[NB * BB];  // res[r] bits, prefetched with the row's weights:
                        // a res load at the store join costs a vmcnt(0)
                        // drain of the whole pipeline EVERY row (seen in
                        // disassembly) even when res == nullptr. When res
                        // is null we load y[r] (valid memory) and zero it
                        // with a bit mask — bitwise, so garbage bits can
                        // never produce a NaN, and no select on the load
                        // path that SimplifyCFG would turn back into a
                        // branch + sunk load.
    const float* resl = res ? res : y;
    const uint32_t rmask = res ? 0xFFFFFFFFu : 0u;

    auto stage = [&](int slot, int i) {
        const int ic = i < rows_my - 1 ? i : rows_my - 1;   // clamp: L2 hit
        const int r = r0 + ic;
        const uint8_t* qrow = qsb + (int64_t)r * qs_rb;
        const uint8_t* hrow = hdb + (int64_t)r * hdr_rb;
        #pragma unroll
        for (int b = 0; b < BB; b++)
            rv[slot * BB + b] = __float_as_uint(resl[(int64_t)b * ldy + r]);
        #pragma unroll
        for (int j = 0; j < CPL; j++)
            qv[slot][j] = *(
                reinterpret_cast<const u32x4*>(qrow) + lane * CPL + j);
        if constexpr (W == DT::DQ4K) {
            hd4[slot] = *reinterpret_cast<const uint2*>(
                hrow + ((lane * CPL) >> 1) * 8);
        } else if constexpr (W == DT::DQ8) {
            if constexpr (CPL == 4)
                hs[slot] = *reinterpret_cast<const uint32_t*>(hrow + lane * 4);
            else
                hs[slot] = *reinterpret_cast<const uint16_t*>(hrow + lane * 2);
        } else {   // DQ6K: 32 B/sb {f16 d; i8 sc[16]}
            const int sb = (lane * CPL) >> 4, s0 = (lane * CPL) & 15;
            const uint8_t* hb = hrow + (int64_t)sb * 32;
            hdd[slot] = *reinterpret_cast<const uint16_t*>(hb);
            if constexpr (CPL == 4)
                hs[slot] = *reinterpret_cast<const uint32_t*>(hb + 4 + s0);
            else
                hs[slot] = *reinterpret_cast<const uint16_t*>(hb + 4 + s0);
        }
    };
    // (float)((dw >> 8n) & 0xff) matches LLVM's CVT_F32_UBYTEn combine:
    // one v_cvt_f32_ubyte{0..3} per weight, no separate extract.
    auto dot16 = [&](uint32_t lo, const float4& x) -> float {
        return (float)(lo & 0xFF) * x.x
             + (float)((lo >> 8) & 0xFF) * x.y
             + (float)((lo >> 16) & 0xFF) * x.z
             + (float)(lo >> 24) * x.w;
    };
    auto sdot16 = [&](uint32_t dw, const float4& x) -> float {
        return (float)(int)(int8_t)(dw & 0xFF) * x.x
             + (float)(int)(int8_t)((dw >> 8) & 0xFF) * x.y
             + (float)(int)(int8_t)((dw >> 16) & 0xFF) * x.z
             + (float)(int)(int8_t)(dw >> 24) * x.w;
    };
    auto compute = [&](int slot, int i) {
        constexpr int XB = SEGF / 4;
        float acc[BB];
        #pragma unroll
        for (int b = 0; b < BB; b++) acc[b] = 0.f;
        if constexpr (W == DT::DQ4K) {
            const uint2 hd = hd4[slot];
            const float d = f16_bits_to_f32(hd.x & 0xFFFF);
            const float dmin = f16_bits_to_f32(hd.x >> 16);
            const float dl = d * (float)(hd.y & 0xFF);
            const float ml = dmin * (float)((hd.y >> 8) & 0xFF);
            const float dh = d * (float)((hd.y >> 16) & 0xFF);
            const float mh = dmin * (float)(hd.y >> 24);
            #pragma unroll
            for (int j = 0; j < CPL; j++) {
                #pragma unroll
                for (int b = 0; b < BB; b++) {
                    float ql = 0.f, qh = 0.f;
                    #pragma unroll
                    for (int t = 0; t < 4; t++) {
                        const uint32_t dw = qv[slot][j][t];
                        ql += dot16(dw & 0x0F0F0F0Fu, xr[b * XB + j * 8 + t]);
                        qh += dot16((dw >> 4) & 0x0F0F0F0Fu,
                                    xr[b * XB + j * 8 + 4 + t]);
                    }
                    acc[b] += dl * ql - ml * sxl[b * CPL + j]
                            + dh * qh - mh * sxh[b * CPL + j];
                }
            }
        } else if constexpr (W == DT::DQ8) {
            #pragma unroll
            for (int j = 0; j < CPL; j++) {
                const float dj = f16_bits_to_f32(
                    (hs[slot] >> (CPL == 4 ? (j >> 1) * 16 : 0)) & 0xFFFF);
                #pragma unroll
                for (int b = 0; b < BB; b++) {
                    float q = 0.f;
                    #pragma unroll
                    for (int t = 0; t < 4; t++)
                        q += sdot16(qv[slot][j][t], xr[b * XB + j * 4 + t]);
                    acc[b] += dj * q;
                }
            }
        } else {   // DQ6K
            const float d = f16_bits_to_f32(hdd[slot]);
            #pragma unroll
            for (int j = 0; j < CPL; j++) {
                const float dj =
                    d * (float)(int)(int8_t)((hs[slot] >> (8 * j)) & 0xFF);
                #pragma unroll
                for (int b = 0; b < BB; b++) {
                    float q = 0.f;
                    #pragma unroll
                    for (int t = 0; t < 4; t++)
                        q += sdot16(qv[slot][j][t], xr[b * XB + j * 4 + t]);
                    acc[b] += dj * q;
                }
            }
        }
        #pragma unroll
        for (int b = 0; b < BB; b++) {
            const float v = wave_reduce_sum(acc[b]);
            if (i < rows_my && lane == 0) {
                const int r = r0 + i;
                y[(int64_t)b * ldy + r] =
                    v + __uint_as_float(rv[slot * BB + b] & rmask);
            }
        }
    };

    if (rows_my > 0) {
        #pragma unroll
        for (int s = 0; s < NB - 1; s++) stage(s, s);
        for (int i0 = 0; i0 < rows_my; i0 += NB) {
            #pragma unroll
            for (int j = 0; j < NB; j++) {
                stage((j + NB - 1) % NB, i0 + j + NB - 1);
                compute(j, i0 + j);
            }
        }
    }
}


// LDS-x row-streaming GEMV for long-K quant mats (the down projection,
// K = 14336): same deep row pipeline and coalesced weight clauses as
// k_gemv_r, but the stripe's activations live in LDS (57 KB/WG -> 2
// WGs/CU) instead of per-lane registers, chunks are assigned lane-
// STRIDED (c = j*64 + lane: each weight instruction reads 1 KB
// contiguous, and every header load is one uint2/u16 per j for any
// chunks-per-lane), and SILU folds into the one coalesced x pass
// (replacing a separate k_silu_mul launch + the legacy kernel's
// per-WG re-evaluation of 14M exps).
template <DT W, int P, int KLT>   // KLT: K in units of 2048
__global__ __launch_bounds__(256) void k_gemv_rl(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ xin, const float* __restrict__ res,
    float* __restrict__ y, int N, int K) {
    static_assert(P != PRE_RMS, "long-K path: norm-free projections only");
    constexpr int KL = KLT * 2048;
    constexpr int CPL = (W == DT::DQ4K) ? KLT : 2 * KLT;
    constexpr int NB = 2;
    const int lane = threadIdx.x & 63;
    const int wid = (int)blockIdx.x * 4 + (threadIdx.x >> 6);
    const int nw = (int)gridDim.x * 4;
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);

    __shared__ float4 xs4[KL / 4];
    {
        const int tid = threadIdx.x;
        #pragma unroll
        for (int i = 0; i < KL / 4 / 256; i++) {
            const int k4 = i * 256 + tid;
            const int k = k4 * 4;
            float4 v;
            if constexpr (P == PRE_SILU) {
                const float4 g = *reinterpret_cast<const float4*>(xin + k);
                const float4 u = *reinterpret_cast<const float4*>(xin + K + k);
                v.x = (g.x / (1.f + __expf(-g.x))) * u.x;
                v.y = (g.y / (1.f + __expf(-g.y))) * u.y;
                v.z = (g.z / (1.f + __expf(-g.z))) * u.z;
                v.w = (g.w / (1.f + __expf(-g.w))) * u.w;
            } else {
                v = *reinterpret_cast<const float4*>(xin + k);
            }
            xs4[k4] = v;
        }
        __syncthreads();
    }

    // per-chunk x bases (stripe-local float4 index) + Q4_K per-run x sums
    float sxl[CPL], sxh[CPL];
    int xb[CPL];
    #pragma unroll
    for (int j = 0; j < CPL; j++) {
        const int c = j * 64 + lane;
        if constexpr (W == DT::DQ4K) {
            const int sb = c >> 3, p = c & 7, q = p >> 1, h = p & 1;
            xb[j] = (sb * 256 + q * 64 + h * 16) >> 2;
            float a = 0.f, b = 0.f;
            #pragma unroll
            for (int t = 0; t < 4; t++) {
                const float4 l = xs4[xb[j] + t], hv = xs4[xb[j] + 8 + t];
                a += l.x + l.y + l.z + l.w;
                b += hv.x + hv.y + hv.z + hv.w;
            }
            sxl[j] = a; sxh[j] = b;
        } else {
            xb[j] = c * 4;
            sxl[j] = 0.f; sxh[j] = 0.f;
        }
    }

    const int rpw = (N + nw - 1) / nw;
    const int r0 = wid * rpw;
    const int rows_my = (N > r0) ? ((N - r0) < rpw ? (N - r0) : rpw) : 0;

    u32x4 qv[NB][CPL];
    uint2 hd4[NB][CPL];        // DQ4K pair per j
    uint32_t hs[NB][CPL];      // DQ8 f16 d / DQ6K sc byte per j
    uint32_t hdd[NB][CPL];     // DQ6K f16 d per j
    uint32_t rv[NB];
    const float* resl = res ? res : y;
    const uint32_t rmask = res ? 0xFFFFFFFFu : 0u;

    auto stage = [&](int slot, int i) {
        const int ic = i < rows_my - 1 ? i : rows_my - 1;
        const int r = r0 + ic;
        const uint8_t* qrow = qs + (int64_t)r * qs_rb;
        const uint8_t* hrow = hdr + (int64_t)r * hdr_rb;
        rv[slot] = __float_as_uint(resl[r]);
        #pragma unroll
        for (int j = 0; j < CPL; j++) {
            const int c = j * 64 + lane;
            qv[slot][j] = *(reinterpret_cast<const u32x4*>(qrow) + c);
            if constexpr (W == DT::DQ4K) {
                hd4[slot][j] = *reinterpret_cast<const uint2*>(
                    hrow + (int64_t)(c >> 1) * 8);
            } else if constexpr (W == DT::DQ8) {
                hs[slot][j] = *reinterpret_cast<const uint16_t*>(
                    hrow + (int64_t)(c >> 1) * 2);
            } else {   // DQ6K
                const int sb = c >> 4;
                const uint8_t* hb = hrow + (int64_t)sb * 32;
                hdd[slot][j] = *reinterpret_cast<const uint16_t*>(hb);
                hs[slot][j] = hb[4 + (lane & 15)];
            }
        }
    };
    auto dot16u = [&](uint32_t lo, const float4& x) -> float {
        return (float)(lo & 0xFF) * x.x + (float)((lo >> 8) & 0xFF) * x.y
             + (float)((lo >> 16) & 0xFF) * x.z + (float)(lo >> 24) * x.w;
    };
    auto dot16s = [&](uint32_t dw, const float4& x) -> float {
        return (float)(int)(int8_t)(dw & 0xFF) * x.x
             + (float)(int)(int8_t)((dw >> 8) & 0xFF) * x.y
             + (float)(int)(int8_t)((dw >> 16) & 0xFF) * x.z
             + (float)(int)(int8_t)(dw >> 24) * x.w;
    };
    auto compute = [&](int slot, int i) {
        float acc = 0.f;
        #pragma unroll
        for (int j = 0; j < CPL; j++) {
            // scheduling fence per chunk: without it the scheduler lifts
            // ALL CPL chunks' LDS x reads to the top of the row compute
            // (224 live floats -> 256 VGPR + AGPR spill, 1 wave/SIMD).
            // Mask 0xF lets ALU/VALU/SALU/MFMA cross (FMA chains of
            // adjacent chunks interleave for ILP) while pinning memory
            // ops — a full sched_barrier(0) costs ~40% in the guide's
            // GEMM ladder by defeating the scheduler outright.
            __builtin_amdgcn_sched_barrier(0x000F);
            if constexpr (W == DT::DQ4K) {
                const uint2 hd = hd4[slot][j];
                const float d = f16_bits_to_f32(hd.x & 0xFFFF);
                const float dmin = f16_bits_to_f32(hd.x >> 16);
                float ql = 0.f, qh = 0.f;
                #pragma unroll
                for (int t = 0; t < 4; t++) {
                    const uint32_t dw = qv[slot][j][t];
                    ql += dot16u(dw & 0x0F0F0F0Fu, xs4[xb[j] + t]);
                    qh += dot16u((dw >> 4) & 0x0F0F0F0Fu, xs4[xb[j] + 8 + t]);
                }
                acc += (d * (float)(hd.y & 0xFF)) * ql
                     - (dmin * (float)((hd.y >> 8) & 0xFF)) * sxl[j]
                     + (d * (float)((hd.y >> 16) & 0xFF)) * qh
                     - (dmin * (float)(hd.y >> 24)) * sxh[j];
            } else {
                float q = 0.f;
                #pragma unroll
                for (int t = 0; t < 4; t++)
                    q += dot16s(qv[slot][j][t], xs4[xb[j] + t]);
                if constexpr (W == DT::DQ8) {
                    acc += f16_bits_to_f32(hs[slot][j]) * q;
                } else {
                    acc += f16_bits_to_f32(hdd[slot][j])
                         * (float)(int)(int8_t)(hs[slot][j]) * q;
                }
            }
        }
        const float v = wave_reduce_sum(acc);
        if (i < rows_my && lane == 0) {
            const int r = r0 + i;
            y[r] = v + __uint_as_float(rv[slot] & rmask);
        }
    };

    if (rows_my > 0) {
        #pragma unroll
        for (int sl = 0; sl < NB - 1; sl++) stage(sl, sl);
        for (int i0 = 0; i0 < rows_my; i0 += NB) {
            #pragma unroll
            for (int j = 0; j < NB; j++) {
                stage((j + NB - 1) % NB, i0 + j + NB - 1);
                compute(j, i0 + j);
            }
        }
    }
}

// Global-x GEMV: no LDS staging — the activation vector (<=57 KB) is
// L1-resident per CU after first touch, so reading it directly unlocks
// full occupancy (no 16 KB+ LDS budget per workgroup) and removes the
// staging phase + barrier entirely (guide: don't LDS-stage what L1/L2
// fits). xin must already be normalized/activated (k_rmsnorm_rows /
// k_silu_rows run once per projection input).
template <DT W>
__global__ __launch_bounds__(256) void k_gemv_g(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const float* __restrict__ xin, const float* __restrict__ res,
    float* __restrict__ y, int N, int K, int B, int ldy) {
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int r = (int)blockIdx.x * 4 + wave;
    const int rc = r < N ? r : N - 1;
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const int64_t hdr_rb = dhdr_row_bytes(W, K);
    const int n_chunks = (int)(qs_rb / 16);
    constexpr int WPC = ChunkTraits<W>::W_PER_CHUNK;

    ChunkRaw<W> cur, nxt;
    const int c0 = lane < n_chunks ? lane : 0;
    load_chunk<W>(qs + (int64_t)rc * qs_rb, hdr + (int64_t)rc * hdr_rb, c0,
                  &cur);
    float acc[2] = {0.f, 0.f};
    for (int c = lane; c < n_chunks; c += 64) {
        const int cn = (c + 64 < n_chunks) ? c + 64 : c;
        load_chunk<W>(qs + (int64_t)rc * qs_rb, hdr + (int64_t)rc * hdr_rb,
                      cn, &nxt);
        float w[WPC];
        int k0;
        decode_chunk_raw<W>(cur, c, w, &k0);
        #pragma unroll 2
        for (int b = 0; b < B; b++) {
            const float4* xb4 = reinterpret_cast<const float4*>(
                xin + (size_t)b * K + k0);
            float sdot = 0.f;
            if constexpr (W == DT::DQ4K) {
                #pragma unroll
                for (int t4 = 0; t4 < 4; t4++) {
                    const float4 xl = xb4[t4];
                    const float4 xh = xb4[8 + t4];
                    sdot += w[t4 * 4 + 0] * xl.x + w[t4 * 4 + 1] * xl.y
                       + w[t4 * 4 + 2] * xl.z + w[t4 * 4 + 3] * xl.w;
                    sdot += w[16 + t4 * 4 + 0] * xh.x + w[16 + t4 * 4 + 1] * xh.y
                       + w[16 + t4 * 4 + 2] * xh.z + w[16 + t4 * 4 + 3] * xh.w;
                }
            } else {
                #pragma unroll
                for (int t4 = 0; t4 < WPC / 4; t4++) {
                    const float4 xv = xb4[t4];
                    sdot += w[t4 * 4 + 0] * xv.x + w[t4 * 4 + 1] * xv.y
                       + w[t4 * 4 + 2] * xv.z + w[t4 * 4 + 3] * xv.w;
                }
            }
            acc[b] += sdot;
        }
        cur = nxt;
    }
    if (r < N) {
        #pragma unroll 2
        for (int b = 0; b < B; b++) {
            float v = wave_reduce_sum(acc[b]);
            if (lane == 0) {
                const int64_t idx = (int64_t)b * ldy + r;
                y[idx] = v + (res ? res[idx] : 0.f);
            }
        }
    }
}


// ------------------------------------------------------------- embedding

template <DT W>
__global__ __launch_bounds__(256) void k_embed(
    const uint8_t* __restrict__ qs, const uint8_t* __restrict__ hdr,
    const int32_t* __restrict__ ids, float* __restrict__ x, int K) {
    const int b = blockIdx.x;
    const int row = ids[b];
    const int64_t qs_rb = dqs_row_bytes(W, K);
    const uint8_t* qs_row = qs + (int64_t)row * qs_rb;
    const uint8_t* hdr_row = hdr + (int64_t)row * dhdr_row_bytes(W, K);
    const int n_chunks = (int)(qs_rb / 16);
    constexpr int WPC = ChunkTraits<W>::W_PER_CHUNK;
    float* xb = x + (size_t)b * K;
    for (int c = threadIdx.x; c < n_chunks; c += 256) {
        float w[WPC];
        int k0;
        ChunkRaw<W> r;
        load_chunk<W>(qs_row, hdr_row, c, &r);
        decode_chunk_raw<W>(r, c, w, &k0);
        if constexpr (W == DT::DQ4K) {
            #pragma unroll
            for (int t = 0; t < 16; t++) xb[k0 + t] = w[t];
            #pragma unroll
            for (int t = 0; t < 16; t++) xb[k0 + 32 + t] = w[16 + t];
        } else {
            #pragma unroll
            for (int t = 0; t < WPC; t++) xb[k0 + t] = w[t];
        }
    }
}

// -------------------------------------------------------- RoPE + append

// grid (S, KVH, B); block 256 = 4 waves; each 16-lane quarter owns a full
// (m, l, o[D]) online-softmax accumulator over its position subsequence,
// for each of the G query heads of this kv group. D = 16*DPL (64 or 128).
// RoPE is fused: every workgroup ropes its q fragment in-register; the
// split that owns position n_past computes the current token's roped k (and
// raw v) from the qkv buffer, uses it in the online softmax, and appends it
// to the cache page for future steps. No separate rope/append kernel.
template <int G, int D>
__global__ __launch_bounds__(256) void k_attn_decode(
    const float* __restrict__ qkv, const float* __restrict__ inv_freq,
    const int32_t* __restrict__ page_table,
    uint16_t* __restrict__ kv_pool, const int32_t* __restrict__ n_past,
    float* __restrict__ part_o,   // [B][NH][S][D]
    float* __restrict__ part_ml,  // [B][NH][S][2]
    int* __restrict__ tickets,    // [B][NKV] fan-in counters (0 on entry)
    float* __restrict__ attn_out, // [B][NH*D]
    int8_t* __restrict__ xq, float* __restrict__ xsc, int M4,  // fused quant
    int NH, int NKV, int S, int page_size, int max_pages,
    int64_t page_stride, float scale) {
    constexpr int DPL = D / 16;   // dims per lane (bf16: 2*DPL bytes)
    const int s = blockIdx.x, kvh = blockIdx.y, b = blockIdx.z;
    const int cap = max_pages * page_size;
    int pos = n_past[b];
    if (pos >= cap) pos = cap - 1;   // serving safety: never index past the
    const int len = pos + 1;         // page table (idle slots, see engine)
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int quarter = lane >> 4, qlane = lane & 15;
    const int sub = wave * 4 + quarter;       // 0..15 position subsequence
    const int d0 = qlane * DPL;               // this lane's dim slice

    int chunk = (len + S - 1) / S;
    chunk = (chunk + 3) & ~3;
    const int start = s * chunk;
    const int end = min(start + chunk, len);

    // rope angles for this lane's dim pairs at the current position
    float cs[DPL / 2], sn[DPL / 2];
    #pragma unroll
    for (int j = 0; j < DPL / 2; j++) {
        const float ang = (float)pos * inv_freq[d0 / 2 + j];
        __sincosf(ang, &sn[j], &cs[j]);
    }

    const float* qkv_b = qkv + (size_t)b * (NH + 2 * NKV) * D;
    float qf[G][DPL];
    #pragma unroll
    for (int g = 0; g < G; g++) {
        const float* qh = qkv_b + (size_t)(kvh * G + g) * D + d0;
        #pragma unroll
        for (int j = 0; j < DPL; j++) qf[g][j] = qh[j];
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {  // in-register RoPE
            const float x0 = qf[g][2 * j], x1 = qf[g][2 * j + 1];
            qf[g][2 * j] = (x0 * cs[j] - x1 * sn[j]) * scale;
            qf[g][2 * j + 1] = (x0 * sn[j] + x1 * cs[j]) * scale;
        }
    }

    float m[G], l[G], o[G][DPL];
    #pragma unroll
    for (int g = 0; g < G; g++) {
        m[g] = -1e30f; l[g] = 0.f;
        #pragma unroll
        for (int j = 0; j < DPL; j++) o[g][j] = 0.f;
    }

    const int cache_end = min(end, pos);  // cached positions only
    // software-pipelined scan: K and V of position p+16 are in flight while
    // p's dot/softmax math runs (the un-prefetched loop measured ~0.4 TB/s
    // — load->use serialization, not bandwidth). Prefetch addresses clamp
    // to the current position so the loads stay branchless (counted
    // s_waitcnt instead of a vmcnt(0) drain — see the GEMV notes).
    auto kv_addr = [&](int p) {
        const int page = page_table[(size_t)b * max_pages + p / page_size];
        return kv_pool + (int64_t)page * page_stride
               + ((int64_t)kvh * 2) * page_size * D
               + (int64_t)(p % page_size) * D + d0;
    };
    const int64_t voff = (int64_t)page_size * D;   // V plane within the slot
    const int p0 = start + sub;
    // depth-2 prefetch: K/V of p+16 AND p+32 in flight while p's math
    // runs (depth 1 left the scan latency-exposed every other step)
    uint32_t kw[DPL / 2], vw[DPL / 2], kwn[DPL / 2], vwn[DPL / 2];
    uint32_t kw2[DPL / 2], vw2[DPL / 2];
    if (p0 < cache_end) {
        const uint16_t* kp = kv_addr(p0);
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            kw[j] = reinterpret_cast<const uint32_t*>(kp)[j];
            vw[j] = reinterpret_cast<const uint32_t*>(kp + voff)[j];
        }
        const int p1 = (p0 + 16 < cache_end) ? p0 + 16 : p0;
        const uint16_t* kp1 = kv_addr(p1);
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            kwn[j] = reinterpret_cast<const uint32_t*>(kp1)[j];
            vwn[j] = reinterpret_cast<const uint32_t*>(kp1 + voff)[j];
        }
    }
    for (int p = p0; p < cache_end; p += 16) {
        {
            const int pn = (p + 32 < cache_end) ? p + 32 : p;
            const uint16_t* kp = kv_addr(pn);
            #pragma unroll
            for (int j = 0; j < DPL / 2; j++) {
                kw2[j] = reinterpret_cast<const uint32_t*>(kp)[j];
                vw2[j] = reinterpret_cast<const uint32_t*>(kp + voff)[j];
            }
        }
        float kf[DPL];
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            kf[2 * j] = bf16_bits_to_f32(kw[j] & 0xFFFF);
            kf[2 * j + 1] = bf16_bits_to_f32(kw[j] >> 16);
        }
        float sc[G];
        #pragma unroll
        for (int g = 0; g < G; g++) {
            float d = 0.f;
            #pragma unroll
            for (int j = 0; j < DPL; j++) d += qf[g][j] * kf[j];
            #pragma unroll
            for (int off = 1; off < 16; off <<= 1) d += __shfl_xor(d, off, 64);
            sc[g] = d;
        }
        float vf[DPL];
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            vf[2 * j] = bf16_bits_to_f32(vw[j] & 0xFFFF);
            vf[2 * j + 1] = bf16_bits_to_f32(vw[j] >> 16);
        }
        #pragma unroll
        for (int g = 0; g < G; g++) {
            const float mn = fmaxf(m[g], sc[g]);
            const float alpha = __expf(m[g] - mn);
            const float w = __expf(sc[g] - mn);
            l[g] = l[g] * alpha + w;
            #pragma unroll
            for (int j = 0; j < DPL; j++) o[g][j] = o[g][j] * alpha + w * vf[j];
            m[g] = mn;
        }
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            kw[j] = kwn[j]; vw[j] = vwn[j];
            kwn[j] = kw2[j]; vwn[j] = vw2[j];
        }
    }

    // current token: the quarter whose subsequence covers `pos` computes
    // roped k / raw v from the qkv buffer, folds it into its accumulator,
    // and appends both to the cache page.
    if (pos >= start && pos < end && ((pos - start) & 15) == sub) {
        const float* kh = qkv_b + (size_t)(NH + kvh) * D + d0;
        const float* vh = qkv_b + (size_t)(NH + NKV + kvh) * D + d0;
        float kf[DPL], vf[DPL];
        #pragma unroll
        for (int j = 0; j < DPL; j++) { kf[j] = kh[j]; vf[j] = vh[j]; }
        #pragma unroll
        for (int j = 0; j < DPL / 2; j++) {
            const float x0 = kf[2 * j], x1 = kf[2 * j + 1];
            kf[2 * j] = x0 * cs[j] - x1 * sn[j];
            kf[2 * j + 1] = x0 * sn[j] + x1 * cs[j];
        }
        // append to cache (bf16)
        const int page = page_table[(size_t)b * max_pages + pos / page_size];
        uint16_t* kdst = kv_pool + (int64_t)page * page_stride
                         + ((int64_t)kvh * 2 + 0) * page_size * D
                         + (int64_t)(pos % page_size) * D + d0;
        uint16_t* vdst = kv_pool + (int64_t)page * page_stride
                         + ((int64_t)kvh * 2 + 1) * page_size * D
                         + (int64_t)(pos % page_size) * D + d0;
        #pragma unroll
        for (int j = 0; j < DPL; j++) {
            kdst[j] = f32_to_bf16_bits(kf[j]);
            vdst[j] = f32_to_bf16_bits(vf[j]);
        }
        // fold into the online softmax (k/v as bf16-rounded, matching what
        // future steps will read from the cache)
        #pragma unroll
        for (int j = 0; j < DPL; j++) {
            kf[j] = bf16_bits_to_f32(kdst[j]);
            vf[j] = bf16_bits_to_f32(vdst[j]);
        }
        float sc[G];
        #pragma unroll
        for (int g = 0; g < G; g++) {
            float d = 0.f;
            #pragma unroll
            for (int j = 0; j < DPL; j++) d += qf[g][j] * kf[j];
            #pragma unroll
            for (int off = 1; off < 16; off <<= 1) d += __shfl_xor(d, off, 64);
            sc[g] = d;
        }
        #pragma unroll
        for (int g = 0; g < G; g++) {
            const float mn = fmaxf(m[g], sc[g]);
            const float alpha = __expf(m[g] - mn);
            const float w = __expf(sc[g] - mn);
            l[g] = l[g] * alpha + w;
            #pragma unroll
            for (int j = 0; j < DPL; j++) o[g][j] = o[g][j] * alpha + w * vf[j];
            m[g] = mn;
        }
    }

    // combine the 16 sub-accumulators through LDS
    __shared__ __attribute__((aligned(16))) float sm[16 * G];
    __shared__ __attribute__((aligned(16))) float sl[16 * G];
    __shared__ __attribute__((aligned(16))) float so[16 * G * D];
    #pragma unroll
    for (int g = 0; g < G; g++) {
        if (qlane == 0) {
            sm[sub * G + g] = m[g];
            sl[sub * G + g] = l[g];
        }
        #pragma unroll
        for (int j = 0; j < DPL; j++) so[(sub * G + g) * D + d0 + j] = o[g][j];
    }
    __syncthreads();
    if (S == 1) {
        // Single split: this WG owns (b, kvh) outright — combine the 16
        // sub-accumulators straight from LDS into attn_out (+ fused
        // quant) and skip the whole fan-in: no part_o round trip, no
        // vmcnt(0) drains, no ticket atomic, and crucially no
        // agent-scope acquire (which invalidates the XCD's L2 on CDNA).
        if (xq) {
            const int jl = tid & 7;
            for (int b32 = tid >> 3; b32 < G * D / 32; b32 += 32) {
                const int g = (b32 * 32) / D;
                const int head = kvh * G + g;
                float mstar = -1e30f;
                #pragma unroll
                for (int t = 0; t < 16; t++)
                    mstar = fmaxf(mstar, sm[t * G + g]);
                float denom = 0.f;
                #pragma unroll
                for (int t = 0; t < 16; t++)
                    denom += __expf(sm[t * G + g] - mstar) * sl[t * G + g];
                float4 v;
                float* vp = &v.x;
                #pragma unroll
                for (int j = 0; j < 4; j++) {
                    const int d = (b32 * 32 + jl * 4 + j) % D;
                    float osum = 0.f;
                    #pragma unroll
                    for (int t = 0; t < 16; t++)
                        osum += __expf(sm[t * G + g] - mstar) *
                                so[(t * G + g) * D + d];
                    vp[j] = osum / denom;
                    attn_out[(size_t)b * NH * D + (size_t)head * D + d] =
                        vp[j];
                }
                quant_block_emit(v, jl, kvh * (G * D / 32) + b32, b,
                                 NH * D, M4, xq, xsc);
            }
        } else {
            for (int idx = tid; idx < G * D; idx += 256) {
                const int g = idx / D, d = idx % D;
                float mstar = -1e30f;
                #pragma unroll
                for (int t = 0; t < 16; t++)
                    mstar = fmaxf(mstar, sm[t * G + g]);
                float lsum = 0.f, osum = 0.f;
                #pragma unroll
                for (int t = 0; t < 16; t++) {
                    const float e = __expf(sm[t * G + g] - mstar);
                    lsum += e * sl[t * G + g];
                    osum += e * so[(t * G + g) * D + d];
                }
                const int head = kvh * G + g;
                attn_out[(size_t)b * NH * D + (size_t)head * D + d] =
                    osum / lsum;
            }
        }
        return;
    }
    for (int idx = tid; idx < G * D; idx += 256) {
        const int g = idx / D, d = idx % D;
        float mstar = -1e30f;
        #pragma unroll
        for (int t = 0; t < 16; t++) mstar = fmaxf(mstar, sm[t * G + g]);
        float lsum = 0.f, osum = 0.f;
        #pragma unroll
        for (int t = 0; t < 16; t++) {
            const float e = __expf(sm[t * G + g] - mstar);
            lsum += e * sl[t * G + g];
            osum += e * so[(t * G + g) * D + d];
        }
        const int head = kvh * G + g;
        part_o[(((size_t)b * NH + head) * S + s) * D + d] = osum;
        if (d == 0) {
            part_ml[(((size_t)b * NH + head) * S + s) * 2 + 0] = mstar;
            part_ml[(((size_t)b * NH + head) * S + s) * 2 + 1] = lsum;
        }
    }

    // ---- fan-in: the LAST split workgroup of (b, kvh) combines ----
    // Guide §6 G16 recipe: every wave drains its stores, __syncthreads,
    // one lane agent-scope release + asm vmcnt(0) (restates the wait the
    // compiler may drop, G16 pitfall 12), relaxed agent ticket add; the
    // last arriver acquires, then reads every split's partials with plain
    // loads. Placement-independent (no XCD/dispatch assumptions).
    __shared__ int is_last;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // every storing wave
    __syncthreads();
    if (tid == 0) {
        __builtin_amdgcn_fence(__ATOMIC_RELEASE, "agent");
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        const int t = __hip_atomic_fetch_add(
            &tickets[(size_t)b * NKV + kvh], 1, __ATOMIC_RELAXED,
            __HIP_MEMORY_SCOPE_AGENT);
        is_last = (t == S - 1);
        if (is_last) {
            __builtin_amdgcn_fence(__ATOMIC_ACQUIRE, "agent");
            // reset for the next launch (we are the only reader)
            __hip_atomic_store(&tickets[(size_t)b * NKV + kvh], 0,
                               __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
    }
    __syncthreads();
    if (!is_last) return;
    if (xq) {
        // combine + fused i8-GEMM activation quantization (8 lanes per
        // 32-block; D%32==0 so a block never crosses heads)
        const int jl = tid & 7;
        for (int b32 = tid >> 3; b32 < G * D / 32; b32 += 32) {
            const int g = (b32 * 32) / D;
            const int head = kvh * G + g;
            const float* ml = part_ml + ((size_t)b * NH + head) * S * 2;
            float mg = -1e30f;
            for (int t = 0; t < S; t++) mg = fmaxf(mg, ml[2 * t]);
            float denom = 0.f;
            for (int t = 0; t < S; t++)
                denom += __expf(ml[2 * t] - mg) * ml[2 * t + 1];
            float4 v;
            float* vp = &v.x;
            #pragma unroll
            for (int j = 0; j < 4; j++) {
                const int d = (b32 * 32 + jl * 4 + j) % D;
                float osum = 0.f;
                #pragma unroll 4
                for (int t = 0; t < S; t++)
                    osum += __expf(ml[2 * t] - mg) *
                            part_o[(((size_t)b * NH + head) * S + t) * D + d];
                vp[j] = osum / denom;
                attn_out[(size_t)b * NH * D + (size_t)head * D + d] = vp[j];
            }
            const int kg = kvh * (G * D / 32) + b32;
            quant_block_emit(v, jl, kg, b, NH * D, M4, xq, xsc);
        }
        return;
    }
    for (int idx = tid; idx < G * D; idx += 256) {
        const int g = idx / D, d = idx % D;
        const int head = kvh * G + g;
        const float* ml = part_ml + ((size_t)b * NH + head) * S * 2;
        float mg = -1e30f;
        for (int t = 0; t < S; t++) mg = fmaxf(mg, ml[2 * t]);
        float denom = 0.f, osum = 0.f;
        #pragma unroll 4
        for (int t = 0; t < S; t++) {
            const float e = __expf(ml[2 * t] - mg);
            denom += e * ml[2 * t + 1];
            osum += e * part_o[(((size_t)b * NH + head) * S + t) * D + d];
        }
        attn_out[(size_t)b * NH * D + (size_t)head * D + d] = osum / denom;
    }
}

// grid (B*NH); block 128. Combine S split-KV partials into attn_out.
// ml pairs staged to LDS first (one coalesced read), then the per-dim o
// reduction issues S independent loads per thread.
__global__ __launch_bounds__(128) void k_attn_combine(
    const float* __restrict__ part_o, const float* __restrict__ part_ml,
    float* __restrict__ attn_out, int NH, int S, int D) {
    const int bh = blockIdx.x;  // b * NH + head
    const int d = threadIdx.x;
    __shared__ __attribute__((aligned(16))) float ml[2 * 64];  // S <= 64
    for (int i = d; i < 2 * S; i += 128)
        ml[i] = part_ml[(size_t)bh * S * 2 + i];
    __syncthreads();
    float mstar = -1e30f;
    for (int s = 0; s < S; s++) mstar = fmaxf(mstar, ml[2 * s]);
    float denom = 0.f;
    for (int s = 0; s < S; s++) denom += __expf(ml[2 * s] - mstar) * ml[2 * s + 1];
    float osum = 0.f;
    #pragma unroll 4
    for (int s = 0; s < S; s++)
        osum += __expf(ml[2 * s] - mstar) * part_o[((size_t)bh * S + s) * D + d];
    attn_out[(size_t)bh * D + d] = osum / denom;
}

// ------------------------------------------------------ argmax sampling

// stage 1: grid (NCHUNK, B); each block scans a slice of logits[b].
__global__ __launch_bounds__(256) void k_argmax_part(
    const float* __restrict__ logits, float* __restrict__ pval,
    int32_t* __restrict__ pidx, int V, int nchunk) {
    const int c = blockIdx.x, b = blockIdx.y;
    const int per = (V + nchunk - 1) / nchunk;
    const int lo = c * per, hi = min(lo + per, V);
    float best = -1e30f;
    int besti = 0;
    for (int i = lo + (int)threadIdx.x; i < hi; i += 256) {
        const float v = logits[(size_t)b * V + i];
        if (v > best) { best = v; besti = i; }
    }
    __shared__ float sv[256];
    __shared__ int si[256];
    sv[threadIdx.x] = best;
    si[threadIdx.x] = besti;
    __syncthreads();
    for (int off = 128; off > 0; off >>= 1) {
        if (threadIdx.x < off && sv[threadIdx.x + off] > sv[threadIdx.x]) {
            sv[threadIdx.x] = sv[threadIdx.x + off];
            si[threadIdx.x] = si[threadIdx.x + off];
        }
        __syncthreads();
    }
    if (threadIdx.x == 0) {
        pval[(size_t)b * nchunk + c] = sv[0];
        pidx[(size_t)b * nchunk + c] = si[0];
    }
}

// stage 2 + state advance: grid B, block 64. Idle slots (slot_active[b]==0)
// do not advance: their n_past stays pinned so the shared decode step never
// scans garbage KV for them (serving: slots are parked between requests).
__global__ __launch_bounds__(64) void k_argmax_final(
    const float* __restrict__ pval, const int32_t* __restrict__ pidx,
    int32_t* __restrict__ cur_ids, int32_t* __restrict__ n_past,
    int32_t* __restrict__ gen_tokens, int32_t* __restrict__ gen_count,
    const uint8_t* __restrict__ slot_active, int nchunk, int gen_cap) {
    const int b = blockIdx.x;
    if (!slot_active[b]) return;
    const int t = threadIdx.x;
    float v = (t < nchunk) ? pval[(size_t)b * nchunk + t] : -1e30f;
    int i = (t < nchunk) ? pidx[(size_t)b * nchunk + t] : 0;
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const float ov = __shfl_xor(v, off, 64);
        const int oi = __shfl_xor(i, off, 64);
        if (ov > v) { v = ov; i = oi; }
    }
    if (t == 0) {
        cur_ids[b] = i;
        const int gc = gen_count[b];
        if (gc < gen_cap) gen_tokens[(size_t)b * gen_cap + gc] = i;
        gen_count[b] = gc + 1;
        n_past[b] = n_past[b] + 1;
    }
}

// TP logits epilogue: undo the rank-major ncclAllGather layout in one
// launch. src is [TP][B][Vl] (each rank's compact [B][Vl] slice gathered in
// rank order), dst is [B][TP*Vl] global logits. One collective of B*Vl per
// step instead of the round-1 per-batch-row loop (B collectives of Vl).
__global__ __launch_bounds__(256) void k_scatter_logits(
    const float* __restrict__ src, float* __restrict__ dst, int B, int Vl) {
    const int r = blockIdx.x, b = blockIdx.y;
    const int TP = gridDim.x;
    const float4* s = reinterpret_cast<const float4*>(
        src + ((size_t)r * B + b) * Vl);
    float4* d = reinterpret_cast<float4*>(
        dst + ((size_t)b * TP + r) * Vl);
    const int n4 = Vl >> 2;
    for (int i = threadIdx.x; i < n4; i += 256) d[i] = s[i];
}

void launch_scatter_logits(const float* src, float* dst, int B, int Vl,
                           int TP, hipStream_t stream) {
    hipLaunchKernelGGL(k_scatter_logits, dim3(TP, B), dim3(256), 0, stream,
                       src, dst, B, Vl);
}


// Clear up to 4 device regions in one launch (replaces several
// hipMemsetAsync nodes ahead of split-K GEMMs: each memset costs a

// Layer-start prep: rmsnorm the residual rows AND clear the split-K
// destinations in one launch (blocks 0..B-1 normalize one row each; the
// rest zero the regions). Replaces a zero4 + rmsnorm_rows pair.
__global__ __launch_bounds__(256) void k_layer_prep(
    const float* __restrict__ X, const float* __restrict__ gw,
    float* __restrict__ xn, int B, int K, float eps,
    float* __restrict__ p0, int n0, float* __restrict__ p1, int n1,
    float* __restrict__ p2, int n2, float* __restrict__ p3, int n3,
    int8_t* __restrict__ xq, float* __restrict__ xsc, int M4) {
    const int blk = blockIdx.x;
    if (blk < B) {
        // rmsnorm row blk (same structure as k_rmsnorm_rows), with the
        // i8-GEMM activation quantization fused (xq non-null): saves one
        // kernel launch + a full re-read of xn per projection input
        const float4* x4 = reinterpret_cast<const float4*>(X + (size_t)blk * K);
        float4* o4 = reinterpret_cast<float4*>(xn + (size_t)blk * K);
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
                quant_block_emit(o, jl, b32, blk, K, M4, xq, xsc);
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
        return;
    }
    const int tid = (blk - B) * 256 + threadIdx.x;
    const int stride = (gridDim.x - B) * 256;
    const float4 z = {0.f, 0.f, 0.f, 0.f};
    for (int i = tid; i < n0 >> 2; i += stride)
        reinterpret_cast<float4*>(p0)[i] = z;
    for (int i = tid; i < n1 >> 2; i += stride)
        reinterpret_cast<float4*>(p1)[i] = z;
    for (int i = tid; i < n2 >> 2; i += stride)
        reinterpret_cast<float4*>(p2)[i] = z;
    for (int i = tid; i < n3 >> 2; i += stride)
        reinterpret_cast<float4*>(p3)[i] = z;
}

void launch_layer_prep(const float* X, const float* gw, float* xn, int B,
                       int K, float eps, float* p0, int64_t n0, float* p1,
                       int64_t n1, float* p2, int64_t n2, float* p3,
                       int64_t n3, int8_t* xq, float* xsc,
                       hipStream_t stream) {
    const int64_t total = (n0 + n1 + n2 + n3) >> 2;
    int zb = (int)((total + 255) / 256);
    if (zb > 1024) zb = 1024;
    if (zb < 1) zb = 1;
    const int M4 = (B + 3) & ~3;
    hipLaunchKernelGGL(k_layer_prep, dim3(B + zb), dim3(256), 0, stream,
                       X, gw, xn, B, K, eps, p0, (int)n0, p1, (int)n1,
                       p2, (int)n2, p3, (int)n3, xq, xsc, M4);
}


// --------------------------------------------------------- launch stubs

#define DISPATCH_DT(DTV, FN)                                          \
    switch (DTV) {                                                    \
        case DT::DQ4K: FN(DT::DQ4K); break;                           \
        case DT::DQ6K: FN(DT::DQ6K); break;                           \
        case DT::DQ8:  FN(DT::DQ8);  break;                           \
        case DT::BF16: FN(DT::BF16); break;                           \
        case DT::F16:  FN(DT::F16);  break;                           \
        case DT::F32:  FN(DT::F32);  break;                           \
        default: throw std::runtime_error("bad dtype");               \
    }

void launch_gemv_q8(const WTensor& w, int pre, const float* xin,
                    const float* gw, const float* res, float* y, int B,
                    int ldy, float eps, hipStream_t stream) {
    const int N = (int)w.n, K = (int)w.k;
    if (B > 2) throw std::runtime_error("GEMV path supports B<=2");
    const int X8B = ((K + (K >> 4)) + 15) & ~15;
    const size_t lds = (size_t)B * X8B + (size_t)B * (K / 32) * 4
                     + (size_t)B * (K / 16) * 4 + 8 * 4;
    dim3 grid((N + 3) / 4), block(256);
    #define GEMVQ_CASE(WT)                                                     \
        do {                                                                   \
            auto kern = (pre == PRE_RMS) ? k_gemv_q8<WT, PRE_RMS>              \
                       : (pre == PRE_SILU) ? k_gemv_q8<WT, PRE_SILU>           \
                       : k_gemv_q8<WT, PRE_NONE>;                              \
            hipLaunchKernelGGL(kern, grid, block, lds, stream,                 \
                (const uint8_t*)w.qs, (const uint8_t*)w.hdr, xin, gw, res, y,  \
                N, K, B, ldy, eps);                                            \
        } while (0)
    switch (w.dtype) {
        case DT::DQ4K: GEMVQ_CASE(DT::DQ4K); break;
        case DT::DQ6K: GEMVQ_CASE(DT::DQ6K); break;
        case DT::DQ8:  GEMVQ_CASE(DT::DQ8);  break;
        default: throw std::runtime_error("q8 gemv: quant dtypes only");
    }
    #undef GEMVQ_CASE
}

static int gemv_r_mode() {   // CLA_GEMV_R=0 -> legacy LDS kernel (A/B)
    static int v = -1;
    if (v < 0) {
        const char* e = getenv("CLA_GEMV_R");
        v = (e && e[0] == '0') ? 0 : 1;
    }
    return v;
}

void launch_gemv(const WTensor& w, int pre, const float* xin, const float* gw,
                 const float* res, float* y, int B, int ldy, float eps,
                 hipStream_t stream) {
    const int N = (int)w.n, K = (int)w.k;
    if (B > 2) throw std::runtime_error("GEMV path supports B<=2");
    const bool quant_w = (w.dtype == DT::DQ4K || w.dtype == DT::DQ6K ||
                          w.dtype == DT::DQ8);
    if (B <= 2 && quant_w && gemv_r_mode() &&
        (K == 2048 || K == 4096)) {
        // Single-stripe shapes only: for K > 4096 (the down projection)
        // a gridDim.z-striped atomic variant measured ~33 us vs the
        // legacy LDS kernel's 23 us (x re-staging there is L2-served and
        // cheap at B=1), so multi-stripe stays unselected.
        // nwg: ~8 waves/CU saturates the stream for big N; smaller N
        // can't afford the per-wave ramp at 512 WGs (rows/wave < 4).
        // B=2 shares one weight stream between both rows (BB=2).
        int nwg = (N >= 16384) ? 512 : 256;
        nwg = std::min(nwg, (N + 3) / 4);
        static int wgs_env = [] {
            const char* e = getenv("CLA_GEMVR_WGS");
            return e ? atoi(e) : 0;
        }();
        if (wgs_env > 0) nwg = std::min(wgs_env, (N + 3) / 4);
        #define GEMVR_SEG(WT, SEGF, BBV)                                       \
            do {                                                               \
                auto kern = (pre == PRE_RMS)                                   \
                        ? k_gemv_r<WT, PRE_RMS, SEGF, BBV>                     \
                        : (pre == PRE_SILU)                                    \
                        ? k_gemv_r<WT, PRE_SILU, SEGF, BBV>                    \
                        : k_gemv_r<WT, PRE_NONE, SEGF, BBV>;                   \
                hipLaunchKernelGGL(kern, dim3(nwg, 1, 1), dim3(256), 0,        \
                    stream, (const uint8_t*)w.qs, (const uint8_t*)w.hdr,       \
                    xin, gw, res, y, N, K, 0, ldy, eps);                       \
            } while (0)
        #define GEMVR_CASE(WT)                                                 \
            do {                                                               \
                if (B == 2) {                                                  \
                    if (K == 4096) GEMVR_SEG(WT, 64, 2);                       \
                    else GEMVR_SEG(WT, 32, 2);                                 \
                } else if (K == 4096) {                                        \
                    GEMVR_SEG(WT, 64, 1);                                      \
                } else {                                                       \
                    GEMVR_SEG(WT, 32, 1);                                      \
                }                                                              \
            } while (0)
        switch (w.dtype) {
            case DT::DQ4K: GEMVR_CASE(DT::DQ4K); return;
            case DT::DQ6K: GEMVR_CASE(DT::DQ6K); return;
            case DT::DQ8:  GEMVR_CASE(DT::DQ8);  return;
            default: break;
        }
        #undef GEMVR_CASE
        #undef GEMVR_SEG
    }
    if (B == 1 && quant_w && gemv_r_mode() && K == 14336 &&
        pre != PRE_RMS) {
        // long-K (down projection): LDS-x row streaming, silu folded
        // into the coalesced x pass (k_gemv_rl doc above)
        const int nwg = std::min(256, (N + 3) / 4);
        #define GEMVRL_CASE(WT)                                                \
            do {                                                               \
                auto kern = (pre == PRE_SILU) ? k_gemv_rl<WT, PRE_SILU, 7>     \
                                              : k_gemv_rl<WT, PRE_NONE, 7>;    \
                hipLaunchKernelGGL(kern, dim3(nwg), dim3(256), 0, stream,      \
                    (const uint8_t*)w.qs, (const uint8_t*)w.hdr, xin, res, y,  \
                    N, K);                                                     \
            } while (0)
        switch (w.dtype) {
            case DT::DQ4K: GEMVRL_CASE(DT::DQ4K); return;
            case DT::DQ6K: GEMVRL_CASE(DT::DQ6K); return;
            case DT::DQ8:  GEMVRL_CASE(DT::DQ8);  return;
            default: break;
        }
        #undef GEMVRL_CASE
    }
    const size_t lds = (size_t)B * (K + (K >> 4)) * 4 + 8 * 4;
    // rows per wave fixed at 1, 256-thread blocks: the measured optimum
    // (A/B-rejected: RPW>1, 3-buffer rotation, pair-unroll, 512-thread
    // blocks, global-x/no-LDS, int8-dot default — see profiles/).
    const bool big = false;
    const int wpb = big ? 8 : 4;                 // waves per block
    dim3 grid((N + wpb - 1) / wpb), block(wpb * 64);
    #define GEMV_NT(WT, NTV)                                                     \
        do {                                                                     \
            auto kern = (pre == PRE_RMS) ? k_gemv<WT, PRE_RMS, 1, NTV>           \
                       : (pre == PRE_SILU) ? k_gemv<WT, PRE_SILU, 1, NTV>        \
                       : k_gemv<WT, PRE_NONE, 1, NTV>;                           \
            if (lds > 64 * 1024) {                                               \
                (void)hipFuncSetAttribute((const void*)kern,                     \
                    hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);       \
            }                                                                    \
            hipLaunchKernelGGL(kern, grid, block, lds, stream,                   \
                (const uint8_t*)w.qs, (const uint8_t*)w.hdr, xin, gw, res, y,    \
                N, K, B, ldy, eps);                                              \
        } while (0)
    #define GEMV_CASE(WT)                                                        \
        do {                                                                     \
            if (big) GEMV_NT(WT, 512);                                           \
            else GEMV_NT(WT, 256);                                               \
        } while (0)
    DISPATCH_DT(w.dtype, GEMV_CASE);
    #undef GEMV_CASE
    #undef GEMV_NT
}

void launch_gemv_g(const WTensor& w, const float* xin, const float* res,
                   float* y, int B, int ldy, hipStream_t stream) {
    const int N = (int)w.n, K = (int)w.k;
    if (B > 2) throw std::runtime_error("GEMV path supports B<=2");
    dim3 grid((N + 3) / 4), block(256);
    #define GEMVG_CASE(WT)                                                     \
        hipLaunchKernelGGL(k_gemv_g<WT>, grid, block, 0, stream,               \
            (const uint8_t*)w.qs, (const uint8_t*)w.hdr, xin, res, y,          \
            N, K, B, ldy)
    DISPATCH_DT(w.dtype, GEMVG_CASE);
    #undef GEMVG_CASE
}


void launch_embed(const WTensor& w, const int32_t* ids, float* x, int B,
                  hipStream_t stream) {
    #define EMBED_CASE(WT)                                                     \
        hipLaunchKernelGGL(k_embed<WT>, dim3(B), dim3(256), 0, stream,         \
            (const uint8_t*)w.qs, (const uint8_t*)w.hdr, ids, x, (int)w.k)
    DISPATCH_DT(w.dtype, EMBED_CASE);
    #undef EMBED_CASE
}

void launch_attn_decode(const float* qkv, const float* inv_freq,
                        const int32_t* page_table,
                        uint16_t* kv_pool, const int32_t* n_past,
                        float* part_o, float* part_ml, int* tickets,
                        float* attn_out, int8_t* xq, float* xsc, int B,
                        int NH, int NKV, int D, int S, int page_size,
                        int max_pages, int64_t page_stride, float scale,
                        hipStream_t stream) {
    const int G = NH / NKV;
    const int M4 = (B + 3) & ~3;
    dim3 grid(S, NKV, B), block(256);
    #define ATTN_CASE(GV, DV)                                                   \
        hipLaunchKernelGGL((k_attn_decode<GV, DV>), grid, block, 0, stream,     \
            qkv, inv_freq, page_table, kv_pool, n_past, part_o, part_ml,        \
            tickets, attn_out, xq, xsc, M4, NH, NKV, S, page_size, max_pages,   \
            page_stride, scale)
    #define ATTN_D(GV)                                                          \
        do { if (D == 128) ATTN_CASE(GV, 128);                                  \
             else if (D == 64) ATTN_CASE(GV, 64);                               \
             else throw std::runtime_error("head_dim must be 64 or 128");       \
        } while (0)
    switch (G) {
        case 1: ATTN_D(1); break;
        case 2: ATTN_D(2); break;
        case 4: ATTN_D(4); break;
        case 8: ATTN_D(8); break;
        default: throw std::runtime_error("unsupported GQA ratio");
    }
    #undef ATTN_D
    #undef ATTN_CASE
}

void launch_attn_combine(const float* part_o, const float* part_ml,
                         float* attn_out, int B, int NH, int S, int D,
                         hipStream_t stream) {
    hipLaunchKernelGGL(k_attn_combine, dim3(B * NH), dim3(D), 0, stream,
                       part_o, part_ml, attn_out, NH, S, D);
}

void launch_argmax(const float* logits, float* pval, int32_t* pidx,
                   int32_t* cur_ids, int32_t* n_past, int32_t* gen_tokens,
                   int32_t* gen_count, const uint8_t* slot_active, int B,
                   int V, int gen_cap, hipStream_t stream) {
    constexpr int NCHUNK = 64;
    hipLaunchKernelGGL(k_argmax_part, dim3(NCHUNK, B), dim3(256), 0, stream,
                       logits, pval, pidx, V, NCHUNK);
    hipLaunchKernelGGL(k_argmax_final, dim3(B), dim3(64), 0, stream,
                       pval, pidx, cur_ids, n_past, gen_tokens, gen_count,
                       slot_active, NCHUNK, gen_cap);
}

}  // namespace cla
