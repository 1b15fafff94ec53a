#include "engine.h"

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <functional>
#include <thread>

#include <rccl/rccl.h>

#include "gguf.h"

namespace cla {

#define NCCL_CHECK(expr)                                                      \
    do {                                                                      \
        ncclResult_t _r = (expr);                                             \
        if (_r != ncclSuccess)                                                \
            throw std::runtime_error(std::string("RCCL error: ") +            \
                                     ncclGetErrorString(_r));                 \
    } while (0)

// kernel launchers (kernels.hip)
void launch_gemv(const WTensor&, int pre, const float* xin, const float* gw,
                 const float* res, float* y, int B, int ldy, float eps,
                 hipStream_t);
void launch_gemv_q8(const WTensor&, int pre, const float* xin,
                    const float* gw, const float* res, float* y, int B,
                    int ldy, float eps, hipStream_t);
void launch_embed(const WTensor&, const int32_t* ids, float* x, int B,
                  hipStream_t);
void launch_attn_decode(const float* qkv, const float* inv_freq,
                        const int32_t* page_table,
                        uint16_t* kv_pool, const int32_t* n_past,
                        float* part_o, float* part_ml, int* tickets,
                        float* attn_out, int8_t* xq, float* xsc, int B,
                        int NH, int NKV, int D, int S, int page_size,
                        int max_pages, int64_t page_stride, float scale,
                        hipStream_t);
void launch_argmax(const float* logits, float* pval, int32_t* pidx,
                   int32_t* cur_ids, int32_t* n_past, int32_t* gen_tokens,
                   int32_t* gen_count, const uint8_t* slot_active, int B,
                   int V, int gen_cap, hipStream_t);
void launch_gemm(const WTensor&, const float* X, const float* res, float* C,
                 int M, int ldc, hipStream_t);
void launch_gemm_ex(const WTensor&, const float* X, const float* X2, int ldx,
                    bool xsilu, const float* res, float* C, int M, int ldc,
                    hipStream_t);
bool gemm_uses_splitk(int N, int K, int M);
void launch_layer_prep(const float* X, const float* gw, float* xn, int B,
                       int K, float eps, float* p0, int64_t n0, float* p1,
                       int64_t n1, float* p2, int64_t n2, float* p3,
                       int64_t n3, int8_t* xq, float* xsc, hipStream_t);
void launch_rmsnorm_rows(const float* X, const float* gw, float* out, int M,
                         int K, float eps, hipStream_t);
void launch_rmsnorm_rows_q(const float* X, const float* gw, float* out,
                           int M, int K, float eps, int8_t* xq, float* xsc,
                           hipStream_t);
void launch_scatter_logits(const float* src, float* dst, int B, int Vl,
                           int TP, hipStream_t);
// gemm_i8.hip: int8-activation MFMA path for quantized weights
void launch_quant_rows(const float* X, int8_t* xq, float* xsc, int M,
                       int K, int ldx, int mode, hipStream_t);
bool gemm_i8_supported(DT dtype, int M, int K);
void launch_gemm_i8(const WTensor& w, const int8_t* xq, const float* xsc,
                    int ldxq, const float* res, float* C, int M, int ldc,
                    hipStream_t, int force_splitk = 0);
void launch_rope_prefill(float* qkv, const float* inv_freq,
                         const int32_t* page_table, uint16_t* kv_pool,
                         int slot, int pos0, int M, int NH, int NKV, int D,
                         int page_size, int max_pages, int64_t page_stride,
                         hipStream_t);
void launch_attn_prefill(const float* qkv, const int32_t* page_table,
                         const uint16_t* kv_pool, float* attn_out, int slot,
                         int pos0, int M, int NH, int NKV, int D,
                         int page_size, int max_pages, int64_t page_stride,
                         float scale, hipStream_t);

namespace {

enum { PRE_NONE = 0, PRE_RMS = 1, PRE_SILU = 2 };

void parallel_for(int64_t n, const std::function<void(int64_t, int64_t)>& fn) {
    const int nt = std::min<int64_t>(std::thread::hardware_concurrency(), 16);
    if (nt <= 1 || n < 4) { fn(0, n); return; }
    std::vector<std::thread> ts;
    const int64_t per = (n + nt - 1) / nt;
    for (int t = 0; t < nt; t++) {
        const int64_t lo = t * per, hi = std::min<int64_t>(lo + per, n);
        if (lo >= hi) break;
        ts.emplace_back([&fn, lo, hi] { fn(lo, hi); });
    }
    for (auto& t : ts) t.join();
}

DT device_dtype(int32_t ggml_type) {
    switch (ggml_type) {
        case 0: return DT::F32;
        case 1: return DT::F16;
        case 30: return DT::BF16;
        case 8: return DT::DQ8;
        case 12: return DT::DQ4K;
        case 14: return DT::DQ6K;
        default:
            throw std::runtime_error("unsupported ggml type " +
                                     std::to_string(ggml_type));
    }
}

// Repack disk-format rows into split qs/hdr device format (see common.h DT).
void repack(int32_t ggml_type, const uint8_t* src, int64_t rows, int64_t k,
            uint8_t* qs_out, uint8_t* hdr_out) {
    const int64_t src_rb = ggml_row_bytes(ggml_type, k);
    switch (ggml_type) {
        case 0: case 1: case 30:  // float passthrough
            std::memcpy(qs_out, src, src_rb * rows);
            return;
        case 12: {  // Q4_K: disk 144B = {f16 d, f16 dmin, u8 sc6[12], qs 128}
            // device hdr: per sub-block PAIR (2 chunks): {d, dmin, sc_lo,
            // mn_lo, sc_hi, mn_hi} — 6-bit scales pre-decoded so the kernel
            // does one aligned 8B load and no divergent unpacking.
            const int64_t nsb = k / 256;
            parallel_for(rows, [&](int64_t lo, int64_t hi) {
                for (int64_t r = lo; r < hi; r++) {
                    const uint8_t* s = src + r * src_rb;
                    uint8_t* q = qs_out + r * nsb * 128;
                    uint8_t* h = hdr_out + r * nsb * 32;
                    for (int64_t b = 0; b < nsb; b++) {
                        const uint8_t* blk = s + b * 144;
                        const uint8_t* sc6 = blk + 4;
                        uint8_t sc[8], mn[8];
                        for (int j = 0; j < 4; j++) {
                            sc[j] = sc6[j] & 63;
                            mn[j] = sc6[j + 4] & 63;
                            sc[j + 4] = (sc6[j + 8] & 0xF) | ((sc6[j] >> 6) << 4);
                            mn[j + 4] = (sc6[j + 8] >> 4) | ((sc6[j + 4] >> 6) << 4);
                        }
                        for (int pr = 0; pr < 4; pr++) {
                            uint8_t* e = h + b * 32 + pr * 8;
                            std::memcpy(e, blk, 4);  // d, dmin (f16 each)
                            e[4] = sc[2 * pr];
                            e[5] = mn[2 * pr];
                            e[6] = sc[2 * pr + 1];
                            e[7] = mn[2 * pr + 1];
                        }
                        std::memcpy(q + b * 128, blk + 16, 128);
                    }
                }
            });
            return;
        }
        case 14: {  // Q6_K: disk 210B = {ql 128, qh 64, sc 16, d 2}
            const int64_t nsb = k / 256;
            parallel_for(rows, [&](int64_t lo, int64_t hi) {
                for (int64_t r = lo; r < hi; r++) {
                    const uint8_t* s = src + r * src_rb;
                    int8_t* q = reinterpret_cast<int8_t*>(qs_out + r * k);
                    uint8_t* h = hdr_out + r * nsb * 32;
                    for (int64_t b = 0; b < nsb; b++) {
                        const uint8_t* ql = s + b * 210;
                        const uint8_t* qh = ql + 128;
                        const uint8_t* sc = ql + 192;
                        // hdr: {f16 d, u8 pad[2], i8 sc[16], pad}
                        std::memcpy(h + b * 32, ql + 208, 2);
                        std::memcpy(h + b * 32 + 4, sc, 16);
                        int8_t* qb = q + b * 256;
                        for (int half = 0; half < 2; half++) {
                            const uint8_t* qlh = ql + half * 64;
                            const uint8_t* qhh = qh + half * 32;
                            for (int l = 0; l < 32; l++) {
                                const int q1 = (qlh[l] & 0xF) | (((qhh[l] >> 0) & 3) << 4);
                                const int q2 = (qlh[l + 32] & 0xF) | (((qhh[l] >> 2) & 3) << 4);
                                const int q3 = (qlh[l] >> 4) | (((qhh[l] >> 4) & 3) << 4);
                                const int q4 = (qlh[l + 32] >> 4) | (((qhh[l] >> 6) & 3) << 4);
                                int8_t* o = qb + half * 128;
                                o[l] = (int8_t)(q1 - 32);
                                o[l + 32] = (int8_t)(q2 - 32);
                                o[l + 64] = (int8_t)(q3 - 32);
                                o[l + 96] = (int8_t)(q4 - 32);
                            }
                        }
                    }
                }
            });
            return;
        }
        case 8: {  // Q8_0: disk 34B = {f16 d, i8 qs[32]}
            const int64_t nb = k / 32;
            parallel_for(rows, [&](int64_t lo, int64_t hi) {
                for (int64_t r = lo; r < hi; r++) {
                    const uint8_t* s = src + r * src_rb;
                    uint8_t* q = qs_out + r * k;
                    uint8_t* h = hdr_out + r * nb * 2;
                    for (int64_t b = 0; b < nb; b++) {
                        std::memcpy(h + b * 2, s + b * 34, 2);
                        std::memcpy(q + b * 32, s + b * 34 + 2, 32);
                    }
                }
            });
            return;
        }
        default:
            throw std::runtime_error("repack: unsupported type");
    }
}

// Slice columns [c0, c1) out of quantized rows (block-aligned: c0/c1 are
// multiples of the format's block; used for tensor-parallel col shards).
std::vector<uint8_t> slice_cols(int32_t ggml_type, const uint8_t* src,
                                int64_t rows, int64_t K, int64_t c0,
                                int64_t c1) {
    const int64_t rb = ggml_row_bytes(ggml_type, K);
    const int64_t b0 = ggml_row_bytes(ggml_type, c0);
    const int64_t bl = ggml_row_bytes(ggml_type, c1 - c0);
    std::vector<uint8_t> out((size_t)rows * bl);
    parallel_for(rows, [&](int64_t lo, int64_t hi) {
        for (int64_t r = lo; r < hi; r++)
            std::memcpy(out.data() + r * bl, src + r * rb + b0, bl);
    });
    return out;
}

}  // namespace

// test-only export of the TP column-slice helper
std::vector<uint8_t> slice_cols_test(int32_t t, const uint8_t* src,
                                     int64_t rows, int64_t k, int64_t c0,
                                     int64_t c1) {
    return slice_cols(t, src, rows, k, c0, c1);
}

Engine::Engine(const std::string& gguf_path, const EngineConfig& cfg)
    : cfg_(cfg) {
    if (cfg_.batch < 1 || cfg_.batch > 64)
        throw std::runtime_error("decode batch must be in [1, 64]");
    HIP_CHECK(hipSetDevice(cfg_.device));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    GGUFFile gf(gguf_path);
    meta_.name = gf.meta_str("general.name", "unknown");
    meta_.vocab = (int)gf.meta_int("llama.vocab_size", 0);
    meta_.hidden = (int)gf.meta_int("llama.embedding_length", 0);
    meta_.layers = (int)gf.meta_int("llama.block_count", 0);
    meta_.heads = (int)gf.meta_int("llama.attention.head_count", 0);
    meta_.kv_heads = (int)gf.meta_int("llama.attention.head_count_kv", meta_.heads);
    meta_.ffn = (int)gf.meta_int("llama.feed_forward_length", 0);
    meta_.max_ctx = (int)gf.meta_int("llama.context_length", 4096);
    meta_.rope_theta = (float)gf.meta_float("llama.rope.freq_base", 10000.0);
    meta_.rms_eps = (float)gf.meta_float("llama.attention.layer_norm_rms_epsilon", 1e-5);
    meta_.head_dim = meta_.hidden / meta_.heads;
    if (meta_.vocab == 0) {
        auto& t = gf.tensor("token_embd.weight");
        meta_.vocab = (int)t.shape[0];
    }
    const int tp = cfg_.tp_size;
    if (tp > 1) {
        if (meta_.kv_heads % tp || meta_.heads % tp || meta_.ffn % tp ||
            meta_.vocab % tp)
            throw std::runtime_error("model dims not divisible by tp_size");
        if ((meta_.hidden / tp) % 256 || (meta_.ffn / tp) % 256)
            throw std::runtime_error("tp column shards not 256-aligned");
        if (cfg_.nccl_id.size() != sizeof(ncclUniqueId))
            throw std::runtime_error("tp_size > 1 requires nccl_id bytes");
        ncclUniqueId id;
        std::memcpy(&id, cfg_.nccl_id.data(), sizeof(id));
        ncclComm_t comm;
        NCCL_CHECK(ncclCommInitRank(&comm, tp, id, cfg_.tp_rank));
        comm_ = comm;
    }
    meta_.heads_l = meta_.heads / tp;
    meta_.kv_heads_l = meta_.kv_heads / tp;
    meta_.ffn_l = meta_.ffn / tp;
    meta_.vocab_l = meta_.vocab / tp;
    load_weights(gf);
    alloc_state();
    reset();
}

Engine::~Engine() {
    if (graph_exec_) hipGraphExecDestroy(graph_exec_);
    if (comm_) ncclCommDestroy((ncclComm_t)comm_);
    for (void* p : allocs_) hipFree(p);
    if (stream_) hipStreamDestroy(stream_);
}

WTensor Engine::upload_tensor(const GGUFFile& gf, const std::string& name) {
    return upload_shard(gf, name, -1, -1, -1, -1);
}

namespace {
struct HostPack {
    DT dtype;
    int64_t rows, k;
    std::vector<uint8_t> qs, hdr;
};

HostPack repack_host(const GGUFTensor& t, int64_t r0, int64_t r1, int64_t c0,
                     int64_t c1) {
    const int64_t k_full = t.shape.back();
    int64_t rows_full = 1;
    for (size_t i = 0; i + 1 < t.shape.size(); i++) rows_full *= t.shape[i];
    if (r0 < 0) { r0 = 0; r1 = rows_full; }
    if (c0 < 0) { c0 = 0; c1 = k_full; }
    HostPack hp;
    hp.rows = r1 - r0;
    hp.k = c1 - c0;
    hp.dtype = device_dtype(t.ggml_type);
    const int64_t src_rb = ggml_row_bytes(t.ggml_type, k_full);
    const uint8_t* src = t.data + r0 * src_rb;
    std::vector<uint8_t> sliced;
    if (c0 != 0 || c1 != k_full) {
        sliced = slice_cols(t.ggml_type, src, hp.rows, k_full, c0, c1);
        src = sliced.data();
    }
    hp.qs.resize(dqs_row_bytes(hp.dtype, hp.k) * hp.rows);
    const int64_t hb = dhdr_row_bytes(hp.dtype, hp.k) * hp.rows;
    hp.hdr.resize(hb ? hb : 1);
    repack(t.ggml_type, src, hp.rows, hp.k, hp.qs.data(), hp.hdr.data());
    return hp;
}
}  // namespace

WTensor Engine::upload_pack(const void* qs, size_t qs_bytes, const void* hdr,
                            size_t hdr_bytes, DT dtype, int64_t rows,
                            int64_t k) {
    WTensor w;
    w.dtype = dtype;
    w.n = rows;
    w.k = k;
    void* d_qs = nullptr;
    HIP_CHECK(hipMalloc(&d_qs, qs_bytes));
    HIP_CHECK(hipMemcpy(d_qs, qs, qs_bytes, hipMemcpyHostToDevice));
    allocs_.push_back(d_qs);
    w.qs = d_qs;
    vram_bytes_ += qs_bytes;
    if (hdr_bytes) {
        void* d_hdr = nullptr;
        HIP_CHECK(hipMalloc(&d_hdr, hdr_bytes));
        HIP_CHECK(hipMemcpy(d_hdr, hdr, hdr_bytes, hipMemcpyHostToDevice));
        allocs_.push_back(d_hdr);
        w.hdr = d_hdr;
        vram_bytes_ += hdr_bytes;
    }
    // transposed, pre-decoded f32 header copy for the i8 GEMM's glds
    // scale staging (see common.h build_hdr2_rows)
    const int64_t h2_rb = dhdr2_row_bytes(dtype, k);
    if (h2_rb) {
        // +1 KB slack: edge-tile glds lanes past the last row read (and
        // discard) up to a wave's width beyond the array
        std::vector<uint8_t> h2((size_t)h2_rb * rows + 1024, 0);
        const uint8_t* hsrc = reinterpret_cast<const uint8_t*>(hdr);
        const int64_t hrb = dhdr_row_bytes(dtype, k);
        parallel_for(rows, [&](int64_t lo, int64_t hi) {
            build_hdr2_rows(dtype, hsrc, hrb, rows, k, lo, hi, h2.data());
        });
        void* d_h2 = nullptr;
        HIP_CHECK(hipMalloc(&d_h2, h2.size()));
        HIP_CHECK(hipMemcpy(d_h2, h2.data(), h2.size(),
                            hipMemcpyHostToDevice));
        allocs_.push_back(d_h2);
        w.hdr2 = d_h2;
        vram_bytes_ += h2.size();
    }
    // GEMM-tiled weight copy (see common.h: the row-major layout
    // over-fetches 4x at BK windows; this one DMAs whole cachelines)
    const int64_t q2_bytes = dqs2_bytes(dtype, rows, k);
    if (q2_bytes) {
        std::vector<uint8_t> q2((size_t)q2_bytes);
        const uint8_t* qsrc = reinterpret_cast<const uint8_t*>(qs);
        const int64_t qrb = dqs_row_bytes(dtype, k);
        const int64_t rows_pad = (rows + I8G_BN - 1) / I8G_BN * I8G_BN;
        parallel_for(rows_pad, [&](int64_t lo, int64_t hi) {
            build_qs2_rows(dtype, qsrc, qrb, rows, k, lo, hi, q2.data());
        });
        void* d_q2 = nullptr;
        HIP_CHECK(hipMalloc(&d_q2, q2.size()));
        HIP_CHECK(hipMemcpy(d_q2, q2.data(), q2.size(),
                            hipMemcpyHostToDevice));
        allocs_.push_back(d_q2);
        w.qs2 = d_q2;
        vram_bytes_ += q2.size();
    }
    return w;
}

// Upload a (possibly sharded) weight matrix: rows [r0,r1), cols [c0,c1);
// -1 = full range. Column shards are block-aligned byte slices.
WTensor Engine::upload_shard(const GGUFFile& gf, const std::string& name,
                             int64_t r0, int64_t r1, int64_t c0, int64_t c1) {
    HostPack hp = repack_host(gf.tensor(name), r0, r1, c0, c1);
    return upload_pack(hp.qs.data(), hp.qs.size(),
                       dhdr_row_bytes(hp.dtype, hp.k) ? hp.hdr.data() : nullptr,
                       dhdr_row_bytes(hp.dtype, hp.k) * hp.rows,
                       hp.dtype, hp.rows, hp.k);
}

const float* Engine::upload_norm(const GGUFFile& gf, const std::string& name) {
    const GGUFTensor& t = gf.tensor(name);
    if (t.ggml_type != 0)
        throw std::runtime_error(name + ": norm weights must be f32");
    void* d = nullptr;
    HIP_CHECK(hipMalloc(&d, t.nbytes));
    HIP_CHECK(hipMemcpy(d, t.data, t.nbytes, hipMemcpyHostToDevice));
    allocs_.push_back(d);
    vram_bytes_ += t.nbytes;
    return reinterpret_cast<const float*>(d);
}

// Load several row-blocks as one projection; each name contributes its
// [rank] row shard when row_shard is set. Parts with identical dtype+K are
// merged into one device tensor so a single kernel launch covers them.
Proj Engine::load_proj(const GGUFFile& gf, const std::vector<std::string>& names,
                       bool row_shard, int64_t c0, int64_t c1) {
    const int tp = cfg_.tp_size, rank = cfg_.tp_rank;
    std::vector<HostPack> packs;
    for (auto& nm : names) {
        const auto& t = gf.tensor(nm);
        int64_t rows = 1;
        for (size_t i = 0; i + 1 < t.shape.size(); i++) rows *= t.shape[i];
        int64_t pr0 = -1, pr1 = -1;
        if (row_shard && tp > 1) {
            const int64_t per = rows / tp;
            pr0 = rank * per;
            pr1 = pr0 + per;
        }
        packs.push_back(repack_host(t, pr0, pr1, c0, c1));
    }
    Proj p;
    bool mergeable = packs.size() > 1;
    for (auto& hp : packs)
        if (hp.dtype != packs[0].dtype || hp.k != packs[0].k) mergeable = false;
    if (mergeable) {
        std::vector<uint8_t> qs, hdr;
        int64_t rows = 0;
        for (auto& hp : packs) {
            qs.insert(qs.end(), hp.qs.begin(), hp.qs.end());
            if (dhdr_row_bytes(hp.dtype, hp.k))
                hdr.insert(hdr.end(), hp.hdr.begin(),
                           hp.hdr.begin() + dhdr_row_bytes(hp.dtype, hp.k) * hp.rows);
            rows += hp.rows;
        }
        WTensor w = upload_pack(qs.data(), qs.size(),
                                hdr.empty() ? nullptr : hdr.data(), hdr.size(),
                                packs[0].dtype, rows, packs[0].k);
        p.parts.push_back({w, 0});
        p.n_total = rows;
        return p;
    }
    int64_t roff = 0;
    for (auto& hp : packs) {
        WTensor w = upload_pack(hp.qs.data(), hp.qs.size(),
                                dhdr_row_bytes(hp.dtype, hp.k) ? hp.hdr.data() : nullptr,
                                dhdr_row_bytes(hp.dtype, hp.k) * hp.rows,
                                hp.dtype, hp.rows, hp.k);
        p.parts.push_back({w, roff});
        roff += w.n;
    }
    p.n_total = roff;
    return p;
}

void Engine::load_weights(const GGUFFile& gf) {
    const int tp = cfg_.tp_size, rank = cfg_.tp_rank;
    embed_ = upload_tensor(gf, "token_embd.weight");  // replicated
    out_norm_ = upload_norm(gf, "output_norm.weight");
    const std::string head_name = gf.has_tensor("output.weight")
                                      ? "output.weight" : "token_embd.weight";
    if (tp > 1) {
        const int64_t per = meta_.vocab / tp;
        head_ = upload_shard(gf, head_name, (int64_t)rank * per,
                             (int64_t)(rank + 1) * per, -1, -1);
    } else {
        head_ = (head_name == "token_embd.weight")
                    ? embed_ : upload_tensor(gf, head_name);
    }
    const int64_t h = meta_.hidden, f = meta_.ffn;
    layers_.resize(meta_.layers);
    for (int i = 0; i < meta_.layers; i++) {
        const std::string p = "blk." + std::to_string(i) + ".";
        Layer& L = layers_[i];
        L.attn_norm = upload_norm(gf, p + "attn_norm.weight");
        L.ffn_norm = upload_norm(gf, p + "ffn_norm.weight");
        L.qkv = load_proj(gf, {p + "attn_q.weight", p + "attn_k.weight",
                               p + "attn_v.weight"}, /*row_shard=*/true);
        // o / down: column shards (local input dims), all-reduced after
        if (tp > 1) {
            L.o = load_proj(gf, {p + "attn_output.weight"}, false,
                            (int64_t)rank * (h / tp),
                            (int64_t)(rank + 1) * (h / tp));
            L.down = load_proj(gf, {p + "ffn_down.weight"}, false,
                               (int64_t)rank * (f / tp),
                               (int64_t)(rank + 1) * (f / tp));
        } else {
            L.o = load_proj(gf, {p + "attn_output.weight"}, false);
            L.down = load_proj(gf, {p + "ffn_down.weight"}, false);
        }
        L.gate_up = load_proj(gf, {p + "ffn_gate.weight", p + "ffn_up.weight"},
                              /*row_shard=*/true);
    }
}

void Engine::alloc_state() {
    const int B = cfg_.batch, H = meta_.hidden, V = meta_.vocab;
    const int NH = meta_.heads_l, NKV = meta_.kv_heads_l, D = meta_.head_dim;
    const int F = meta_.ffn_l;
    // KV split count: 256/(B*NKV) with a floor of 2 up to B=32 — S=2 at
    // B=16 measured best with the round-2 i8 GEMM step (4548 vs 4396
    // tok/s at S=4), B=32 regressed at S=1 (5549 vs 6260 at S=2), and
    // B=64 has enough workgroups without splits
    attn_splits_ = std::max(1, std::min(32, 256 / std::max(1, B * NKV)));
    // floor of 2 up to B=16 only: with the fence-free S=1 combine path,
    // B=32 measured best at its natural S=1 (6732 vs 6498 tok/s) while
    // B=16 still wants S=2 (4883 vs 4766) and long contexts more
    if (attn_splits_ < 2 && B <= 16) attn_splits_ = 2;
    // B<=2: cap at 8 — at short contexts the in-kernel split combine
    // (ticket spin + fences) dominates; S=8 measured 399 vs 363 tok/s at
    // S=32 on llama3-8b B=1 (S=4 equal, S=2 worse: too few WGs)
    if (B <= 2 && attn_splits_ > 8) attn_splits_ = 8;
    if (const char* e = getenv("CLA_ATTN_SPLITS")) {   // on-HW sweeps
        const int v = atoi(e);
        if (v >= 1 && v <= 64) attn_splits_ = v;
    }
    max_pages_ = (cfg_.max_seq + cfg_.page_size - 1) / cfg_.page_size;
    // pool layout: [page][layer][kvh][2][page_size][D] bf16 — one pool, all
    // layers; per-layer base pointer passed at launch.
    layer_stride_ = (int64_t)NKV * 2 * cfg_.page_size * D;
    page_stride_ = layer_stride_ * meta_.layers;

    auto dalloc = [&](size_t bytes) {
        void* p = nullptr;
        HIP_CHECK(hipMalloc(&p, bytes));
        HIP_CHECK(hipMemset(p, 0, bytes));
        allocs_.push_back(p);
        vram_bytes_ += bytes;
        return p;
    };
    x_ = (float*)dalloc((size_t)B * H * 4);
    xn_ = (float*)dalloc((size_t)B * H * 4);
    x2_ = (float*)dalloc((size_t)B * H * 4);
    x3_ = (float*)dalloc((size_t)B * H * 4);
    tmp_h_ = (float*)dalloc((size_t)B * H * 4);
    {   // prefill scratch
        const int Mc = cfg_.prefill_chunk;
        const int QKV = (NH + 2 * NKV) * D;
        xp_ = (float*)dalloc((size_t)Mc * H * 4);
        xp2_ = (float*)dalloc((size_t)Mc * H * 4);
        xnp_ = (float*)dalloc((size_t)Mc * H * 4);
        qkvp_ = (float*)dalloc((size_t)Mc * QKV * 4);
        attnp_ = (float*)dalloc((size_t)Mc * NH * D * 4);
        gup_ = (float*)dalloc((size_t)Mc * 2 * F * 4);
        pids_ = (int32_t*)dalloc((size_t)Mc * 4);
        tmp_hp_ = (float*)dalloc((size_t)Mc * H * 4);
    }
    qkv_ = (float*)dalloc((size_t)B * (NH + 2 * NKV) * D * 4);
    attn_out_ = (float*)dalloc((size_t)B * NH * D * 4);
    gu_ = (float*)dalloc((size_t)B * 2 * F * 4);
    logits_ = (float*)dalloc((size_t)B * V * 4);
    if (cfg_.tp_size > 1) {   // compact local slice + rank-major gather buf
        logits_tp_ = (float*)dalloc((size_t)B * meta_.vocab_l * 4);
        gather_tp_ = (float*)dalloc((size_t)B * V * 4);
    }
    part_o_ = (float*)dalloc((size_t)B * NH * attn_splits_ * D * 4);
    part_ml_ = (float*)dalloc((size_t)B * NH * attn_splits_ * 2 * 4);
    attn_tickets_ = (int32_t*)dalloc((size_t)B * NKV * 4);  // zeroed once;
    // the last arriver resets its counter each launch
    amax_val_ = (float*)dalloc((size_t)B * 64 * 4);
    amax_idx_ = (int32_t*)dalloc((size_t)B * 64 * 4);
    cur_ids_ = (int32_t*)dalloc((size_t)B * 4);
    n_past_ = (int32_t*)dalloc((size_t)B * 4);
    gen_tokens_ = (int32_t*)dalloc((size_t)B * cfg_.gen_cap * 4);
    gen_count_ = (int32_t*)dalloc((size_t)B * 4);
    page_table_ = (int32_t*)dalloc((size_t)B * max_pages_ * 4);
    slot_active_ = (uint8_t*)dalloc((size_t)B);
    HIP_CHECK(hipMemset(slot_active_, 1, B));  // all-active by default
    kv_pool_ = (uint16_t*)dalloc((size_t)B * max_pages_ * page_stride_ * 2);
    {   // i8-GEMM activation-quant scratch, sized for the widest input of
        // any projection at the bigger of decode batch / prefill chunk
        const size_t maxK = std::max<size_t>(
            {(size_t)H, (size_t)F, (size_t)NH * D});
        const size_t maxM = std::max<size_t>(B, cfg_.prefill_chunk);
        xq_ = (int8_t*)dalloc(maxM * maxK);
        xsc_ = (float*)dalloc((maxM + 4) * (maxK / 32) * 2 * 4);
    }
    // rope frequency table
    std::vector<float> invf(D / 2);
    for (int i = 0; i < D / 2; i++)
        invf[i] = std::pow(meta_.rope_theta, -2.0f * i / (float)D);
    inv_freq_ = (float*)dalloc(invf.size() * 4);
    HIP_CHECK(hipMemcpy(inv_freq_, invf.data(), invf.size() * 4,
                        hipMemcpyHostToDevice));
    // static page table: slot b owns pages [b*max_pages_, (b+1)*max_pages_)
    std::vector<int32_t> pt((size_t)B * max_pages_);
    for (int b = 0; b < B; b++)
        for (int pg = 0; pg < max_pages_; pg++)
            pt[(size_t)b * max_pages_ + pg] = b * max_pages_ + pg;
    HIP_CHECK(hipMemcpy(page_table_, pt.data(), pt.size() * 4,
                        hipMemcpyHostToDevice));
}

void Engine::reset() {
    const int B = cfg_.batch;
    HIP_CHECK(hipMemset(n_past_, 0, B * 4));
    HIP_CHECK(hipMemset(gen_count_, 0, B * 4));
    HIP_CHECK(hipMemset(cur_ids_, 0, B * 4));
    HIP_CHECK(hipMemset(slot_active_, 1, B));
    HIP_CHECK(hipDeviceSynchronize());
}

static bool is_quant_dt(DT t) {
    return t == DT::DQ4K || t == DT::DQ6K || t == DT::DQ8;
}

// Activation-quant scratch for the i8 GEMM path (engine-owned buffers).
struct QBufs {
    int8_t* xq = nullptr;
    float* xsc = nullptr;   // interleaved [K/32][2][M4] scales/sums
};

// Launch a projection through the GEMM path: pre-zero C when split-K
// accumulation is in play (see launch_gemm), then one launch per part.
// With qb set (cfg.act_q8), quantized-weight parts run the int8-MFMA
// kernel (gemm_i8.hip): activations are block-quantized once here (silu
// fused into the quantizer for the down projection), float parts and
// unsupported shapes fall back to the bf16-staging kernel.
static bool proj_i8_ok(const Proj& p, int M) {
    bool ok = false;
    for (auto& pt : p.parts)
        ok |= gemm_i8_supported(pt.w.dtype, M, (int)pt.w.k);
    return ok;
}

static void gemm_proj(const Proj& p, const float* X, const float* res,
                      float* C, int M, hipStream_t s,
                      bool pre_zeroed = false, const float* X2 = nullptr,
                      int ldx = -1, bool xsilu = false,
                      const QBufs* qb = nullptr, bool pre_quant = false) {
    bool zero = false;
    for (auto& pt : p.parts)
        zero |= gemm_uses_splitk((int)pt.w.n, (int)pt.w.k, M);
    if (zero && !pre_zeroed)
        HIP_CHECK(hipMemsetAsync(C, 0, (size_t)M * p.n_total * 4, s));
    const int Kq = (int)p.parts[0].w.k;
    bool any_i8 = false;
    if (qb)
        for (auto& pt : p.parts)
            any_i8 |= gemm_i8_supported(pt.w.dtype, M, (int)pt.w.k);
    if (any_i8 && !pre_quant)
        launch_quant_rows(X, qb->xq, qb->xsc, M, Kq,
                          ldx < 0 ? Kq : ldx, xsilu ? 1 : 0, s);
    for (auto& pt : p.parts) {
        if (qb && gemm_i8_supported(pt.w.dtype, M, (int)pt.w.k)) {
            launch_gemm_i8(pt.w, qb->xq, qb->xsc, Kq,
                           res ? res + pt.row_off : nullptr,
                           C + pt.row_off, M, (int)p.n_total, s);
        } else {
            launch_gemm_ex(pt.w, X, X2, ldx < 0 ? (int)pt.w.k : ldx, xsilu,
                           res ? res + pt.row_off : nullptr,
                           C + pt.row_off, M, (int)p.n_total, s);
        }
    }
}

// GEMV dispatch honoring the act_q8 config for quantized weights.
static void gemv_pick(bool act_q8, const WTensor& w, int pre,
                      const float* xin, const float* gw, const float* res,
                      float* y, int B, int ldy, float eps, hipStream_t s) {
    if (act_q8 && is_quant_dt(w.dtype))
        launch_gemv_q8(w, pre, xin, gw, res, y, B, ldy, eps, s);
    else
        launch_gemv(w, pre, xin, gw, res, y, B, ldy, eps, s);
}

void Engine::reset_slot(int slot) {
    // async on the engine stream: ordered behind any in-flight decode work,
    // no host serialization (round-1 advisor: per-stride sync memcpys here
    // serialized the batcher thread against the stream).
    HIP_CHECK(hipMemsetAsync(n_past_ + slot, 0, 4, stream_));
    HIP_CHECK(hipMemsetAsync(gen_count_ + slot, 0, 4, stream_));
    HIP_CHECK(hipMemsetAsync(cur_ids_ + slot, 0, 4, stream_));
}

// Park / unpark a slot: idle slots do not advance n_past during shared
// decode steps (k_argmax_final gate), so the attention scan of an idle
// slot stays at length 1 instead of creeping with every stride.
void Engine::set_slot_active(int slot, bool active) {
    HIP_CHECK(hipMemsetAsync(slot_active_ + slot, active ? 1 : 0, 1, stream_));
}

void Engine::step(hipStream_t s) {
    const int B = cfg_.batch;
    const int NH = meta_.heads_l, NKV = meta_.kv_heads_l, D = meta_.head_dim;
    const float scale = 1.0f / std::sqrt((float)D);
    const float eps = meta_.rms_eps;
    const bool tp = cfg_.tp_size > 1;
    const bool r0 = cfg_.tp_rank == 0;
    auto allreduce = [&](const float* send, float* recv, size_t n) {
        NCCL_CHECK(ncclAllReduce(send, recv, n, ncclFloat, ncclSum,
                                 (ncclComm_t)comm_, s));
    };

    launch_embed(embed_, cur_ids_, x_, B, s);
    // B=1 always; B=2 when the register-x GEMV covers the projections
    // (quant weights, hidden a single 2048/4096 stripe): its BB=2 form
    // shares one weight stream between both rows, beating the split-K
    // GEMM that previously won at B=2 (~440 tok/s agg)
    const bool gemv_path =
        B <= 1 ||
        (B == 2 && is_quant_dt(layers_[0].qkv.parts[0].w.dtype) &&
         (meta_.hidden == 2048 || meta_.hidden == 4096));
    const QBufs qbufs{xq_, xsc_};
    const QBufs* qb = cfg_.act_q8 ? &qbufs : nullptr;
    int li = 0;
    for (auto& L : layers_) {
        uint16_t* kv_layer = kv_pool_ + (int64_t)li * layer_stride_;
        li++;
        if (gemv_path) {
            for (auto& pt : L.qkv.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_RMS, x_, L.attn_norm, nullptr,
                            qkv_ + pt.row_off, B, (int)L.qkv.n_total, eps, s);
        } else {
            // 3-buffer rotation: in --(+attn)--> mid --(+ffn)--> out, with
            // one fused clear of every split-K destination per layer
            // (replaces 3-4 ~5 us hipMemsetAsync dispatches).
            float* lin = ((li - 1) & 1) ? x2_ : x_;  // li already advanced
            float* lout = ((li - 1) & 1) ? x_ : x2_;
            const bool fq = qb && proj_i8_ok(L.qkv, B);
            launch_layer_prep(lin, L.attn_norm, xn_, B, meta_.hidden, eps,
                              qkv_, (int64_t)B * (NH + 2 * NKV) * D,
                              gu_, (int64_t)B * 2 * meta_.ffn_l,
                              x3_, (int64_t)B * meta_.hidden,
                              lout, (int64_t)B * meta_.hidden,
                              fq ? xq_ : nullptr, fq ? xsc_ : nullptr, s);
            gemm_proj(L.qkv, xn_, nullptr, qkv_, B, s, /*pre_zeroed=*/true,
                      nullptr, -1, false, qb, /*pre_quant=*/fq);
        }
        const bool fqo = !gemv_path && qb && proj_i8_ok(L.o, B);
        launch_attn_decode(qkv_, inv_freq_, page_table_, kv_layer, n_past_,
                           part_o_, part_ml_, attn_tickets_, attn_out_,
                           fqo ? xq_ : nullptr, fqo ? xsc_ : nullptr, B, NH,
                           NKV, D, attn_splits_, cfg_.page_size, max_pages_,
                           page_stride_, scale, s);
        if (gemv_path && !tp) {
            for (auto& pt : L.o.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_NONE, attn_out_, nullptr, x_ + pt.row_off,
                            x_ + pt.row_off, B, (int)L.o.n_total, eps, s);
            for (auto& pt : L.gate_up.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_RMS, x_, L.ffn_norm, nullptr,
                            gu_ + pt.row_off, B, (int)L.gate_up.n_total, eps, s);
            for (auto& pt : L.down.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_SILU, gu_, nullptr, x_ + pt.row_off,
                            x_ + pt.row_off, B, (int)L.down.n_total, eps, s);
        } else if (gemv_path) {
            // TP: local partial -> all-reduce; rank 0 folds the residual so
            // the summed result is residual + sum(partials) on every rank.
            for (auto& pt : L.o.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_NONE, attn_out_, nullptr,
                            r0 ? x_ + pt.row_off : nullptr, tmp_h_ + pt.row_off,
                            B, (int)L.o.n_total, eps, s);
            allreduce(tmp_h_, x2_, (size_t)B * meta_.hidden);
            for (auto& pt : L.gate_up.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_RMS, x2_, L.ffn_norm, nullptr,
                            gu_ + pt.row_off, B, (int)L.gate_up.n_total, eps, s);
            for (auto& pt : L.down.parts)
                gemv_pick(cfg_.gemv_q8, pt.w, PRE_SILU, gu_, nullptr,
                            r0 ? x2_ + pt.row_off : nullptr, tmp_h_ + pt.row_off,
                            B, (int)L.down.n_total, eps, s);
            allreduce(tmp_h_, x_, (size_t)B * meta_.hidden);
        } else {
            float* lin = ((li - 1) & 1) ? x2_ : x_;
            float* lout = ((li - 1) & 1) ? x_ : x2_;
            if (tp) {
                gemm_proj(L.o, attn_out_, r0 ? lin : nullptr, tmp_h_, B, s,
                          false, nullptr, -1, false, qb, fqo);
            } else {
                gemm_proj(L.o, attn_out_, lin, x3_, B, s, /*pre_zeroed=*/true,
                          nullptr, -1, false, qb, fqo);
            }
            if (tp) allreduce(tmp_h_, x3_, (size_t)B * meta_.hidden);
            const bool fqg = qb && proj_i8_ok(L.gate_up, B);
            if (fqg)
                launch_rmsnorm_rows_q(x3_, L.ffn_norm, xn_, B, meta_.hidden,
                                      eps, xq_, xsc_, s);
            else
                launch_rmsnorm_rows(x3_, L.ffn_norm, xn_, B, meta_.hidden,
                                    eps, s);
            gemm_proj(L.gate_up, xn_, nullptr, gu_, B, s, /*pre_zeroed=*/true,
                      nullptr, -1, false, qb, fqg);
            // silu fused into the down GEMM's X staging (gate | up halves;
            // i8 path: fused into the activation quantizer instead)
            if (tp) {
                gemm_proj(L.down, gu_, r0 ? x3_ : nullptr, tmp_h_, B, s,
                          false, gu_ + meta_.ffn_l, 2 * meta_.ffn_l, true, qb);
                allreduce(tmp_h_, lout, (size_t)B * meta_.hidden);
            } else {
                gemm_proj(L.down, gu_, x3_, lout, B, s, /*pre_zeroed=*/true,
                          gu_ + meta_.ffn_l, 2 * meta_.ffn_l, true, qb);
            }
        }
    }
    // GEMM path: the final residual lands in x_ for even layer counts,
    // x2_ for odd (3-buffer rotation); the GEMV path is in-place in x_.
    float* xfinal = x_;
    if (!gemv_path && (meta_.layers & 1)) xfinal = x2_;
    // TP: each rank computes its compact [B][vocab_l] slice, ONE all-gather
    // of B*vocab_l moves every slice, then a scatter kernel restores the
    // [B][V] layout (round-1 review: B separate per-row collectives).
    float* lg = tp ? logits_tp_ : logits_;
    const int ldl = tp ? meta_.vocab_l : meta_.vocab;
    if (gemv_path) {
        gemv_pick(cfg_.gemv_q8, head_, PRE_RMS, x_, out_norm_, nullptr, lg, B,
                    ldl, eps, s);
    } else {
        const bool fqh = qb && gemm_i8_supported(head_.dtype, B,
                                                 (int)head_.k) &&
                         head_.hdr2 && head_.qs2;
        if (fqh)
            launch_rmsnorm_rows_q(xfinal, out_norm_, xn_, B, meta_.hidden,
                                  eps, xq_, xsc_, s);
        else
            launch_rmsnorm_rows(xfinal, out_norm_, xn_, B, meta_.hidden,
                                eps, s);
        bool zero = gemm_uses_splitk((int)head_.n, (int)head_.k, B);
        if (zero)
            HIP_CHECK(hipMemsetAsync(lg, 0, (size_t)B * ldl * 4, s));
        if (fqh) {
            launch_gemm_i8(head_, xq_, xsc_, (int)head_.k, nullptr,
                           lg, B, ldl, s);
        } else {
            launch_gemm(head_, xn_, nullptr, lg, B, ldl, s);
        }
    }
    if (tp) {
        NCCL_CHECK(ncclAllGather(logits_tp_, gather_tp_,
                                 (size_t)B * meta_.vocab_l, ncclFloat,
                                 (ncclComm_t)comm_, s));
        launch_scatter_logits(gather_tp_, logits_, B, meta_.vocab_l,
                              cfg_.tp_size, s);
    }
    launch_argmax(logits_, amax_val_, amax_idx_, cur_ids_, n_past_,
                  gen_tokens_, gen_count_, slot_active_, B, meta_.vocab,
                  cfg_.gen_cap, s);
}

void Engine::ensure_graph() {
    if (graph_exec_ || !cfg_.use_graph) return;
    // warm-up eager step is NOT run here; capture directly (all state is
    // device-resident, shapes static). Capture of in-graph RCCL collectives
    // (tp>1) is the finicky case SURVEY §7.3 flags: on any capture failure
    // fall back to eager stepping instead of dying.
    hipGraph_t graph = nullptr;
    try {
        // Relaxed: other engines in the process (mixed-fleet serving
        // runs one engine per model) may touch the null stream while this
        // engine captures; ThreadLocal mode made their sync memcpys fail
        // with "legacy stream would depend on a capturing stream".
        HIP_CHECK(hipStreamBeginCapture(stream_,
                                        hipStreamCaptureModeRelaxed));
        step(stream_);
        HIP_CHECK(hipStreamEndCapture(stream_, &graph));
        HIP_CHECK(hipGraphInstantiate(&graph_exec_, graph, nullptr,
                                      nullptr, 0));
        HIP_CHECK(hipGraphDestroy(graph));
    } catch (const std::exception& e) {
        if (graph) hipGraphDestroy(graph);
        hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
        hipStreamIsCapturing(stream_, &st);
        if (st != hipStreamCaptureStatusNone) {
            hipGraph_t dead = nullptr;
            hipStreamEndCapture(stream_, &dead);  // discard partial capture
            if (dead) hipGraphDestroy(dead);
        }
        graph_exec_ = nullptr;
        cfg_.use_graph = false;
        fprintf(stderr,
                "[cla engine] hipGraph capture failed (%s); decoding eagerly\n",
                e.what());
    }
}

void Engine::prefill_chunk_pass(int slot, int pos0, int m) {
    // One GEMM pass over m prompt rows (ids already in pids_).
    hipStream_t s = stream_;
    const int NH = meta_.heads_l, NKV = meta_.kv_heads_l, D = meta_.head_dim;
    const float scale = 1.0f / std::sqrt((float)D);
    const float eps = meta_.rms_eps;
    const bool tp = cfg_.tp_size > 1;
    const bool r0 = cfg_.tp_rank == 0;
    const QBufs qbufs{xq_, xsc_};
    // i8 path engages automatically for chunks <= 128 rows (same act_q8
    // quantization semantics as the decode step)
    const QBufs* qb = cfg_.act_q8 ? &qbufs : nullptr;
    launch_embed(embed_, pids_, xp_, m, s);
    int li = 0;
    for (auto& L : layers_) {
        uint16_t* kv_layer = kv_pool_ + (int64_t)li * layer_stride_;
        li++;
        launch_rmsnorm_rows(xp_, L.attn_norm, xnp_, m, meta_.hidden, eps, s);
        gemm_proj(L.qkv, xnp_, nullptr, qkvp_, m, s, false, nullptr, -1,
                  false, qb);
        launch_rope_prefill(qkvp_, inv_freq_, page_table_, kv_layer, slot,
                            pos0, m, NH, NKV, D, cfg_.page_size, max_pages_,
                            page_stride_, s);
        launch_attn_prefill(qkvp_, page_table_, kv_layer, attnp_, slot, pos0,
                            m, NH, NKV, D, cfg_.page_size, max_pages_,
                            page_stride_, scale, s);
        if (tp) {
            gemm_proj(L.o, attnp_, r0 ? xp_ : nullptr, tmp_hp_, m, s,
                      false, nullptr, -1, false, qb);
            NCCL_CHECK(ncclAllReduce(tmp_hp_, xp2_, (size_t)m * meta_.hidden,
                                     ncclFloat, ncclSum, (ncclComm_t)comm_, s));
        } else {
            gemm_proj(L.o, attnp_, xp_, xp2_, m, s, false, nullptr, -1,
                      false, qb);
        }
        launch_rmsnorm_rows(xp2_, L.ffn_norm, xnp_, m, meta_.hidden, eps, s);
        gemm_proj(L.gate_up, xnp_, nullptr, gup_, m, s, false, nullptr, -1,
                  false, qb);
        if (tp) {
            gemm_proj(L.down, gup_, r0 ? xp2_ : nullptr, tmp_hp_, m, s,
                      false, gup_ + meta_.ffn_l, 2 * meta_.ffn_l, true, qb);
            NCCL_CHECK(ncclAllReduce(tmp_hp_, xp_, (size_t)m * meta_.hidden,
                                     ncclFloat, ncclSum, (ncclComm_t)comm_, s));
        } else {
            gemm_proj(L.down, gup_, xp2_, xp_, m, s, false,
                      gup_ + meta_.ffn_l, 2 * meta_.ffn_l, true, qb);
        }
    }
}

void Engine::prefill_slot(int slot, const std::vector<int32_t>& ids) {
    if (ids.empty())
        throw std::runtime_error("prefill_slot: empty prompt");
    const int len = (int)ids.size();
    if (len >= cfg_.max_seq)
        throw std::runtime_error("prompt longer than max_seq");
    int pos0 = 0;  // prefill restarts the slot
    int done = 0;
    const float eps = meta_.rms_eps;
    while (done < len) {
        const int m = std::min(cfg_.prefill_chunk, len - done);
        HIP_CHECK(hipMemcpyAsync(pids_, ids.data() + done, (size_t)m * 4,
                                 hipMemcpyHostToDevice, stream_));
        prefill_chunk_pass(slot, pos0, m);
        HIP_CHECK(hipStreamSynchronize(stream_));
        pos0 += m;
        done += m;
    }
    // logits of the LAST prompt row -> slot's logits; sample + advance state
    const int last = ((len - 1) % cfg_.prefill_chunk);
    const int64_t voff = (int64_t)cfg_.tp_rank * meta_.vocab_l;
    gemv_pick(cfg_.gemv_q8, head_, PRE_RMS, xp_ + (size_t)last * meta_.hidden, out_norm_,
                nullptr, logits_ + (size_t)slot * meta_.vocab + voff, 1,
                meta_.vocab, eps, stream_);
    if (cfg_.tp_size > 1)
        NCCL_CHECK(ncclAllGather(
            logits_ + (size_t)slot * meta_.vocab + voff,
            logits_ + (size_t)slot * meta_.vocab, meta_.vocab_l, ncclFloat,
            (ncclComm_t)comm_, stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));
    // n_past[slot] = len - 1; argmax's advance makes it len.
    const int32_t npast = len - 1, zero = 0;
    HIP_CHECK(hipMemcpy(n_past_ + slot, &npast, 4, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(gen_count_ + slot, &zero, 4, hipMemcpyHostToDevice));
    // prefill implies the slot is (re)activated before its first argmax
    HIP_CHECK(hipMemsetAsync(slot_active_ + slot, 1, 1, stream_));
    launch_argmax(logits_ + (size_t)slot * meta_.vocab,
                  amax_val_ + (size_t)slot * 64, amax_idx_ + (size_t)slot * 64,
                  cur_ids_ + slot, n_past_ + slot,
                  gen_tokens_ + (size_t)slot * cfg_.gen_cap,
                  gen_count_ + slot, slot_active_ + slot, 1, meta_.vocab,
                  cfg_.gen_cap, stream_);
    HIP_CHECK(hipStreamSynchronize(stream_));
}

void Engine::prefill(const std::vector<int32_t>& ids, int len) {
    const int B = cfg_.batch;
    if ((int)ids.size() != B * len)
        throw std::runtime_error("prefill: ids must be batch*len");
    for (int b = 0; b < B; b++)
        prefill_slot(b, std::vector<int32_t>(ids.begin() + (size_t)b * len,
                                             ids.begin() + (size_t)(b + 1) * len));
}

void Engine::decode(int n_steps) {
    ensure_graph();
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    HIP_CHECK(hipEventRecord(e0, stream_));
    for (int i = 0; i < n_steps; i++) {
        if (graph_exec_) {
            HIP_CHECK(hipGraphLaunch(graph_exec_, stream_));
        } else {
            step(stream_);
        }
    }
    HIP_CHECK(hipEventRecord(e1, stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));
    float ms = 0;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    last_decode_ms_ = ms;
    hipEventDestroy(e0);
    hipEventDestroy(e1);
}

std::vector<int32_t> Engine::gen_tokens(int slot) {
    int32_t count = 0;
    HIP_CHECK(hipMemcpy(&count, gen_count_ + slot, 4, hipMemcpyDeviceToHost));
    count = std::min(count, cfg_.gen_cap);
    std::vector<int32_t> out(count);
    if (count)
        HIP_CHECK(hipMemcpy(out.data(), gen_tokens_ + (size_t)slot * cfg_.gen_cap,
                            count * 4, hipMemcpyDeviceToHost));
    return out;
}

std::vector<float> Engine::logits(int slot) {
    std::vector<float> out(meta_.vocab);
    HIP_CHECK(hipMemcpy(out.data(), logits_ + (size_t)slot * meta_.vocab,
                        meta_.vocab * 4, hipMemcpyDeviceToHost));
    return out;
}

void Engine::set_cur_token(int slot, int32_t id) {
    HIP_CHECK(hipMemcpy(cur_ids_ + slot, &id, 4, hipMemcpyHostToDevice));
}

int32_t Engine::cur_token(int slot) {
    int32_t id = 0;
    HIP_CHECK(hipMemcpy(&id, cur_ids_ + slot, 4, hipMemcpyDeviceToHost));
    return id;
}

std::vector<int32_t> Engine::n_past() {
    std::vector<int32_t> out(cfg_.batch);
    HIP_CHECK(hipMemcpy(out.data(), n_past_, cfg_.batch * 4,
                        hipMemcpyDeviceToHost));
    return out;
}

}  // namespace cla
