// MI355X-native decode engine: owns device weights, paged KV cache, and the
// per-token decode step (hipGraph-captured). This is the worker-side compute
// that replaces the reference's shell-out to Ollama (reference
// pkg/crowdllama/api.go:45-160 -> Ollama -> llama.cpp; here it is first-party).
#pragma once

#include <hip/hip_runtime.h>

#include <memory>
#include <string>
#include <vector>

#include "common.h"

namespace cla {

class GGUFFile;

struct EngineConfig {
    int batch = 1;          // decode slots
    int max_seq = 4096;     // max positions per slot
    int page_size = 64;     // KV page tokens
    int gen_cap = 2048;     // generated-token ring per slot
    int device = 0;
    bool use_graph = true;
    int prefill_chunk = 1024;  // prompt tokens per prefill GEMM pass
    // tensor parallelism over RCCL/xGMI (capability extension; the
    // reference has no collectives at all — SURVEY.md §2.3)
    int tp_rank = 0;
    int tp_size = 1;
    // int8-quantized activations (per-32 symmetric; semantics replicated
    // by ref_numpy(act_q8=True)) for quantized-weight GEMMs — the
    // v_mfma_i32 batched-decode path (gemm_i8.hip). Default ON: measured
    // +33% at B=16 over the bf16-staging GEMM (round 2). Exact-f32
    // comparisons (tests) set this false.
    bool act_q8 = true;
    // int8-activation GEMV (v_dot4) for the B=1 decode path. Off: measured
    // slower than the f32 GEMV (273 vs 292 tok/s llama3-8b B=1) even with
    // lane-parallel staging — B=1 is bandwidth-, not issue-bound.
    bool gemv_q8 = false;
    std::string nccl_id;       // ncclUniqueId bytes (rank 0 creates)
};

struct ModelMeta {
    std::string name;
    int vocab = 0, hidden = 0, layers = 0, heads = 0, kv_heads = 0;
    int ffn = 0, head_dim = 0, max_ctx = 0;
    float rope_theta = 10000.f, rms_eps = 1e-5f;
    // per-rank geometry under tensor parallelism (== globals at tp=1)
    int heads_l = 0, kv_heads_l = 0, ffn_l = 0, vocab_l = 0;
};

// One logical projection: a list of row-blocks (merged when dtypes match).
struct Proj {
    struct Part {
        WTensor w;
        int64_t row_off = 0;  // offset into the output vector
    };
    std::vector<Part> parts;
    int64_t n_total = 0;
};

struct Layer {
    const float* attn_norm = nullptr;
    const float* ffn_norm = nullptr;
    Proj qkv, o, gate_up, down;
};

class Engine {
public:
    Engine(const std::string& gguf_path, const EngineConfig& cfg);
    ~Engine();
    Engine(const Engine&) = delete;

    const ModelMeta& meta() const { return meta_; }
    const EngineConfig& config() const { return cfg_; }

    // Reset all slots (empty KV).
    void reset();
    // Reset one slot (serving: reclaim a finished/idle slot).
    void reset_slot(int slot);
    // Park / unpark a slot: parked slots do not advance during decode.
    void set_slot_active(int slot, bool active);
    // Feed prompt tokens (same length for all slots in this call) through
    // the GEMM prefill path; afterwards each slot's first generated token is
    // in gen_tokens[slot][0]. ids is [batch][len] row-major.
    void prefill(const std::vector<int32_t>& ids, int len);
    // Prefill one slot (chunked GEMM path); used for serving.
    void prefill_slot(int slot, const std::vector<int32_t>& ids);
    // Run n decode steps back-to-back (graph replays; one sync at the end).
    void decode(int n_steps);
    // Fetch generated tokens for a slot (gen_count entries).
    std::vector<int32_t> gen_tokens(int slot);
    // Copy a slot's current logits to host (for host-side sampling).
    std::vector<float> logits(int slot);
    // Override a slot's current input token (host-side sampling path).
    void set_cur_token(int slot, int32_t id);
    int32_t cur_token(int slot);
    std::vector<int32_t> n_past();

    size_t vram_bytes() const { return vram_bytes_; }
    double last_decode_ms() const { return last_decode_ms_; }

private:
    void load_weights(const GGUFFile& gf);
    Proj load_proj(const GGUFFile& gf, const std::vector<std::string>& names,
                   bool row_shard = false, int64_t c0 = -1, int64_t c1 = -1);
    WTensor upload_tensor(const GGUFFile& gf, const std::string& name);
    WTensor upload_shard(const GGUFFile& gf, const std::string& name,
                         int64_t r0, int64_t r1, int64_t c0, int64_t c1);
    WTensor upload_pack(const void* qs, size_t qs_bytes, const void* hdr,
                        size_t hdr_bytes, DT dtype, int64_t rows, int64_t k);
    const float* upload_norm(const GGUFFile& gf, const std::string& name);
    void alloc_state();
    void step(hipStream_t stream);
    void ensure_graph();

    EngineConfig cfg_;
    ModelMeta meta_;
    hipStream_t stream_ = nullptr;

    // weights
    std::vector<Layer> layers_;
    WTensor embed_, head_;
    const float* out_norm_ = nullptr;
    std::vector<void*> allocs_;
    size_t vram_bytes_ = 0;

    // state buffers (device)
    void prefill_chunk_pass(int slot, int pos0, int m);
    float* x_ = nullptr;        // [B][h]
    float* xn_ = nullptr;       // [B][h] (GEMM decode path rmsnorm out)
    float* x2_ = nullptr;       // [B][h] residual ping-pong (split-K GEMM)
    float* x3_ = nullptr;       // [B][h] mid buffer (post-attention)
    // prefill scratch ([Mchunk] rows)
    float* xp_ = nullptr;
    float* xp2_ = nullptr;
    float* xnp_ = nullptr;
    float* qkvp_ = nullptr;
    float* attnp_ = nullptr;
    float* gup_ = nullptr;
    int32_t* pids_ = nullptr;   // prefill token ids
    float* qkv_ = nullptr;      // [B][(NH+2KV)*D]
    float* attn_out_ = nullptr; // [B][NH*D]
    float* gu_ = nullptr;       // [B][2F]
    float* logits_ = nullptr;   // [B][V]
    float* part_o_ = nullptr;   // [B][NH][S][D]
    float* part_ml_ = nullptr;  // [B][NH][S][2]
    int32_t* attn_tickets_ = nullptr;  // [B][NKV] fan-in counters
    float* amax_val_ = nullptr; // [B][64]
    int32_t* amax_idx_ = nullptr;
    float* inv_freq_ = nullptr; // [D/2]
    int32_t* cur_ids_ = nullptr;   // [B]
    int32_t* n_past_ = nullptr;    // [B]
    int32_t* gen_tokens_ = nullptr;  // [B][gen_cap]
    int32_t* gen_count_ = nullptr;   // [B]
    int32_t* page_table_ = nullptr;  // [B][max_pages]
    uint8_t* slot_active_ = nullptr; // [B] decode-advance gate
    uint16_t* kv_pool_ = nullptr;
    int attn_splits_ = 16;
    int max_pages_ = 0;
    int64_t page_stride_ = 0;   // bytes-in-elements per page (all layers)
    int64_t layer_stride_ = 0;  // per-layer offset within a page

    hipGraphExec_t graph_exec_ = nullptr;
    double last_decode_ms_ = 0.0;
    void* comm_ = nullptr;      // ncclComm_t when tp_size > 1
    float* tmp_h_ = nullptr;    // [B][hidden] all-reduce staging
    float* tmp_hp_ = nullptr;   // [Mchunk][hidden] prefill all-reduce staging
    float* logits_tp_ = nullptr;  // [B][vocab_l] local slice (tp>1)
    float* gather_tp_ = nullptr;  // [tp][B][vocab_l] all-gather recv (tp>1)
    int8_t* xq_ = nullptr;        // [maxM][maxK] i8-GEMM activation quant
    float* xsc_ = nullptr;        // [maxK/32][2][M4] scales + dx*sum(qx)
};

}  // namespace cla
