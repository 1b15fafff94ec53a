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

// Transposed header copy for the i8 GEMM's LDS-DMA scale staging
// (gemm_i8.hip): headers grouped so one k-window's headers for ALL rows
// are contiguous. DQ4K: [K/64 q-groups][N][8B pair header];
// DQ8: [K/32][N][f16 d]. Bytes per row (total = n * this):
__host__ __device__ inline int64_t dhdr2_row_bytes(DT t, int64_t k) {
    switch (t) {
        case DT::DQ4K: return k / 64 * 8;
        case DT::DQ8: return k / 32 * 2;
        default: return 0;
    }
}

// A weight matrix on device: N rows of K quantized columns.
struct WTensor {
    DT dtype = DT::F32;
    int64_t n = 0;   // rows (output features)
    int64_t k = 0;   // cols (input features)
    const void* qs = nullptr;
    const void* hdr = nullptr;
    const void* hdr2 = nullptr;  // transposed headers (quant GEMM dtypes)
};

}  // namespace cla
