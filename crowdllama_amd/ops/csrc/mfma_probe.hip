// MFMA fragment-layout probe (test-only): one wave computes
// C[16][16] = A[16][32] x B[32][16] with v_mfma_f32_16x16x32_bf16 under the
// assumed lane mappings:
//   A: lane l holds row (l&15), k = (l>>4)*8 + i  (8 bf16)
//   B: lane l holds col (l&15), k = (l>>4)*8 + i  (8 bf16)
//   C: lane l holds col (l&15), rows (l>>4)*4 + r (4 f32)
// tests/test_gpu_kernels.py diffs it against numpy (asymmetric inputs, so a
// transposed mapping cannot pass).
#include "common.h"

namespace cla {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(64) void k_mfma_probe(
    const uint16_t* __restrict__ A,   // [16][32] bf16 row-major
    const uint16_t* __restrict__ B,   // [32][16] bf16 row-major
    float* __restrict__ C) {          // [16][16] f32 row-major
    const int lane = threadIdx.x & 63;
    const int half = lane >> 4;       // 0..3 (k-groups of 8)
    const int idx = lane & 15;
    bf16x8 a, b;
    #pragma unroll
    for (int i = 0; i < 8; i++) {
        const int k = half * 8 + i;
        uint16_t av = A[idx * 32 + k];      // A[row=idx][k]
        uint16_t bv = B[k * 16 + idx];      // B[k][col=idx]
        a[i] = *reinterpret_cast<__bf16*>(&av);
        b[i] = *reinterpret_cast<__bf16*>(&bv);
    }
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    #pragma unroll
    for (int r = 0; r < 4; r++) {
        const int row = half * 4 + r;
        C[row * 16 + idx] = acc[r];
    }
}

void launch_mfma_probe_test(const uint16_t* A, const uint16_t* B, float* C) {
    void *dA, *dB, *dC;
    HIP_CHECK(hipMalloc(&dA, 16 * 32 * 2));
    HIP_CHECK(hipMalloc(&dB, 32 * 16 * 2));
    HIP_CHECK(hipMalloc(&dC, 16 * 16 * 4));
    HIP_CHECK(hipMemcpy(dA, A, 16 * 32 * 2, hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dB, B, 32 * 16 * 2, hipMemcpyHostToDevice));
    hipLaunchKernelGGL(k_mfma_probe, dim3(1), dim3(64), 0, nullptr,
                       (const uint16_t*)dA, (const uint16_t*)dB, (float*)dC);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipMemcpy(C, dC, 16 * 16 * 4, hipMemcpyDeviceToHost));
    hipFree(dA); hipFree(dB); hipFree(dC);
}

}  // namespace cla
