// MFMA fragment-map probe: computes C = A @ B for one 16x16x32 bf16 MFMA
// with the layout assumed by attn_prefill.hip. tests/test_ops_gpu.py checks
// it against torch.matmul with ASYMMETRIC operands (guide: a symmetric B
// passes a row/col-swapped C-write silently).
#include "common.h"

__global__ void mfma_probe_kernel(const unsigned short* A,  // [16][32]
                                  const unsigned short* B,  // [32][16]
                                  float* C) {               // [16][16]
    const int l = threadIdx.x;
    bf16x8 a, b;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        const int kk = (l >> 4) * 8 + e;
        a[e] = *reinterpret_cast<const __bf16*>(&A[(l & 15) * 32 + kk]);
        b[e] = *reinterpret_cast<const __bf16*>(&B[kk * 16 + (l & 15)]);
    }
    f32x4 c{0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

extern "C" void launch_mfma_probe(const unsigned short* A,
                                  const unsigned short* B, float* C,
                                  hipStream_t stream) {
    hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream, A, B, C);
}
