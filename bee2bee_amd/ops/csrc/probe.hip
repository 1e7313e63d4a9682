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

// fp8 (OCP e4m3) variant of the layout probe: assumes the SAME index map
// as the bf16 op (lane -> row/col, k = (l>>4)*8 + e over 8 fp8 BYTES) —
// tests/test_ops_gpu.py::test_mfma_fp8_fragment_map verifies it on HW
// before any fp8-MFMA kernel relies on it.
__global__ void mfma_probe_fp8_kernel(const unsigned char* A,  // [16][32]
                                      const unsigned char* B,  // [32][16]
                                      float* C) {              // [16][16]
    const int l = threadIdx.x;
    uchar8 a, b;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
        const int kk = (l >> 4) * 8 + e;
        a[e] = A[(l & 15) * 32 + kk];
        b[e] = B[kk * 16 + (l & 15)];
    }
    f32x4 c{0.f, 0.f, 0.f, 0.f};
    c = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
        *reinterpret_cast<const long*>(&a),
        *reinterpret_cast<const long*>(&b), c, 0, 0, 0);
#pragma unroll
    for (int r = 0; r < 4; ++r) C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

extern "C" void launch_mfma_probe_fp8(const unsigned char* A,
                                      const unsigned char* B, float* C,
                                      hipStream_t stream) {
    hipLaunchKernelGGL(mfma_probe_fp8_kernel, dim3(1), dim3(64), 0, stream,
                       A, B, C);
}

// ---------------------------------------------------------------------------
// Bandwidth probes over a bf16 pool — used to locate the decode-attention
// bandwidth ceiling (scripts/bench_attn.py --probe). Each accumulates a
// checksum so loads cannot be DCE'd.

// (a) linear roofline: grid-stride dwordx4
__global__ void bw_linear_kernel(const unsigned short* __restrict__ src,
                                 long n_elems, float* __restrict__ sink) {
    float acc = 0.f;
    const long stride = (long)gridDim.x * blockDim.x * 8;
    for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
         i < n_elems; i += stride) {
        short8 v = *reinterpret_cast<const short8*>(src + i);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc += (float)(short)v[j];
    }
    if (acc == 1234.5678f) sink[0] = acc;  // never true; keeps loads live
}

// (b) phase-1 pattern: lane owns one 256 B row, reads it in 16 B pieces.
// rows_per_block keys per workgroup (4 waves x 64), row stride = hd elems.
__global__ void bw_rowperlane_kernel(const unsigned short* __restrict__ src,
                                     long n_rows, int hd, int batch8,
                                     float* __restrict__ sink) {
    const long row = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= n_rows) return;
    const unsigned short* r = src + row * hd;
    float acc = 0.f;
    if (batch8) {
        for (int d = 0; d + 64 <= hd; d += 64) {
            short8 raw[8];
#pragma unroll
            for (int ii = 0; ii < 8; ++ii)
                raw[ii] = *reinterpret_cast<const short8*>(r + d + ii * 8);
#pragma unroll
            for (int ii = 0; ii < 8; ++ii)
#pragma unroll
                for (int j = 0; j < 8; ++j) acc += (float)(short)raw[ii][j];
        }
    } else {
        for (int d = 0; d < hd; d += 8) {
            short8 v = *reinterpret_cast<const short8*>(r + d);
#pragma unroll
            for (int j = 0; j < 8; ++j) acc += (float)(short)v[j];
        }
    }
    if (acc == 1234.5678f) sink[0] = acc;
}

// (c) phase-2 pattern: whole wave reads one row per instruction (4 B/lane),
// iterating rows in batches of 8 independent loads.
__global__ void bw_rowperinstr_kernel(const unsigned short* __restrict__ src,
                                      long n_rows, int hd, int rows_per_wg,
                                      float* __restrict__ sink) {
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;
    const int waves = blockDim.x / WAVE;
    const long base = (long)blockIdx.x * rows_per_wg;
    const int d0 = (lane * 2) % hd;
    float a0 = 0.f, a1 = 0.f;
    for (long t0 = base + wid * 8; t0 < base + rows_per_wg;
         t0 += (long)waves * 8) {
        short2v vv[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            const long row = (t0 + j < n_rows) ? t0 + j : 0;
            vv[j] = *reinterpret_cast<const short2v*>(src + row * hd + d0);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
            a0 += (float)(short)vv[j][0];
            a1 += (float)(short)vv[j][1];
        }
    }
    if (a0 + a1 == 1234.5678f) sink[0] = a0;
}

// (d) paged variant of (b): rows resolved through a block table like the
// KV cache (16 KB pages scattered over the pool)
__global__ void bw_paged_kernel(const unsigned short* __restrict__ src,
                                const int* __restrict__ table, long n_rows,
                                int hd, int rows_per_page,
                                float* __restrict__ sink) {
    const long row = (long)blockIdx.x * blockDim.x + threadIdx.x;
    if (row >= n_rows) return;
    const long page = table[row / rows_per_page];
    const unsigned short* r =
        src + (page * rows_per_page + row % rows_per_page) * (long)hd;
    float acc = 0.f;
    for (int d = 0; d + 64 <= hd; d += 64) {
        short8 raw[8];
#pragma unroll
        for (int ii = 0; ii < 8; ++ii)
            raw[ii] = *reinterpret_cast<const short8*>(r + d + ii * 8);
#pragma unroll
        for (int ii = 0; ii < 8; ++ii)
#pragma unroll
            for (int j = 0; j < 8; ++j) acc += (float)(short)raw[ii][j];
    }
    if (acc == 1234.5678f) sink[0] = acc;
}

extern "C" {
void launch_bw_paged(const unsigned short* src, const int* table, long n_rows,
                     int hd, int rows_per_page, float* sink, hipStream_t st) {
    long blocks = (n_rows + 255) / 256;
    hipLaunchKernelGGL(bw_paged_kernel, dim3(blocks), dim3(256), 0, st, src,
                       table, n_rows, hd, rows_per_page, sink);
}
void launch_bw_linear(const unsigned short* src, long n, float* sink,
                      hipStream_t st) {
    long blocks = (n / 8 + 255) / 256;
    if (blocks > 2048) blocks = 2048;
    hipLaunchKernelGGL(bw_linear_kernel, dim3(blocks), dim3(256), 0, st, src,
                       n, sink);
}
void launch_bw_rowperlane(const unsigned short* src, long n_rows, int hd,
                          int batch8, float* sink, hipStream_t st) {
    long blocks = (n_rows + 255) / 256;
    hipLaunchKernelGGL(bw_rowperlane_kernel, dim3(blocks), dim3(256), 0, st,
                       src, n_rows, hd, batch8, sink);
}
void launch_bw_rowperinstr(const unsigned short* src, long n_rows, int hd,
                           int rows_per_wg, float* sink, hipStream_t st) {
    long blocks = (n_rows + rows_per_wg - 1) / rows_per_wg;
    hipLaunchKernelGGL(bw_rowperinstr_kernel, dim3(blocks), dim3(256), 0, st,
                       src, n_rows, hd, rows_per_wg, sink);
}
}
