// Grouped GEMM for MoE expert MLPs (CDNA4 MFMA).
//
//   out[S, N] = for each expert e: x[offs[e]:offs[e+1], :] @ w[e].T
//
// x is the token-sorted activation matrix (tokens grouped by expert), w is
// [E, N, K] in the HF [out, in] convention, offs is a device int32 [E+1]
// prefix (row partition of [0, S)). The segment sizes live ONLY on the
// device: grid and shapes are static, so the op is hipGraph-capturable and
// the host never syncs on routing counts (torch-side padded bmm needed a
// counts.max() sync per MoE layer — 32 pipeline drains per Mixtral step).
//
// Decode-shaped design (M per expert ~10^1-10^3, N/K in the thousands):
//   grid (N/64, E), one workgroup = 4 waves = 64 output columns of one
//   expert; the wave's 16-column W panel streams from HBM exactly ONCE
//   while up to 16 row-tiles (256 rows) accumulate in registers; x K-tiles
//   are staged cooperatively in LDS (padded rows -> conflict-free
//   ds_read_b128) and reused by all 4 waves. Experts with more than 256
//   rows loop in 256-row passes (W is re-read per pass; at decode batches
//   a pass covers everything).
//
// Numerics reference: ops/reference.py grouped_gemm.
#include "common.h"

#define GG_BLOCK 256
#define GG_WAVES 4
#define GG_BN 64          // output cols per workgroup (16 per wave)
#define GG_MROWS 256      // rows per pass (16 m-tiles)
#define GG_MT (GG_MROWS / 16)
#define GG_BK 32          // K step (one mfma_16x16x32)
#define GG_XPAD 8         // LDS row padding (bf16 elems)

__global__ __launch_bounds__(GG_BLOCK) void grouped_gemm_kernel(
    const unsigned short* __restrict__ x,   // [S, K]
    const unsigned short* __restrict__ w,   // [E, N, K]
    const int* __restrict__ offs,           // [E+1]
    unsigned short* __restrict__ out,       // [S, N]
    int N, int K) {
    const int e = blockIdx.y;
    const int n0 = blockIdx.x * GG_BN;
    const int m_lo = offs[e];
    const int m_hi = offs[e + 1];
    if (m_lo >= m_hi) return;
    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wid = tid / WAVE;
    const int lg = lane >> 4;
    const int li = lane & 15;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    // x K-tile: [GG_MROWS][GG_BK + GG_XPAD] bf16 (80 B rows -> the 16-lane
    // ds_read_b128 groups land on distinct banks)
    unsigned short* x_s = reinterpret_cast<unsigned short*>(smem_raw);
    constexpr int XS = GG_BK + GG_XPAD;

    const unsigned short* wp = w + ((long)e * N + n0 + wid * 16 + li) * K;

    for (int pass = m_lo; pass < m_hi; pass += GG_MROWS) {
        const int m_cnt = min(GG_MROWS, m_hi - pass);
        const int n_mt = (m_cnt + 15) / 16;

        f32x4 acc[GG_MT];
#pragma unroll
        for (int t = 0; t < GG_MT; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

        for (int k0 = 0; k0 < K; k0 += GG_BK) {
            __syncthreads();
            // stage the x K-tile: thread i covers row i/2, dims (i%2)*16
            for (int i = tid; i < GG_MROWS * 2; i += GG_BLOCK) {
                const int r = i / 2;
                const int d = (i % 2) * 16;
                short8 v{};
                if (r < m_cnt)
                    v = *reinterpret_cast<const short8*>(
                        x + (long)(pass + r) * K + k0 + d);
                *reinterpret_cast<short8*>(x_s + r * XS + d) = v;
                *reinterpret_cast<short8*>(x_s + r * XS + d + 8) =
                    (r < m_cnt)
                        ? *reinterpret_cast<const short8*>(
                              x + (long)(pass + r) * K + k0 + d + 8)
                        : short8{};
            }
            __syncthreads();

            // B fragment: this wave's 16 W rows (= output cols), k-seg
            const bf16x8 b =
                *reinterpret_cast<const bf16x8*>(wp + k0 + lg * 8);
            // static unroll with a guard: a runtime-indexed acc[t] would be
            // demoted to scratch (register arrays need constant indices)
#pragma unroll
            for (int t = 0; t < GG_MT; ++t) {
                if (t < n_mt) {
                    const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                        x_s + (t * 16 + li) * XS + lg * 8);
                    acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        a, b, acc[t], 0, 0, 0);
                }
            }
        }

        // epilogue: C row = lg*4 + r, col = li
#pragma unroll
        for (int t = 0; t < GG_MT; ++t) {
            if (t >= n_mt) continue;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = t * 16 + lg * 4 + r;
                if (row >= m_cnt) continue;
                out[(long)(pass + row) * N + n0 + wid * 16 + li] =
                    f2bf(acc[t][r]);
            }
        }
    }
}

extern "C" void launch_grouped_gemm(
    const unsigned short* x, const unsigned short* w, const int* offs,
    unsigned short* out, int E, int S, int N, int K, hipStream_t stream) {
    dim3 grid(N / GG_BN, E);
    const int smem = GG_MROWS * (GG_BK + GG_XPAD) * 2;
    hipLaunchKernelGGL(grouped_gemm_kernel, grid, dim3(GG_BLOCK), smem,
                       stream, x, w, offs, out, N, K);
}
