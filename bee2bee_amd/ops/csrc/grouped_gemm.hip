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
//   while up to 8 row-tiles (128 rows) accumulate in registers. A-fragments
//   read straight from global memory — x is a few MB and L2-resident, so
//   staging it through LDS only added two barriers per 32-deep K-step
//   (measured 2x slower than the barrier-free form at decode sizes).
//   Experts with more rows loop in 128-row passes (W re-read per pass).
//
// Numerics reference: ops/reference.py grouped_gemm.
#include "common.h"

#define GG_BLOCK 256
#define GG_WAVES 4
#define GG_BN 64          // output cols per workgroup (16 per wave)
#define GG_MROWS 64       // rows per pass (VGPR-bound: 128 rows -> 252 VGPRs)
#define GG_MT (GG_MROWS / 16)
#define GG_BK 32          // K step (one mfma_16x16x32)
#define GG_KG 128         // K-group staged per barrier pair (4 K-steps)
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

    const unsigned short* wp = w + ((long)e * N + n0 + wid * 16 + li) * K;
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* x_s = reinterpret_cast<unsigned short*>(smem_raw);
    constexpr int XS = GG_KG + GG_XPAD;  // 272 B rows: conflict-free b128

    for (int pass = m_lo; pass < m_hi; pass += GG_MROWS) {
        const int m_cnt = min(GG_MROWS, m_hi - pass);
        const int n_mt = (m_cnt + 15) / 16;

        f32x4 acc[GG_MT];
#pragma unroll
        for (int t = 0; t < GG_MT; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

        // K-groups of 128 dims: the x tile stages cooperatively into LDS
        // (8 independent 16 B loads in flight per thread), the wave's B
        // panel loads 4 fragments per group — one barrier pair per 32
        // MFMAs. (Register-resident A hoisted to 250 VGPRs; per-step LDS
        // staging was barrier-bound: this is the middle ground.)
        for (int k0 = 0; k0 < K; k0 += GG_KG) {
            __syncthreads();
            for (int i = tid; i < GG_MROWS * (GG_KG / 16); i += GG_BLOCK) {
                const int r = i / (GG_KG / 16);
                const int d = (i % (GG_KG / 16)) * 16;
                short8 v0{}, v1{};
                if (r < m_cnt) {
                    const unsigned short* src =
                        x + (long)(pass + r) * K + k0 + d;
                    v0 = *reinterpret_cast<const short8*>(src);
                    v1 = *reinterpret_cast<const short8*>(src + 8);
                }
                *reinterpret_cast<short8*>(x_s + r * XS + d) = v0;
                *reinterpret_cast<short8*>(x_s + r * XS + d + 8) = v1;
            }
            bf16x8 bg[GG_KG / GG_BK];
#pragma unroll
            for (int g = 0; g < GG_KG / GG_BK; ++g)
                bg[g] = *reinterpret_cast<const bf16x8*>(
                    wp + k0 + g * GG_BK + lg * 8);
            __syncthreads();
#pragma unroll
            for (int g = 0; g < GG_KG / GG_BK; ++g) {
#pragma unroll
                for (int t = 0; t < GG_MT; ++t) {
                    if (t < n_mt) {
                        const bf16x8 a = *reinterpret_cast<const bf16x8*>(
                            x_s + (t * 16 + li) * XS + g * GG_BK + lg * 8);
                        acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                            a, bg[g], acc[t], 0, 0, 0);
                    }
                }
                // fence the scheduler per K-step: without it all 32 LDS
                // A-fragments are hoisted live -> 252 VGPRs, 1 wave/SIMD
                __builtin_amdgcn_sched_barrier(0);
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
    const int smem = GG_MROWS * (GG_KG + GG_XPAD) * 2;
    hipLaunchKernelGGL(grouped_gemm_kernel, grid, dim3(GG_BLOCK), smem,
                       stream, x, w, offs, out, N, K);
}
