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
// Decode-shaped design (M per expert ~10^1-10^3, N/K in the thousands —
// W streaming from HBM is the bound: Mixtral reads 2.8 GB of expert
// weights per MoE layer, so the target is "W once, at HBM rate"):
//   grid (N/64, E); one workgroup = 4 waves = 64 output columns of one
//   expert, accumulating up to GG_MROWS=128 rows so one pass covers the
//   whole segment at decode sizes (the previous 64-row pass re-read W
//   twice at S=1024/E=8 — measured 1.7 TB/s effective).
//   The inner loop is the T14 software pipeline proved out in
//   attn_pv_mfma_kernel (attn_decode.hip): double-buffered x tiles in LDS
//   with register prefetch TWO K-groups ahead, W fragments double-buffered
//   in registers, SET/BUF as LITERALS so the compiler tracks exactly which
//   outstanding loads each write waits on (runtime-indexed sets forced
//   vmcnt(0) per write in the PV kernel: measured 2.1 TB/s vs 4.1).
//   Experts with more rows loop in 128-row passes (W re-read per pass —
//   fine: at M>=128 per pass the arithmetic intensity makes each pass
//   compute-, not W-, bound).
//
// Numerics reference: ops/reference.py grouped_gemm.
#include "common.h"

#define GG_BLOCK 512
#define GG_WAVES 8
#define GG_BN 64          // output cols per workgroup (16 per col-group)
#define GG_MROWS 128      // rows per pass (8 MFMA row-tiles)
#define GG_MT (GG_MROWS / 16)
#define GG_WMT (GG_MT / 2)  // row-tiles per wave (rows split across 2
                            // wave-halves: halves the per-wave registers ->
                            // 2 blocks/CU instead of 1; the duplicate W
                            // fragment loads of a row-half pair hit L2)
#define GG_BK 32          // K step (one mfma_16x16x32)
#define GG_KG 64          // K-group per pipeline stage (2 K-steps)
#define GG_XPAD 8         // LDS row padding (bf16 elems): 272 B rows keep
                          // b128 reads conflict-free at the 64-dword modulus

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

    const int colg = wid & 3;   // wave's 16-col group
    const int rowh = wid >> 2;  // wave's 64-row half
    const unsigned short* wp = w + ((long)e * N + n0 + colg * 16 + li) * K;
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* x_s = reinterpret_cast<unsigned short*>(smem_raw);
    constexpr int XS = GG_KG + GG_XPAD;
    // x_s: double-buffered [2][GG_MROWS][XS]

    // x staging ownership: thread t covers GG_NCHUNK of the row×16-dim
    // chunks of a (128 x 128) tile — independent 16 B loads per group/set
    constexpr int DCH = GG_KG / 16;        // 16-dim chunks per row
    constexpr int GG_NCHUNK = GG_MROWS * DCH / GG_BLOCK;
    constexpr int GG_RSTRIDE = GG_BLOCK / DCH;
    const int st_row0 = tid / DCH;         // base row; +GG_RSTRIDE per chunk
    const int st_d = (tid % DCH) * 16;     // dim start within the group
    const int ng = K / GG_KG;              // K-groups (K % 128 == 0 checked
                                           // host-side)

    for (int pass = m_lo; pass < m_hi; pass += GG_MROWS) {
        const int m_cnt = min(GG_MROWS, m_hi - pass);
        const int n_mt = (m_cnt + 15) / 16;

        f32x4 acc[GG_WMT];
#pragma unroll
        for (int t = 0; t < GG_WMT; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

        short8 xr[2][GG_NCHUNK][2];   // [set][chunk][2 x 16B]
        bf16x8 wf[2][GG_KG / GG_BK];  // [set][kstep]

#define GG_XLOAD(G, SET)                                                       \
    do {                                                                       \
        const int k0_ = (G) * GG_KG;                                           \
        _Pragma("unroll") for (int c = 0; c < GG_NCHUNK; ++c) {                \
            const int r_ = st_row0 + c * GG_RSTRIDE;                           \
            xr[SET][c][0] = short8{};                                          \
            xr[SET][c][1] = short8{};                                          \
            if (r_ < m_cnt) {                                                  \
                const unsigned short* src_ =                                   \
                    x + (long)(pass + r_) * K + k0_ + st_d;                    \
                xr[SET][c][0] = *reinterpret_cast<const short8*>(src_);        \
                xr[SET][c][1] = *reinterpret_cast<const short8*>(src_ + 8);    \
            }                                                                  \
        }                                                                      \
    } while (0)
#define GG_WLOAD(G, SET)                                                       \
    do {                                                                       \
        const int k0_ = (G) * GG_KG;                                           \
        _Pragma("unroll") for (int g = 0; g < GG_KG / GG_BK; ++g)              \
            wf[SET][g] = *reinterpret_cast<const bf16x8*>(                     \
                wp + k0_ + g * GG_BK + lg * 8);                                \
    } while (0)
#define GG_XWRITE(SET, BUF)                                                    \
    do {                                                                       \
        unsigned short* dst_ = x_s + (BUF) * GG_MROWS * XS;                    \
        _Pragma("unroll") for (int c = 0; c < GG_NCHUNK; ++c) {                \
            const int r_ = st_row0 + c * GG_RSTRIDE;                           \
            *reinterpret_cast<short8*>(dst_ + r_ * XS + st_d) =                \
                xr[SET][c][0];                                                 \
            *reinterpret_cast<short8*>(dst_ + r_ * XS + st_d + 8) =            \
                xr[SET][c][1];                                                 \
        }                                                                      \
    } while (0)

// one K-group step: write this set's x tile to LDS, prefetch the tile two
// groups ahead, barrier, then 4 K-steps x up-to-8 row-tiles of MFMA; the
// NEXT group's W fragments load after the MFMAs release this set's regs
// (they have a full alternate-set step to land)
#define GG_STEP(G, SET)                                                        \
    do {                                                                       \
        GG_XWRITE(SET, SET);                                                   \
        if ((G) + 2 < ng) GG_XLOAD((G) + 2, SET);                              \
        __syncthreads();                                                       \
        const unsigned short* xb_ = x_s + (SET) * GG_MROWS * XS;               \
        _Pragma("unroll") for (int g_ = 0; g_ < GG_KG / GG_BK; ++g_) {         \
            _Pragma("unroll") for (int t = 0; t < GG_WMT; ++t) {               \
                const bf16x8 a_ = *reinterpret_cast<const bf16x8*>(            \
                    xb_ + ((rowh * GG_WMT + t) * 16 + li) * XS +               \
                    g_ * GG_BK + lg * 8);                                      \
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(              \
                    a_, wf[SET][g_], acc[t], 0, 0, 0);                         \
            }                                                                  \
            __builtin_amdgcn_sched_barrier(0);                                 \
        }                                                                      \
        if ((G) + 2 < ng) GG_WLOAD((G) + 2, SET);                              \
    } while (0)
// (one barrier per group: reads of buffer SET in step G are ordered before
// the step-G+2 rewrite by step G+1's own barrier)

        GG_XLOAD(0, 0);
        GG_WLOAD(0, 0);
        if (ng > 1) {
            GG_XLOAD(1, 1);
            GG_WLOAD(1, 1);
        }
        int grp = 0;
        for (; grp + 1 < ng; grp += 2) {
            GG_STEP(grp, 0);
            GG_STEP(grp + 1, 1);
        }
        if (grp < ng) GG_STEP(grp, 0);
#undef GG_STEP
#undef GG_XWRITE
#undef GG_WLOAD
#undef GG_XLOAD

        // epilogue: C row = lg*4 + r, col = li
#pragma unroll
        for (int t = 0; t < GG_WMT; ++t) {
            if (rowh * GG_WMT + t >= n_mt) continue;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = (rowh * GG_WMT + t) * 16 + lg * 4 + r;
                if (row >= m_cnt) continue;
                out[(long)(pass + row) * N + n0 + colg * 16 + li] =
                    f2bf(acc[t][r]);
            }
        }
        __syncthreads();  // x_s buffers reused by the next 128-row pass
    }
}

extern "C" void launch_grouped_gemm(
    const unsigned short* x, const unsigned short* w, const int* offs,
    unsigned short* out, int E, int S, int N, int K, hipStream_t stream) {
    dim3 grid(N / GG_BN, E);
    const int smem = 2 * GG_MROWS * (GG_KG + GG_XPAD) * 2;
    hipLaunchKernelGGL(grouped_gemm_kernel, grid, dim3(GG_BLOCK), smem,
                       stream, x, w, offs, out, N, K);
}
