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
//   grid (N/64, E); one 512-thread workgroup = 64 output columns x up to
//   128 rows of one expert (one pass covers the whole segment at decode
//   sizes; a 64-row pass re-read W twice at S=1024/E=8 -> 1.7 TB/s).
//   BOTH operands stage through double-buffered LDS tiles (classic GEMM
//   pipeline): per K-group of 64 dims each thread issues exactly one 16 B
//   W load and one 32 B x load, cooperatively; waves then eat MFMA
//   fragments out of LDS. This de-duplicates the W fragments the 2
//   row-half waves share (a register-direct W variant loaded each panel
//   twice and measured 2.5-2.7 TB/s) and makes the W staging writes
//   bank-conflict-free (16 consecutive granule slots per quarter-wave;
//   the register-direct variant measured 13.8% LDSBankConflict cycles).
//   Loads prefetch TWO K-groups ahead with SET/BUF as literals so the
//   compiler tracks exactly which outstanding loads each LDS write waits
//   on (a runtime-indexed set forced vmcnt(0): measured 2x slower in
//   attn_pv_mfma_kernel, attn_decode.hip).
//   Experts with more rows loop in 128-row passes (W re-read per pass —
//   at M>=128/pass the arithmetic intensity is past the memory knee).
//
// Numerics reference: ops/reference.py grouped_gemm.
#include "common.h"

#define GG_BLOCK 512
#define GG_WAVES 8
#define GG_BN 64          // output cols per workgroup (16 per col-group)
#define GG_MROWS 128      // rows per pass (8 MFMA row-tiles)
#define GG_MT (GG_MROWS / 16)
#define GG_WMT (GG_MT / 2)  // row-tiles per wave (8 waves = 4 col-groups
                            // x 2 row-halves)
#define GG_BK 32          // K step (one mfma_16x16x32)
#define GG_KG 64          // K-group per pipeline stage (2 K-steps)
#define GG_PAD 8          // LDS row padding (bf16 elems): 144 B rows keep
                          // b128 fragment reads on 16 distinct banks at
                          // the 64-dword modulus (9-granule stride)

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
    BB_KASSERT(m_lo >= 0 && m_lo <= m_hi);  // offsets must be a prefix
    if (m_lo >= m_hi) return;
    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wid = tid / WAVE;
    const int lg = lane >> 4;
    const int li = lane & 15;
    const int colg = wid & 3;   // wave's 16-col group
    const int rowh = wid >> 2;  // wave's 64-row half

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* x_s = reinterpret_cast<unsigned short*>(smem_raw);
    constexpr int XS = GG_KG + GG_PAD;
    // x_s: [2][GG_MROWS][XS]; w_s: [2][GG_BN][XS]
    unsigned short* w_s = x_s + 2 * GG_MROWS * XS;

    // x staging: thread covers one 32 B chunk: row tid/4, dims (tid%4)*16
    const int xs_row = tid >> 2;
    const int xs_d = (tid & 3) * 16;
    // W staging: thread covers one 16 B chunk: col tid/8, dims (tid%8)*8
    // (quarter-wave = 2 cols x 8 chunks -> 16 consecutive granule slots,
    // conflict-free)
    const int ws_col = tid >> 3;
    const int ws_d = (tid & 7) * 8;
    const unsigned short* wrow = w + ((long)e * N + n0 + ws_col) * K + ws_d;
    const int ng = K / GG_KG;   // K % 128 == 0 checked host-side

    for (int pass = m_lo; pass < m_hi; pass += GG_MROWS) {
        const int m_cnt = min(GG_MROWS, m_hi - pass);
        const int n_mt = (m_cnt + 15) / 16;

        f32x4 acc[GG_WMT];
#pragma unroll
        for (int t = 0; t < GG_WMT; ++t) acc[t] = f32x4{0.f, 0.f, 0.f, 0.f};

        short8 xr[4][2];   // [reg set][2 x 16B] — loads prefetch FOUR
        short8 wr[4];      // groups ahead in registers (two LDS buffers:
                           // regs sit 3 steps, written 1 step before use);
                           // ~96 B/lane outstanding covers HBM latency

#define GG_LOAD(G, SET)                                                        \
    do {                                                                       \
        const int k0_ = (G) * GG_KG;                                           \
        /* rows past the segment re-load row 0 instead of branching: each  */\
        /* C row depends only on its own A row and the epilogue never      */\
        /* writes rows >= m_cnt, so the garbage stays contained            */\
        const unsigned short* xsrc_ =                                          \
            x + (long)(pass + (xs_row < m_cnt ? xs_row : 0)) * K + k0_ + xs_d; \
        xr[SET][0] = *reinterpret_cast<const short8*>(xsrc_);                  \
        xr[SET][1] = *reinterpret_cast<const short8*>(xsrc_ + 8);              \
        wr[SET] = *reinterpret_cast<const short8*>(wrow + k0_);                \
    } while (0)
#define GG_WRITE(RS, SET)                                                      \
    do {                                                                       \
        unsigned short* xd_ = x_s + (SET) * GG_MROWS * XS + xs_row * XS + xs_d;\
        *reinterpret_cast<short8*>(xd_) = xr[RS][0];                           \
        *reinterpret_cast<short8*>(xd_ + 8) = xr[RS][1];                       \
        *reinterpret_cast<short8*>(                                            \
            w_s + (SET) * GG_BN * XS + ws_col * XS + ws_d) = wr[RS];           \
    } while (0)

// one K-group step: write this set's tiles to LDS, prefetch two groups
// ahead, barrier, then 2 K-steps x 4 row-tiles of MFMA out of LDS
#define GG_STEP(G, RS, SET)                                                    \
    do {                                                                       \
        GG_WRITE(RS, SET);                                                     \
        if ((G) + 4 < ng) GG_LOAD((G) + 4, RS);                                \
        __syncthreads();                                                       \
        const unsigned short* xb_ = x_s + (SET) * GG_MROWS * XS;               \
        const unsigned short* wb_ = w_s + (SET) * GG_BN * XS;                  \
        _Pragma("unroll") for (int g_ = 0; g_ < GG_KG / GG_BK; ++g_) {         \
            const bf16x8 b_ = *reinterpret_cast<const bf16x8*>(                \
                wb_ + (colg * 16 + li) * XS + g_ * GG_BK + lg * 8);            \
            _Pragma("unroll") for (int t = 0; t < GG_WMT; ++t) {               \
                const bf16x8 a_ = *reinterpret_cast<const bf16x8*>(            \
                    xb_ + ((rowh * GG_WMT + t) * 16 + li) * XS +               \
                    g_ * GG_BK + lg * 8);                                      \
                acc[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(              \
                    a_, b_, acc[t], 0, 0, 0);                                  \
            }                                                                  \
            __builtin_amdgcn_sched_barrier(0);                                 \
        }                                                                      \
    } while (0)
// (one barrier per group: reads of buffer SET in step G are ordered before
// the step-G+2 rewrite by step G+1's own barrier)

        GG_LOAD(0, 0);
        if (ng > 1) GG_LOAD(1, 1);
        if (ng > 2) GG_LOAD(2, 2);
        if (ng > 3) GG_LOAD(3, 3);
        int grp = 0;
        for (; grp + 3 < ng; grp += 4) {
            GG_STEP(grp, 0, 0);
            GG_STEP(grp + 1, 1, 1);
            GG_STEP(grp + 2, 2, 0);
            GG_STEP(grp + 3, 3, 1);
        }
        // K is a multiple of 128 => ng even: at most 2 groups remain
        if (grp < ng) GG_STEP(grp, 0, 0);
        if (grp + 1 < ng) GG_STEP(grp + 1, 1, 1);
#undef GG_STEP
#undef GG_WRITE
#undef GG_LOAD

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
        __syncthreads();  // LDS buffers reused by the next 128-row pass
    }
}

extern "C" void launch_grouped_gemm(
    const unsigned short* x, const unsigned short* w, const int* offs,
    unsigned short* out, int E, int S, int N, int K, hipStream_t stream) {
    dim3 grid(N / GG_BN, E);
    const int smem = 2 * (GG_MROWS + GG_BN) * (GG_KG + GG_PAD) * 2;
    hipLaunchKernelGGL(grouped_gemm_kernel, grid, dim3(GG_BLOCK), smem,
                       stream, x, w, offs, out, N, K);
}
