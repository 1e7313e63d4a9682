// Varlen causal prefill attention (flash-style, no S x S materialization).
//
// Both kernels are templated over PAGED: when true, K/V come from the paged
// KV cache ([nb, nkv, bs, hd] + block table) instead of the packed fresh
// projections, and each query chunk attends to the full cached history
// (chunked prefill: query row j of sequence b has absolute position
// hist + j where hist = seq_lens[b] - query_len[b]). The chunk's own K/V
// were stored to the cache by kv_store before this kernel runs.
//
// Two kernels:
//  * attn_prefill_mfma — the production path for head_dim 64/128: per
//    workgroup 4 waves x 16 q-rows (64-row Q tile), 32-key K/V tiles staged
//    in LDS (+8-element row padding => conflict-free ds_read_b128, guide
//    T2/G4), QK^T and P.V on v_mfma_f32_16x16x32_bf16, online softmax in
//    registers with a 16-lane-group shuffle reduce, V transposed at staging
//    so the PV B-fragment is a contiguous 16 B LDS read.
//  * attn_prefill_basic — correctness fallback for small/odd head dims
//    (test models): per (seq, head) workgroup, lanes-over-keys scores +
//    lanes-over-dims PV like the decode kernel.
//
// Numerics reference: ops/reference.py attn_prefill.
#include "common.h"

// KV element loaders shared by the bf16 and fp8 (OCP e4m3) cache paths
__device__ __forceinline__ float kv_elem_f32(unsigned short x) {
    return bf2f(x);
}
__device__ __forceinline__ float kv_elem_f32(unsigned char x) {
    return fp8x2_2f((unsigned short)x).x;
}
__device__ __forceinline__ bf16x8 load_kv8_bf16(const unsigned short* p) {
    return *reinterpret_cast<const bf16x8*>(p);
}
__device__ __forceinline__ bf16x8 load_kv8_bf16(const unsigned char* p) {
    float f[8];
    load_fp8x8(p, f);
    bf16x8 r;
#pragma unroll
    for (int e = 0; e < 8; ++e) r[e] = (__bf16)f[e];
    return r;
}

// ------------------------------------------------------------------ basic
template <bool PAGED, typename KVT>
__global__ __launch_bounds__(64) void attn_prefill_basic(
    const unsigned short* __restrict__ q,  // [T, nq, hd]
    const KVT* __restrict__ k,  // [T,nkv,hd] | PAGED: [nb,nkv,bs,hd]
    const KVT* __restrict__ v,
    const int* __restrict__ cu,            // [nseq+1] (query tokens)
    const int* __restrict__ block_table,   // PAGED only: [nseq, W]
    const int* __restrict__ seq_lens,      // PAGED only: total len incl chunk
    unsigned short* __restrict__ out,      // [T, nq, hd]
    int nq, int nkv, int hd, long s_q, long s_k, long s_v, int W, int bs,
    float scale, int causal) {
    const int seq = blockIdx.y;
    const int h = blockIdx.x;
    const int kvh = h / (nq / nkv);
    const int s0 = cu[seq], s1 = cu[seq + 1];
    const int QL = s1 - s0;
    const int L = PAGED ? seq_lens[seq] : QL;
    const int hist = L - QL;
    const int* bt = PAGED ? block_table + (long)seq * W : nullptr;
    const int lane = threadIdx.x;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    float* q_s = reinterpret_cast<float*>(smem_raw);  // [hd]
    float* p_s = q_s + hd;                            // [WAVE]

    for (int r = 0; r < QL; ++r) {  // one q row at a time
        for (int d = lane; d < hd; d += WAVE)
            q_s[d] = bf2f(q[(long)(s0 + r) * s_q + (long)h * hd + d]) * scale;
        __builtin_amdgcn_s_barrier();
        const int kmax = causal ? hist + r + 1 : L;
        float m = -1e30f, lsum = 0.f, o0 = 0.f, o1 = 0.f;
        const int d0 = lane * 2;
        for (int base = 0; base < kmax; base += WAVE) {
            const int key = base + lane;
            float s = -1e30f;
            if (key < kmax) {
                s = 0.f;
                const KVT* kr;
                if (PAGED)
                    kr = k + (((long)bt[key / bs] * nkv + kvh) * bs +
                              key % bs) * hd;
                else
                    kr = k + (long)(s0 + key) * s_k + (long)kvh * hd;
                for (int d = 0; d < hd; ++d)
                    s = fmaf(kv_elem_f32(kr[d]), q_s[d], s);
            }
            float cmax = wave_max(s);
            float mn = fmaxf(m, cmax);
            float p = (key < kmax) ? __expf(s - mn) : 0.f;
            float alpha = __expf(m - mn);
            m = mn;
            lsum = lsum * alpha + wave_sum(p);
            o0 *= alpha;
            o1 *= alpha;
            p_s[lane] = p;
            const int nk = min(WAVE, kmax - base);
            if (d0 < hd) {
                for (int t = 0; t < nk; ++t) {
                    const int vk = base + t;
                    const KVT* vr;
                    if (PAGED)
                        vr = v + (((long)bt[vk / bs] * nkv + kvh) * bs +
                                  vk % bs) * hd;
                    else
                        vr = v + (long)(s0 + vk) * s_v + (long)kvh * hd;
                    o0 = fmaf(p_s[t], kv_elem_f32(vr[d0]), o0);
                    o1 = fmaf(p_s[t], kv_elem_f32(vr[d0 + 1]), o1);
                }
            }
        }
        if (d0 < hd) {
            const float inv = (lsum > 0.f) ? 1.f / lsum : 0.f;
            unsigned short* orow = out + ((long)(s0 + r) * nq + h) * hd;
            orow[d0] = f2bf(o0 * inv);
            orow[d0 + 1] = f2bf(o1 * inv);
        }
        __builtin_amdgcn_s_barrier();
    }
}

// ------------------------------------------------------------------- mfma
// Fragment maps for v_mfma_f32_16x16x32_bf16 (gfx950):
//   A (16x32): lane holds row (l&15), k = (l>>4)*8 + e   (8 bf16)
//   B (32x16): lane holds col (l&15), k = (l>>4)*8 + e   (8 bf16)
//   C (16x16): lane holds col (l&15), rows (l>>4)*4 + r  (4 f32)
// Verified on hardware by tests/test_ops_gpu.py::test_mfma_fragment_map.
#define PF_QROWS 16   // q rows per wave
#define PF_WAVES 4
#define PF_TM (PF_QROWS * PF_WAVES)  // 64-row q tile per workgroup
#define PF_TN 64                     // kv tile (4 x 16-key fragments)
#define PF_PAD 8                     // LDS row padding (bf16 elems)

template <int HD, bool PAGED, typename KVT>
__global__ __launch_bounds__(256) void attn_prefill_mfma(
    const unsigned short* __restrict__ q,
    const KVT* __restrict__ k,  // PAGED: k_cache [nb,nkv,bs,hd]
    const KVT* __restrict__ v,
    const int* __restrict__ cu,            // [nseq+1] query tokens
    const int* __restrict__ block_table,   // PAGED only
    const int* __restrict__ seq_lens,      // PAGED only
    unsigned short* __restrict__ out,
    int nq, int nkv, long s_q, long s_k, long s_v, int W, int bs,
    float scale) {
    constexpr int KSTEPS = HD / 32;
    const int seq = blockIdx.z;
    const int h = blockIdx.y;
    const int kvh = h / (nq / nkv);
    const int s0 = cu[seq], s1 = cu[seq + 1];
    const int QL = s1 - s0;                // query rows in this chunk
    const int L = PAGED ? seq_lens[seq] : QL;  // keys incl. cached history
    const int hist = L - QL;
    const int* bt = PAGED ? block_table + (long)seq * W : nullptr;
    const int q0 = blockIdx.x * PF_TM;  // q tile base (chunk-local row)
    if (q0 >= QL) return;

    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wid = tid / WAVE;
    const int lg = lane >> 4;   // 4 lane groups of 16
    const int li = lane & 15;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    // K tile [PF_TN][HD+PAD], V^T tile [HD][PF_TN+PAD], both bf16.
    // The K buffer is later reused as the P tile [PF_TM][PF_TN+PAD], so the
    // V^T carve starts past max(K tile, P tile) — at HD=64 the P tile is the
    // larger one and a K-sized carve would let P overwrite V^T.
    constexpr int VT_STRIDE = PF_TN + PF_PAD;
    constexpr int K_ELEMS = PF_TN * (HD + PF_PAD);
    constexpr int P_ELEMS = PF_TM * VT_STRIDE;
    constexpr int VT_OFF = (K_ELEMS > P_ELEMS ? K_ELEMS : P_ELEMS);
    unsigned short* k_s = reinterpret_cast<unsigned short*>(smem_raw);
    unsigned short* vt_s = k_s + VT_OFF;

    // ---- load this wave's Q rows into A-fragments (registers), scaled
    bf16x8 a_q[KSTEPS];
    const int my_qrow = q0 + wid * PF_QROWS + li;  // A row = li
    const bool row_ok = my_qrow < QL;
    {
        const unsigned short* qr = q + (long)(s0 + my_qrow) * s_q + (long)h * HD;
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks) {
            const int d = ks * 32 + lg * 8;
            if (row_ok) {
                short8 raw = *reinterpret_cast<const short8*>(qr + d);
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    a_q[ks][e] = (__bf16)(bf2f((unsigned short)raw[e]) * scale);
            } else {
#pragma unroll
                for (int e = 0; e < 8; ++e) a_q[ks][e] = (__bf16)0.f;
            }
        }
    }

    // online softmax state: this lane covers rows (lg*4 + r) of the wave's
    // 16-row block; stats per C-row slot r (4 rows), kept redundantly by
    // the 16 lanes of each group.
    float m_run[4], l_run[4];
    f32x4 o_acc[HD / 16];  // C frags: [dim block][4 rows]
#pragma unroll
    for (int r = 0; r < 4; ++r) { m_run[r] = -1e30f; l_run[r] = 0.f; }
#pragma unroll
    for (int nb = 0; nb < HD / 16; ++nb) o_acc[nb] = f32x4{0.f, 0.f, 0.f, 0.f};

    // causal: keys up to the last absolute q position of this tile
    const int kv_end = min(L, hist + q0 + PF_TM);

    for (int kb = 0; kb < kv_end; kb += PF_TN) {
        // ---- stage K tile [32][HD] and V^T tile [HD][32]
        __syncthreads();
        {
            // each thread loads 16 B: key = tid / (HD/8), dims 8*(tid % ...)
            constexpr int THREADS_PER_ROW = HD / 8;
            for (int i = tid; i < PF_TN * THREADS_PER_ROW; i += 256) {
                const int key = i / THREADS_PER_ROW;
                const int d = (i % THREADS_PER_ROW) * 8;
                const int gk = kb + key;
                bf16x8 kraw{}, vraw{};
                if (gk < kv_end) {
                    if (PAGED) {
                        const long koff = (((long)bt[gk / bs] * nkv + kvh) *
                                           bs + gk % bs) * HD + d;
                        kraw = load_kv8_bf16(k + koff);
                        vraw = load_kv8_bf16(v + koff);
                    } else {
                        kraw = load_kv8_bf16(
                            k + (long)(s0 + gk) * s_k + (long)kvh * HD + d);
                        vraw = load_kv8_bf16(
                            v + (long)(s0 + gk) * s_v + (long)kvh * HD + d);
                    }
                }
                *reinterpret_cast<bf16x8*>(k_s + key * (HD + PF_PAD) + d) = kraw;
#pragma unroll
                for (int e = 0; e < 8; ++e)
                    vt_s[(d + e) * VT_STRIDE + key] =
                        ((const unsigned short*)&vraw)[e];
            }
        }
        __syncthreads();

        // ---- S = Q K^T : PF_TN/16 16-key column fragments
        f32x4 sfrag[PF_TN / 16];
#pragma unroll
        for (int f = 0; f < PF_TN / 16; ++f) {
            f32x4 acc{0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int ks = 0; ks < KSTEPS; ++ks) {
                // B-frag: col = key (li + 16f), k = dim ks*32 + lg*8 + e
                bf16x8 bk = *reinterpret_cast<const bf16x8*>(
                    k_s + (f * 16 + li) * (HD + PF_PAD) + ks * 32 + lg * 8);
                acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_q[ks], bk, acc,
                                                              0, 0, 0);
            }
            sfrag[f] = acc;
        }

        // ---- causal mask + online softmax (rows rr = lg*4 + r)
        float pmax[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) pmax[r] = -1e30f;
#pragma unroll
        for (int f = 0; f < PF_TN / 16; ++f) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int qrow = q0 + wid * PF_QROWS + lg * 4 + r;
                const int key = kb + f * 16 + li;
                if (key > hist + qrow || qrow >= QL || key >= kv_end)
                    sfrag[f][r] = -1e30f;
                pmax[r] = fmaxf(pmax[r], sfrag[f][r]);
            }
        }
        // row max across the 16 lanes of the group (cols)
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
#pragma unroll
            for (int r = 0; r < 4; ++r)
                pmax[r] = fmaxf(pmax[r], __shfl_xor(pmax[r], off));

        float alpha[4], psum[4];
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const float mn = fmaxf(m_run[r], pmax[r]);
            alpha[r] = __expf(m_run[r] - mn);
            m_run[r] = mn;
            psum[r] = 0.f;
        }
        // P = exp(S - m); convert to bf16 A-fragment via LDS staging:
        // reuse k_s as the P tile [PF_TM][PF_TN+PAD] (barrier-protected).
        unsigned short* p_s = k_s;  // reuse; careful with sizes (TM*(TN+8))
        __syncthreads();  // everyone done reading K before overwrite
#pragma unroll
        for (int f = 0; f < PF_TN / 16; ++f) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                float pv = __expf(sfrag[f][r] - m_run[r]);
                if (sfrag[f][r] <= -1e29f) pv = 0.f;
                psum[r] += pv;
                p_s[(wid * PF_QROWS + lg * 4 + r) * VT_STRIDE + f * 16 + li] =
                    f2bf(pv);
            }
        }
        // row sum across group lanes
#pragma unroll
        for (int off = 1; off < 16; off <<= 1)
#pragma unroll
            for (int r = 0; r < 4; ++r) psum[r] += __shfl_xor(psum[r], off);
#pragma unroll
        for (int r = 0; r < 4; ++r)
            l_run[r] = l_run[r] * alpha[r] + psum[r];

        // rescale O by alpha (alpha is per C-row r, uniform across cols)
#pragma unroll
        for (int nb = 0; nb < HD / 16; ++nb)
#pragma unroll
            for (int r = 0; r < 4; ++r) o_acc[nb][r] *= alpha[r];

        // ---- O += P V : A = P rows of this wave, B = V^T
        // (p_s writes above are within-wave for our rows; but other waves
        // share the LDS buffer -> sync so K-tile reuse is safe)
        __syncthreads();
        bf16x8 a_p[PF_TN / 32];  // A k-dim = keys: 32 keys per fragment
#pragma unroll
        for (int ks2 = 0; ks2 < PF_TN / 32; ++ks2)
            a_p[ks2] = *reinterpret_cast<const bf16x8*>(
                p_s + (wid * PF_QROWS + li) * VT_STRIDE + ks2 * 32 + lg * 8);
#pragma unroll
        for (int nb = 0; nb < HD / 16; ++nb) {
#pragma unroll
            for (int ks2 = 0; ks2 < PF_TN / 32; ++ks2) {
                bf16x8 bv = *reinterpret_cast<const bf16x8*>(
                    vt_s + (nb * 16 + li) * VT_STRIDE + ks2 * 32 + lg * 8);
                o_acc[nb] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    a_p[ks2], bv, o_acc[nb], 0, 0, 0);
            }
        }
    }

    // ---- epilogue: normalize and store (C layout: col=li, row=lg*4+r)
    const int orow_base = q0 + wid * PF_QROWS;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int qrow = orow_base + lg * 4 + r;
        if (qrow >= QL) continue;
        const float inv = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
        unsigned short* orow = out + ((long)(s0 + qrow) * nq + h) * HD;
#pragma unroll
        for (int nb = 0; nb < HD / 16; ++nb)
            orow[nb * 16 + li] = f2bf(o_acc[nb][r] * inv);
    }
}

static int prefill_mfma_smem(int hd) {
    // LDS: max(K tile, P tile) + V^T tile, bf16; must match VT_OFF
    const int k_bytes = PF_TN * (hd + PF_PAD) * 2;
    const int p_bytes = PF_TM * (PF_TN + PF_PAD) * 2;
    const int vt_bytes = hd * (PF_TN + PF_PAD) * 2;
    return (k_bytes > p_bytes ? k_bytes : p_bytes) + vt_bytes;
}

extern "C" void launch_attn_prefill(
    const unsigned short* q, const unsigned short* k, const unsigned short* v,
    const int* cu, unsigned short* out, int nseq, int nq, int nkv, int hd,
    long s_q, long s_k, long s_v, int max_seqlen, float scale, int causal,
    hipStream_t stream) {
    if ((hd == 64 || hd == 128) && causal) {
        const int tiles = (max_seqlen + PF_TM - 1) / PF_TM;
        dim3 grid(tiles, nq, nseq);
        const int smem = prefill_mfma_smem(hd);
        if (hd == 128)
            hipLaunchKernelGGL((attn_prefill_mfma<128, false, unsigned short>),
                               grid, dim3(256), smem, stream, q, k, v, cu,
                               nullptr, nullptr, out, nq, nkv, s_q, s_k, s_v,
                               0, 0, scale);
        else
            hipLaunchKernelGGL((attn_prefill_mfma<64, false, unsigned short>),
                               grid, dim3(256), smem, stream, q, k, v, cu,
                               nullptr, nullptr, out, nq, nkv, s_q, s_k, s_v,
                               0, 0, scale);
        return;
    }
    dim3 grid(nq, nseq);
    const int smem = (hd + WAVE) * 4;
    hipLaunchKernelGGL((attn_prefill_basic<false, unsigned short>), grid,
                       dim3(WAVE), smem, stream, q, k, v, cu, nullptr,
                       nullptr, out, nq, nkv, hd, s_q, s_k, s_v, 0, 0, scale,
                       causal);
}

// Chunked prefill: query chunks (packed varlen, cu) attend to the full
// paged history; the chunk's K/V are already in the cache (kv_store ran
// first), so K/V come exclusively from the paged pool.
extern "C" void launch_attn_prefill_paged(
    const unsigned short* q, const void* k_cache, const void* v_cache,
    const int* cu, const int* block_table, const int* seq_lens,
    unsigned short* out, int nseq, int nq, int nkv, int hd, long s_q, int W,
    int bs, int max_qlen, float scale, int fp8, hipStream_t stream) {
#define PFP_LAUNCH(KVT)                                                        \
    do {                                                                       \
        const KVT* kc = reinterpret_cast<const KVT*>(k_cache);                 \
        const KVT* vc = reinterpret_cast<const KVT*>(v_cache);                 \
        if (hd == 64 || hd == 128) {                                           \
            const int tiles = (max_qlen + PF_TM - 1) / PF_TM;                  \
            dim3 grid(tiles, nq, nseq);                                        \
            const int smem = prefill_mfma_smem(hd);                            \
            if (hd == 128)                                                     \
                hipLaunchKernelGGL((attn_prefill_mfma<128, true, KVT>), grid,  \
                                   dim3(256), smem, stream, q, kc, vc, cu,     \
                                   block_table, seq_lens, out, nq, nkv, s_q,   \
                                   0, 0, W, bs, scale);                        \
            else                                                               \
                hipLaunchKernelGGL((attn_prefill_mfma<64, true, KVT>), grid,   \
                                   dim3(256), smem, stream, q, kc, vc, cu,     \
                                   block_table, seq_lens, out, nq, nkv, s_q,   \
                                   0, 0, W, bs, scale);                        \
            return;                                                            \
        }                                                                      \
        dim3 grid(nq, nseq);                                                   \
        const int smem = (hd + WAVE) * 4;                                      \
        hipLaunchKernelGGL((attn_prefill_basic<true, KVT>), grid, dim3(WAVE),  \
                           smem, stream, q, kc, vc, cu, block_table, seq_lens, \
                           out, nq, nkv, hd, s_q, 0, 0, W, bs, scale, 1);      \
    } while (0)
    if (fp8) PFP_LAUNCH(unsigned char);
    else PFP_LAUNCH(unsigned short);
#undef PFP_LAUNCH
}
