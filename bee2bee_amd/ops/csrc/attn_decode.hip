// Paged GQA decode attention — three-stage flash-decoding.
//
// The decode step is memory-bound on KV reads (per key: 2*hd*2 B moved).
// A single fused kernel could not reach the HBM roofline: its score phase
// (lanes-over-keys) and PV phase (lanes-over-dims) have opposing register
// needs, and together they pushed VGPRs past 100 -> 2 waves/SIMD -> too few
// bytes in flight (measured 1.7-2.5 TB/s vs 6.0 TB/s for each access
// pattern probed in isolation — see probe.hip bw_* kernels). Splitting into
// per-chunk kernels keeps each at <=64 VGPRs with deep independent-load
// batches:
//   1. attn_scores:  S = q.K^T + chunk-level softmax -> p, (m,l) scratch
//   2. attn_pv:      partial_o = p.V per chunk (p from scratch, LDS-staged)
//   3. attn_combine: merge chunk partials, normalize, write bf16 out
// Scratch p traffic is ~3% of KV traffic.
//
// Replaces: the reference's decode path inside transformers.generate()
// (bee2bee/hf.py:84-108). Numerics reference: ops/reference.py attn_decode.
#include "common.h"

#define DEC_BLOCK 256
#define DEC_WAVES 4
#define DEC_CHUNK 256  // keys per workgroup (DEC_WAVES x 64)

// KV element loaders, templated on the cache storage type:
//   unsigned short = bf16 (2 B/elem), unsigned char = OCP fp8 e4m3 (1 B).
template <typename KVT>
__device__ __forceinline__ bf16x8 load_kv_bf16x8(const KVT* p);
template <>
__device__ __forceinline__ bf16x8 load_kv_bf16x8(const unsigned short* p) {
    return *reinterpret_cast<const bf16x8*>(p);
}
template <>
__device__ __forceinline__ bf16x8 load_kv_bf16x8(const unsigned char* p) {
    float f[8];
    load_fp8x8(p, f);
    bf16x8 r;
#pragma unroll
    for (int e = 0; e < 8; ++e) r[e] = (__bf16)f[e];
    return r;
}
// raw 2-element cache pair (batched loads keep the raw bytes in flight,
// converting only when consumed)
template <typename KVT> struct kv_pair_t;
template <> struct kv_pair_t<unsigned short> { short2v v; };
template <> struct kv_pair_t<unsigned char> { u8x2 v; };
__device__ __forceinline__ float2v kv_pair_f32(kv_pair_t<unsigned short> p) {
    return float2v{bf2f((unsigned short)p.v[0]), bf2f((unsigned short)p.v[1])};
}
__device__ __forceinline__ float2v kv_pair_f32(kv_pair_t<unsigned char> p) {
    unsigned short w = (unsigned short)p.v[0] |
                       ((unsigned short)p.v[1] << 8);
    return fp8x2_2f(w);
}

// MFMA fragment plumbing for the scores kernel: bf16 caches feed the bf16
// MFMA; fp8 caches feed v_mfma_f32_16x16x32_fp8_fp8 DIRECTLY (same lane
// map, hardware-verified by test_mfma_fp8_fragment_map) — K bytes go
// straight from the load into the matrix op, no convert VALU. q is
// quantized to e4m3 once at fragment setup (the fp8 cache mode is already
// approximate; documented).
template <typename KVT> struct kfrag_of { using type = bf16x8; };
template <> struct kfrag_of<unsigned char> { using type = long; };
template <typename KVT>
__device__ __forceinline__ typename kfrag_of<KVT>::type load_kfrag(
    const KVT* p) {
    return *reinterpret_cast<const typename kfrag_of<KVT>::type*>(p);
}
__device__ __forceinline__ bf16x8 qfrag_from_lds_bf16(
    const unsigned short* p) {
    return *reinterpret_cast<const bf16x8*>(p);
}
__device__ __forceinline__ long qfrag_from_lds_fp8(const unsigned short* p) {
    unsigned short w[4];
#pragma unroll
    for (int i = 0; i < 4; ++i)
        w[i] = f2fp8x2(bf2f(p[2 * i]), bf2f(p[2 * i + 1]));
    long r;
    unsigned short* rp = reinterpret_cast<unsigned short*>(&r);
#pragma unroll
    for (int i = 0; i < 4; ++i) rp[i] = w[i];
    return r;
}
template <typename KVT>
__device__ __forceinline__ typename kfrag_of<KVT>::type qfrag_from_lds(
    const unsigned short* p);
template <>
__device__ __forceinline__ bf16x8 qfrag_from_lds<unsigned short>(
    const unsigned short* p) {
    return qfrag_from_lds_bf16(p);
}
template <>
__device__ __forceinline__ long qfrag_from_lds<unsigned char>(
    const unsigned short* p) {
    return qfrag_from_lds_fp8(p);
}
__device__ __forceinline__ f32x4 mfma_kq(bf16x8 a, bf16x8 b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}
__device__ __forceinline__ f32x4 mfma_kq(long a, long b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, c, 0, 0, 0);
}

template <typename KVT>
__device__ __forceinline__ float2v load_kv_f32x2(const KVT* p);
template <>
__device__ __forceinline__ float2v load_kv_f32x2(const unsigned short* p) {
    const short2v v = *reinterpret_cast<const short2v*>(p);
    return float2v{bf2f((unsigned short)v[0]), bf2f((unsigned short)v[1])};
}
template <>
__device__ __forceinline__ float2v load_kv_f32x2(const unsigned char* p) {
    return fp8x2_2f(*reinterpret_cast<const unsigned short*>(p));
}

// ------------------------------------------------------------ stage 1: S
// Whole K row staged into registers (HD/8 independent 16 B loads in one
// burst), then G dot products against the LDS-held q. Compile-time HD keeps
// the row array in registers (a runtime bound would spill it to scratch).
template <int G, int HD, typename KVT>
__device__ __forceinline__ void score_row(const KVT* kr,
                                          const unsigned short* q_s,
                                          float* s) {
    bf16x8 krow[HD / 8];
#pragma unroll
    for (int ii = 0; ii < HD / 8; ++ii)
        krow[ii] = load_kv_bf16x8<KVT>(kr + ii * 8);
#pragma unroll
    for (int g = 0; g < G; ++g) {
#pragma unroll
        for (int ii = 0; ii < HD / 8; ++ii) {
            const short8 qraw =
                *reinterpret_cast<const short8*>(q_s + g * HD + ii * 8);
#pragma unroll
            for (int j = 0; j < 8; ++j)
                s[g] = fmaf((float)krow[ii][j],
                            bf2f((unsigned short)qraw[j]), s[g]);
        }
        // stop the scheduler hoisting every g's LDS q-reads to the top
        // (otherwise G=4/HD=128 allocates 256 VGPRs -> 1 wave/SIMD)
        __builtin_amdgcn_sched_barrier(0);
    }
}

template <int G, typename KVT>
__global__ __launch_bounds__(DEC_BLOCK) void attn_scores_kernel(
    const unsigned short* __restrict__ q,        // [B, nq, hd] (strided)
    const KVT* __restrict__ k_cache,             // [nb, nkv, bs, hd]
    const int* __restrict__ block_table,         // [B, W]
    const int* __restrict__ seq_lens,            // [B]
    unsigned short* __restrict__ p_out,          // [B, nkv, C, CHUNK, G] bf16
    float* __restrict__ part_ml,                 // [B, nkv, C, G, 2]
    int nkv, int W, int bs, int hd, int C, long q_stride, float scale) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int L = seq_lens[b];
    if ((int)blockIdx.z * DEC_CHUNK >= L) return;
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* q_s = reinterpret_cast<unsigned short*>(smem_raw);  // [G*hd]
    float* red = reinterpret_cast<float*>(smem_raw + ((G * hd * 2 + 15) & ~15));
    // red: [DEC_WAVES][G][2] wave-level m / sum

    for (int i = threadIdx.x; i < G * hd; i += DEC_BLOCK) {
        const int g = i / hd, d = i % hd;
        q_s[i] = f2bf(
            bf2f(q[(long)b * q_stride + (kvh * G + g) * (long)hd + d]) * scale);
    }
    __syncthreads();

    const int* bt = block_table + (long)b * W;
    // loop chunks with stride gridDim.z — the PV and combine stages walk
    // ALL chunks of (b, kvh), so every chunk's p/(m,l) must be produced even
    // when the launcher picks Z < ceil(L/DEC_CHUNK) (a single-
    // chunk kernel left chunks >= Z uninitialized for hd not in {64,128})
    for (int chunk = blockIdx.z; chunk * DEC_CHUNK < L; chunk += gridDim.z) {
    const int start = chunk * DEC_CHUNK;
    const int key = start + threadIdx.x;  // one key per thread
    const bool valid = key < L;

    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) s[g] = -1e30f;
    if (valid) {
        const int page = bt[key / bs];
        BB_KASSERT(page >= 0);  // block table row must be fully mapped
        const KVT* kr =
            k_cache + (((long)page * nkv + kvh) * bs + key % bs) * hd;
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = 0.f;
        if (hd == 128) {
            score_row<G, 128, KVT>(kr, q_s, s);
        } else if (hd == 64) {
            score_row<G, 64, KVT>(kr, q_s, s);
        } else {
            for (int d = 0; d < hd; d += 8) {  // tail models (hd 16/24/...)
                const bf16x8 kraw = load_kv_bf16x8<KVT>(kr + d);
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    const short8 qraw =
                        *reinterpret_cast<const short8*>(q_s + g * hd + d);
#pragma unroll
                    for (int j = 0; j < 8; ++j)
                        s[g] = fmaf((float)kraw[j],
                                    bf2f((unsigned short)qraw[j]), s[g]);
                }
            }
        }
    }

    // chunk-level softmax: wave max -> LDS -> chunk max; then exp + sums
    float wmax[G];
#pragma unroll
    for (int g = 0; g < G; ++g) wmax[g] = wave_max(s[g]);
    if (lane == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) red[(wid * G + g) * 2] = wmax[g];
    }
    __syncthreads();
    float M[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
        float m = red[g * 2];
        for (int w2 = 1; w2 < DEC_WAVES; ++w2)
            m = fmaxf(m, red[(w2 * G + g) * 2]);
        M[g] = m;
    }
    float p[G], wsum[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
        // round to bf16 BEFORE summing so l matches the p that PV consumes
        p[g] = valid ? bf2f(f2bf(__expf(s[g] - M[g]))) : 0.f;
        wsum[g] = wave_sum(p[g]);
    }
    __syncthreads();  // red reuse (second slot written below)
    if (lane == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) red[(wid * G + g) * 2 + 1] = wsum[g];
    }
    // coalesced p write: [key][G] bf16 (p in [0,1]; bf16 halves the
    // G-proportional scratch traffic)
    {
        unsigned short* prow =
            p_out + ((((long)b * nkv + kvh) * C + chunk) * DEC_CHUNK +
                     threadIdx.x) * G;
#pragma unroll
        for (int g = 0; g < G; ++g) prow[g] = f2bf(p[g]);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        float* pml = part_ml + ((((long)b * nkv + kvh) * C + chunk) * G) * 2;
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float lsum = 0.f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2)
                lsum += red[(w2 * G + g) * 2 + 1];
            pml[g * 2] = M[g];
            pml[g * 2 + 1] = lsum;
        }
    }
    __syncthreads();  // red reused by the next chunk iteration
    }  // chunk loop
}


// MFMA variant of stage 1 (hd 64/128): the G dot products per key run on
// the matrix pipe as S^T tiles = K_tile(16 keys) x q^T(16 cols, G used).
// A-fragment rows map exactly to "lane reads its key row's 16 B segment",
// so K streams from HBM straight into MFMA operands — no LDS staging, no
// VALU fma chains, ~70 VGPRs -> high occupancy.
// Fragment maps (verified on HW by tests/test_ops_gpu.py::test_mfma_fragment_map):
//   A (16x32): lane = row (l&15), k = (l>>4)*8+e
//   B (32x16): lane = col (l&15), k = (l>>4)*8+e
//   C (16x16): lane = col (l&15), row = (l>>4)*4+r
template <int G, int HD, typename KVT>
__global__ __launch_bounds__(DEC_BLOCK) void attn_scores_mfma_kernel(
    const unsigned short* __restrict__ q,
    const KVT* __restrict__ k_cache,
    const int* __restrict__ block_table,
    const int* __restrict__ seq_lens,
    unsigned short* __restrict__ p_out,  // bf16
    float* __restrict__ part_ml,
    int nkv, int W, int bs, int C, long q_stride, float scale) {
    constexpr int KSTEPS = HD / 32;
    constexpr int TILES = WAVE / 16;  // 4 key-tiles of 16 per wave
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int L = seq_lens[b];
    if ((int)blockIdx.z * DEC_CHUNK >= L) return;
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;
    const int lg = lane >> 4;  // lane group 0..3
    const int li = lane & 15;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* q_s = reinterpret_cast<unsigned short*>(smem_raw);  // [G*HD]
    float* red = reinterpret_cast<float*>(smem_raw + ((G * HD * 2 + 15) & ~15));
    // red: [DEC_WAVES][G][2]

    for (int i = threadIdx.x; i < G * HD; i += DEC_BLOCK) {
        const int g = i / HD, d = i % HD;
        q_s[i] = f2bf(
            bf2f(q[(long)b * q_stride + (kvh * G + g) * (long)HD + d]) * scale);
    }
    __syncthreads();

    // B-fragments: col = q head (li), k = dim lg*8+e within each kstep
    using kfrag_t = typename kfrag_of<KVT>::type;
    kfrag_t bq[KSTEPS];
#pragma unroll
    for (int ks = 0; ks < KSTEPS; ++ks) {
        if (li < G) {
            bq[ks] = qfrag_from_lds<KVT>(q_s + li * HD + ks * 32 + lg * 8);
        } else {
            bq[ks] = kfrag_t{};
        }
    }

    const int* bt = block_table + (long)b * W;
    // loop chunks with stride gridDim.z: the q staging amortizes and the
    // per-chunk softmax tail overlaps other chunks' K bursts (single-chunk
    // blocks measured ~66% memory-idle from cold-start/tail phases)
    for (int chunk = blockIdx.z; chunk * DEC_CHUNK < L; chunk += gridDim.z) {
    const int start = chunk * DEC_CHUNK;
    const int wave_key0 = start + wid * WAVE;  // this wave covers 64 keys

    // scores: all TILES*KSTEPS A-fragments issue before any MFMA (probe-
    // style load batching: 16 independent 16 B loads in flight per wave)
    f32x4 acc[TILES];
    kfrag_t afr[TILES][KSTEPS];
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
        const int key = wave_key0 + t * 16 + li;  // A row = li
        const KVT* kr;
        if (key < L) {
            const int page = bt[key / bs];
            kr = k_cache + (((long)page * nkv + kvh) * bs + key % bs) * HD;
        } else {
            // safe dummy row; masked later via the key<L score mask
            kr = k_cache;
        }
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks)
            afr[t][ks] = load_kfrag<KVT>(kr + ks * 32 + lg * 8);
    }
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
        f32x4 c{0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int ks = 0; ks < KSTEPS; ++ks)
            c = mfma_kq(afr[t][ks], bq[ks], c);
        acc[t] = c;
    }

    // this lane holds scores for g = li, keys (t*16 + lg*4 + r)
    float sv[TILES * 4];
    float lmax = -1e30f;
#pragma unroll
    for (int t = 0; t < TILES; ++t) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int key = wave_key0 + t * 16 + lg * 4 + r;
            float v = (key < L) ? acc[t][r] : -1e30f;
            sv[t * 4 + r] = v;
            lmax = fmaxf(lmax, v);
        }
    }
    // reduce over the 4 lane groups holding the same g (stride-16 lanes)
    lmax = fmaxf(lmax, __shfl_xor(lmax, 16));
    lmax = fmaxf(lmax, __shfl_xor(lmax, 32));
    if (lg == 0 && li < G) red[(wid * G + li) * 2] = lmax;
    __syncthreads();
    float M = -1e30f;  // chunk max for this lane's g
    if (li < G) {
#pragma unroll
        for (int w2 = 0; w2 < DEC_WAVES; ++w2)
            M = fmaxf(M, red[(w2 * G + li) * 2]);
    }
    float lsum = 0.f;
#pragma unroll
    for (int i = 0; i < TILES * 4; ++i) {
        // round to bf16 BEFORE summing so l matches the p that PV consumes
        const float pv =
            (sv[i] <= -1e29f) ? 0.f : bf2f(f2bf(__expf(sv[i] - M)));
        sv[i] = pv;
        lsum += pv;
    }
    lsum += __shfl_xor(lsum, 16);
    lsum += __shfl_xor(lsum, 32);
    __syncthreads();  // red slot-0 reads done
    if (lg == 0 && li < G) red[(wid * G + li) * 2 + 1] = lsum;

    // p write-out: [key][G]; lanes with the same (t, r) write G consecutive
    // floats for 4 key-groups -> 4 x 64 B segments per instruction
    if (li < G) {
        unsigned short* pbase =
            p_out + (((long)b * nkv + kvh) * C + chunk) * DEC_CHUNK * G;
#pragma unroll
        for (int t = 0; t < TILES; ++t) {
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int key = wid * WAVE + t * 16 + lg * 4 + r;
                pbase[key * G + li] = f2bf(sv[t * 4 + r]);
            }
        }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
        float* pml = part_ml + ((((long)b * nkv + kvh) * C + chunk) * G) * 2;
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float ls = 0.f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2)
                ls += red[(w2 * G + g) * 2 + 1];
            float mg = -1e30f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2)
                mg = fmaxf(mg, red[(w2 * G + g) * 2]);
            pml[g * 2] = mg;
            pml[g * 2 + 1] = ls;
        }
    }
    __syncthreads();  // red reused by the next chunk iteration
    }  // chunk loop
}

// ----------------------------------------------------------- stage 2: PV
// When FOLD (launcher sets it iff gridDim.z == 1, i.e. every chunk of this
// (b, kvh) runs serially in THIS workgroup — always true at flagship batch
// sizes where B*nkv >= 4096), the kernel carries the running flash-decoding
// merge (M, l, acc) across chunks in registers and writes the normalized
// bf16 output directly: no part_o round-trip, no combine kernel.
template <int G, bool FOLD, typename KVT>
__global__ __launch_bounds__(DEC_BLOCK) void attn_pv_kernel(
    const KVT* __restrict__ v_cache,             // [nb, nkv, bs, hd]
    const int* __restrict__ block_table,
    const int* __restrict__ seq_lens,
    const unsigned short* __restrict__ p_in,     // [B,nkv,C,CHUNK,G] bf16
    float* __restrict__ part_o,                  // [B, nkv, C, G, hd]
    const float* __restrict__ part_ml,           // [B, nkv, C, G, 2]
    unsigned short* __restrict__ out,            // [B, nq, hd] (FOLD only)
    float* __restrict__ out_ml,                  // optional [B, nq, 2]:
                                                 // final (m, l) per q head —
                                                 // the cross-rank merge
                                                 // state for context
                                                 // parallelism (parallel/cp)
    int nkv, int W, int bs, int hd, int C) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int L = seq_lens[b];
    if ((int)blockIdx.z * DEC_CHUNK >= L) return;
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    float* p_s = reinterpret_cast<float*>(smem_raw);  // [CHUNK][G]
    long* voff_s = reinterpret_cast<long*>(p_s + DEC_CHUNK * G);  // [CHUNK]
    float* merge = reinterpret_cast<float*>(voff_s + DEC_CHUNK);  // [Wv][G][hd]
    // FOLD running state lives in LDS so the hot loop stays at ~64 VGPRs
    // (a register version measured 101 VGPRs -> 5 waves/SIMD). Each head
    // group g is OWNED by wave g%DEC_WAVES: the per-chunk merge + fold and
    // the final normalize run G-way parallel across waves instead of
    // serializing on wave 0 (the r1 kernel's measured latency tail), and
    // per-g slices stay single-writer so no extra barriers appear.
    float* run = merge + DEC_WAVES * G * hd;   // [G][hd] running acc
    float* runml = run + G * hd;               // [G][2] running (m, l)
    if (FOLD) {
        for (int g = wid; g < G; g += DEC_WAVES) {
            for (int i = lane; i < hd; i += WAVE) run[g * hd + i] = 0.f;
            if (lane == 0) {
                runml[g * 2] = -1e30f;
                runml[g * 2 + 1] = 0.f;
            }
        }
    }

    for (int chunk = blockIdx.z; chunk * DEC_CHUNK < L; chunk += gridDim.z) {
    const int start = chunk * DEC_CHUNK;
    // stage p for the whole chunk into LDS, widening bf16 -> fp32 so the
    // inner fma loop reads full-rate f32x4
    {
        const unsigned short* psrc =
            p_in + (((long)b * nkv + kvh) * C + chunk) * DEC_CHUNK * G;
        for (int i = threadIdx.x; i < DEC_CHUNK * G / 8; i += DEC_BLOCK) {
            float v[8];
            load_bf16x8(psrc + i * 8, v);
#pragma unroll
            for (int j = 0; j < 8; ++j) p_s[i * 8 + j] = v[j];
        }
    }
    // per-thread V row offset (one chained block-table load per key)
    {
        const int* bt = block_table + (long)b * W;
        const int key = start + threadIdx.x;
        long voff = 0;
        if (key < L) {
            const int page = bt[key / bs];
            voff = (((long)page * nkv + kvh) * bs + key % bs) * hd;
        }
        voff_s[threadIdx.x] = voff;
    }
    __syncthreads();

    // wave wid accumulates keys [wid*64, wid*64+64) of the chunk;
    // lane owns output dims (d0, d0+1)
    const int d0 = lane * 2;
    const int wave_start = wid * WAVE;
    const int nkeys = min(WAVE, L - (start + wave_start));
    float o0[G], o1[G];
#pragma unroll
    for (int g = 0; g < G; ++g) o0[g] = o1[g] = 0.f;
    if (d0 < hd && nkeys > 0) {
        const long* voffs = voff_s + wave_start;
        const float* pw = p_s + wave_start * G;
        int t = 0;
        for (; t + 32 <= nkeys; t += 32) {  // 32 V loads in flight
            kv_pair_t<KVT> vv[32];
#pragma unroll
            for (int j = 0; j < 32; ++j)
                vv[j] = *reinterpret_cast<const kv_pair_t<KVT>*>(
                    v_cache + voffs[t + j] + d0);
#pragma unroll
            for (int j = 0; j < 32; ++j) {
                const float2v vf = kv_pair_f32(vv[j]);
                const float* pt = pw + (t + j) * G;
#pragma unroll
                for (int g = 0; g < G; ++g) {
                    o0[g] = fmaf(pt[g], vf.x, o0[g]);
                    o1[g] = fmaf(pt[g], vf.y, o1[g]);
                }
            }
        }
        for (; t < nkeys; ++t) {
            const kv_pair_t<KVT> vv =
                *reinterpret_cast<const kv_pair_t<KVT>*>(
                    v_cache + voffs[t] + d0);
            const float2v vf = kv_pair_f32(vv);
            const float* pt = pw + t * G;
#pragma unroll
            for (int g = 0; g < G; ++g) {
                o0[g] = fmaf(pt[g], vf.x, o0[g]);
                o1[g] = fmaf(pt[g], vf.y, o1[g]);
            }
        }
    }

    // all waves used the same chunk max, so cross-wave merge is a plain sum
    float* mw = merge + wid * G * hd;
    if (d0 < hd) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            mw[g * hd + d0] = o0[g];
            mw[g * hd + d0 + 1] = o1[g];
        }
    }
    __syncthreads();
    if (d0 < hd) {
        const long base = (((long)b * nkv + kvh) * C + chunk) * G;
        for (int g = wid; g < G; g += DEC_WAVES) {
            float a0 = 0.f, a1 = 0.f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2) {
                a0 += merge[(w2 * G + g) * hd + d0];
                a1 += merge[(w2 * G + g) * hd + d0 + 1];
            }
            if (FOLD) {
                // online merge with this chunk's (m, l) from the scores pass
                const float mc = part_ml[(base + g) * 2];
                const float lc = part_ml[(base + g) * 2 + 1];
                const float mold = runml[g * 2];
                const float mn = fmaxf(mold, mc);
                const float alpha = __expf(mold - mn);
                const float beta = __expf(mc - mn);
                run[g * hd + d0] = run[g * hd + d0] * alpha + a0 * beta;
                run[g * hd + d0 + 1] =
                    run[g * hd + d0 + 1] * alpha + a1 * beta;
                if (lane == 0) {
                    runml[g * 2] = mn;
                    runml[g * 2 + 1] = runml[g * 2 + 1] * alpha + lc * beta;
                }
            } else {
                float* po = part_o + (base + g) * hd;
                po[d0] = a0;
                po[d0 + 1] = a1;
            }
        }
    }
    __syncthreads();  // LDS buffers reused by the next chunk iteration
    }  // chunk loop

    if (FOLD) {
        const int d0f = lane * 2;
        const int nq = nkv * G;
        if (d0f < hd) {
            for (int g = wid; g < G; g += DEC_WAVES) {
                const float lr = runml[g * 2 + 1];
                const float inv = (lr > 0.f) ? 1.f / lr : 0.f;
                unsigned short* orow =
                    out + ((long)b * nq + kvh * G + g) * hd;
                orow[d0f] = f2bf(run[g * hd + d0f] * inv);
                orow[d0f + 1] = f2bf(run[g * hd + d0f + 1] * inv);
            }
        }
        if (out_ml != nullptr && lane == 0) {
            for (int g = wid; g < G; g += DEC_WAVES) {
                float* mlrow = out_ml + ((long)b * nq + kvh * G + g) * 2;
                mlrow[0] = runml[g * 2];
                mlrow[1] = runml[g * 2 + 1];
            }
        }
    }
}

// ------------------------------------------- stage 2 (MFMA variant, hd 128)
// P.V on the matrix pipe: out[g][d] = sum_k P[g][k] V[k][d] as 16x16x32
// MFMA tiles (A = P rows staged transposed [g][key] in LDS, B = V^T staged
// [d][key] in LDS). The waves SPLIT THE 128 OUTPUT DIMS (2 dim-groups of 16
// each) and share one cooperative 32-key V subtile per iteration, so there
// is no cross-wave merge and the per-wave state is 2 C-fragments (8 VGPRs).
// Per lane per 32 keys this is ~25 ops vs ~420 for the VALU PV (2G-fma
// chains + G LDS reads per key) — the G-proportional tail the roadmap
// attributed the 4.2-4.7 TB/s plateau to.
// vt row stride 40 elems: 80 B rows keep ds_read_b128 16 B-aligned and hit
// 16 distinct banks at the 64-dword b128 modulus (bank math in the CDNA4
// guide §2); the b16 scatter writes are 2-way conflicted (acceptable).
#define PVM_PAD 8
#define PVM_STRIDE (32 + PVM_PAD)

template <int G>
__global__ __launch_bounds__(DEC_BLOCK) void attn_pv_mfma_kernel(
    const unsigned short* __restrict__ v_cache,  // [nb, nkv, bs, 128]
    const int* __restrict__ block_table,
    const int* __restrict__ seq_lens,
    const unsigned short* __restrict__ p_in,     // [B,nkv,C,CHUNK,G] bf16
    float* __restrict__ part_o,                  // [B, nkv, C, G, 128]
    int nkv, int W, int bs, int C) {
    constexpr int HD = 128;
    constexpr int PT_STRIDE = DEC_CHUNK + 8;
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int L = seq_lens[b];
    if ((int)blockIdx.z * DEC_CHUNK >= L) return;
    const int tid = threadIdx.x;
    const int lane = tid % WAVE;
    const int wid = tid / WAVE;
    const int lg = lane >> 4;
    const int li = lane & 15;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    unsigned short* p_t = reinterpret_cast<unsigned short*>(smem_raw);
    const int off_v = (G * PT_STRIDE * 2 + 15) & ~15;
    long* voff_s = reinterpret_cast<long*>(smem_raw + off_v);  // [CHUNK]
    unsigned short* vt =
        reinterpret_cast<unsigned short*>(smem_raw + off_v + DEC_CHUNK * 8);
    // vt: double-buffered [2][HD][PVM_STRIDE]

    for (int chunk = blockIdx.z; chunk * DEC_CHUNK < L; chunk += gridDim.z) {
        const int start = chunk * DEC_CHUNK;
        // stage p transposed ([key][G] -> [g][key]) and the V row offsets
        {
            const unsigned short* psrc =
                p_in + (((long)b * nkv + kvh) * C + chunk) * DEC_CHUNK * G;
            for (int i = tid; i < DEC_CHUNK * G; i += DEC_BLOCK)
                p_t[(i % G) * PT_STRIDE + i / G] = psrc[i];
            const int* btb = block_table + (long)b * W;
            const int key = start + tid;
            long vo = 0;
            if (key < L)
                vo = (((long)btb[key / bs] * nkv + kvh) * bs + key % bs) * HD;
            voff_s[tid] = vo;
        }
        __syncthreads();

        const int nkeys = min(DEC_CHUNK, L - start);
        const int nsub = (nkeys + 31) / 32;
        f32x4 acc[2];
        acc[0] = acc[1] = f32x4{0.f, 0.f, 0.f, 0.f};

        // Cooperative 32-key x 128-dim V subtiles, transposed through LDS.
        // T14 register staging (guide): loads for subtile t+2 issue BEFORE
        // the LDS write of t+1 waits on t+1's loads, so each load has a full
        // iteration (barrier + 2 MFMAs + epilogue traffic) to land. Two
        // register sets = 2 subtiles of load latency in flight per thread.
        // Thread t owns key t/8, dims [(t%8)*16, +16): two 16 B loads,
        // sixteen b16 column writes.
        const int st_key = tid >> 3;
        const int st_d = (tid & 7) * 16;
        short8 ra[2], rb[2];
#define PVM_LOAD(SUB, SET)                                                     \
    do {                                                                       \
        const int k_ = (SUB) * 32 + st_key;                                    \
        ra[SET] = short8{};                                                    \
        rb[SET] = short8{};                                                    \
        if (k_ < nkeys) {                                                      \
            const unsigned short* vr_ = v_cache + voff_s[k_] + st_d;           \
            ra[SET] = *reinterpret_cast<const short8*>(vr_);                   \
            rb[SET] = *reinterpret_cast<const short8*>(vr_ + 8);               \
        }                                                                      \
    } while (0)
#define PVM_WRITE(SET, BUF)                                                    \
    do {                                                                       \
        unsigned short* w_ = vt + (BUF) * HD * PVM_STRIDE + st_key;            \
        _Pragma("unroll") for (int e = 0; e < 8; ++e)                          \
            w_[(st_d + e) * PVM_STRIDE] = (unsigned short)ra[SET][e];          \
        _Pragma("unroll") for (int e = 0; e < 8; ++e)                          \
            w_[(st_d + 8 + e) * PVM_STRIDE] = (unsigned short)rb[SET][e];      \
    } while (0)

// one 32-key step: LDS-write reg set, prefetch 2 subtiles ahead, barrier,
// 2 MFMAs. SET/BUF are LITERALS so the compiler tracks which outstanding
// loads each write actually needs (a runtime-indexed set forced
// vmcnt(0) at every write: measured 2.1 TB/s vs 4.1)
#define PVM_STEP(SUB, SET)                                                     \
    do {                                                                       \
        PVM_WRITE(SET, SET);                                                   \
        if ((SUB) + 2 < nsub) PVM_LOAD((SUB) + 2, SET);                        \
        __syncthreads();                                                       \
        const unsigned short* vb_ = vt + (SET) * HD * PVM_STRIDE;              \
        const bf16x8 ap_ = *reinterpret_cast<const bf16x8*>(                   \
            p_t + prow + (SUB) * 32 + lg * 8);                                 \
        _Pragma("unroll") for (int dgi = 0; dgi < 2; ++dgi) {                  \
            const bf16x8 bv_ = *reinterpret_cast<const bf16x8*>(               \
                vb_ + ((wid * 2 + dgi) * 16 + li) * PVM_STRIDE + lg * 8);      \
            acc[dgi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                \
                ap_, bv_, acc[dgi], 0, 0, 0);                                  \
        }                                                                      \
    } while (0)

        PVM_LOAD(0, 0);
        if (nsub > 1) PVM_LOAD(1, 1);
        const int prow = (li < G ? li : G - 1) * PT_STRIDE;
        int sub = 0;
        for (; sub + 1 < nsub; sub += 2) {
            PVM_STEP(sub, 0);
            PVM_STEP(sub + 1, 1);
        }
        if (sub < nsub) PVM_STEP(sub, 0);
#undef PVM_STEP
#undef PVM_LOAD
#undef PVM_WRITE

        // C layout: lane holds rows g = lg*4 + r, col d = dgroup*16 + li
        const long base = (((long)b * nkv + kvh) * C + chunk) * G;
#pragma unroll
        for (int dgi = 0; dgi < 2; ++dgi) {
            const int d = (wid * 2 + dgi) * 16 + li;
#pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int g = lg * 4 + r;
                if (g < G) part_o[(base + g) * HD + d] = acc[dgi][r];
            }
        }
        __syncthreads();  // p_t / voff_s reused by the next chunk
    }
}

// ------------------------------------------------------ stage 3: combine
template <int G>
__global__ __launch_bounds__(DEC_BLOCK) void attn_decode_combine_kernel(
    const float* __restrict__ part_o,   // [B, nkv, C, G, hd]
    const float* __restrict__ part_ml,  // [B, nkv, C, G, 2]
    const int* __restrict__ seq_lens,
    unsigned short* __restrict__ out,   // [B, nq, hd]
    float* __restrict__ out_ml,         // optional [B, nq, 2] (see PV)
    int nkv, int hd, int C) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int g = blockIdx.z;  // one block per (seq, kv head, q head)
    const int nq = nkv * G;
    const int L = seq_lens[b];
    const int nc = (L + DEC_CHUNK - 1) / DEC_CHUNK;
    const long base = ((long)b * nkv + kvh) * C;

    // stage this head's (m, l) pairs once
    __shared__ float ml_s[128][2];  // C <= 128 (32768-token contexts)
    for (int c = threadIdx.x; c < nc; c += DEC_BLOCK) {
        ml_s[c][0] = part_ml[((base + c) * G + g) * 2];
        ml_s[c][1] = part_ml[((base + c) * G + g) * 2 + 1];
    }
    __syncthreads();
    float M = -1e30f;
    for (int c = 0; c < nc; ++c) M = fmaxf(M, ml_s[c][0]);
    float Ltot = 0.f;
    for (int c = 0; c < nc; ++c) Ltot += __expf(ml_s[c][0] - M) * ml_s[c][1];
    const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
    if (out_ml != nullptr && threadIdx.x == 0) {
        float* mlrow = out_ml + ((long)b * nq + kvh * G + g) * 2;
        mlrow[0] = M;
        mlrow[1] = Ltot;
    }

    for (int d = threadIdx.x; d < hd; d += DEC_BLOCK) {
        float acc = 0.f;
        for (int c = 0; c < nc; ++c)
            acc += __expf(ml_s[c][0] - M) *
                   part_o[((base + c) * G + g) * hd + d];
        out[((long)b * nq + kvh * G + g) * hd + d] = f2bf(acc * inv);
    }
}

extern "C" void launch_attn_decode(
    const unsigned short* q, const void* k_cache, const void* v_cache,
    const int* block_table, const int* seq_lens, unsigned short* p_buf,
    float* part_o, float* part_ml, unsigned short* out, float* out_ml,
    int B, int nkv,
    int G, int W, int bs, int hd, int C, long q_stride, float scale,
    int fp8, hipStream_t stream) {
    // enough blocks to fill the chip, but chunks loop within a block so the
    // cold-start/tail phases amortize over multiple 64 KB K/V bursts
    int Z = (4096 + B * nkv - 1) / (B * nkv);
    if (Z > C) Z = C;
    if (Z < 1) Z = 1;
    dim3 grid(B, nkv, Z);
    dim3 cgrid(B, nkv, G);
    const int smem_s = ((G * hd * 2 + 15) & ~15) + DEC_WAVES * G * 2 * 4;
    const int smem_pv = DEC_CHUNK * G * 4 + DEC_CHUNK * 8 +
                        DEC_WAVES * G * hd * 4 +
                        G * hd * 4 + G * 8;  // + FOLD running state
    const int smem_pvm = ((G * (DEC_CHUNK + 8) * 2 + 15) & ~15) +
                         DEC_CHUNK * 8 + 2 * hd * PVM_STRIDE * 2;
    // MFMA PV (attn_pv_mfma_kernel) measured 4.2 TB/s vs the VALU PV's
    // 4.5 at B768/G4: the barrier-paced LDS pipeline loses more latency
    // hiding than the matrix pipe saves (PV is latency-, not VALU-bound).
    // Kept as opt-in infrastructure for future tr_b16 staging work.
    static int pv_mfma = -1;
    if (pv_mfma < 0) {
        const char* e = getenv("BEE2BEE_PV_MFMA");
        pv_mfma = (e != nullptr && e[0] == '1') ? 1 : 0;
    }
#define LAUNCH_T(GG, KVT)                                                      \
    do {                                                                       \
        const KVT* kc = reinterpret_cast<const KVT*>(k_cache);                 \
        const KVT* vc = reinterpret_cast<const KVT*>(v_cache);                 \
        if (hd == 128)                                                         \
            hipLaunchKernelGGL((attn_scores_mfma_kernel<GG, 128, KVT>), grid,  \
                               dim3(DEC_BLOCK), smem_s, stream, q, kc,         \
                               block_table, seq_lens, p_buf, part_ml, nkv, W,  \
                               bs, C, q_stride, scale);                        \
        else if (hd == 64)                                                     \
            hipLaunchKernelGGL((attn_scores_mfma_kernel<GG, 64, KVT>), grid,   \
                               dim3(DEC_BLOCK), smem_s, stream, q, kc,         \
                               block_table, seq_lens, p_buf, part_ml, nkv, W,  \
                               bs, C, q_stride, scale);                        \
        else                                                                   \
            hipLaunchKernelGGL((attn_scores_kernel<GG, KVT>), grid,            \
                               dim3(DEC_BLOCK), smem_s, stream, q, kc,         \
                               block_table, seq_lens, p_buf, part_ml, nkv, W,  \
                               bs, hd, C, q_stride, scale);                    \
        if (hd == 128 && pv_mfma && !fp8) {                                    \
            hipLaunchKernelGGL(attn_pv_mfma_kernel<GG>, grid,                  \
                               dim3(DEC_BLOCK), smem_pvm, stream,              \
                               reinterpret_cast<const unsigned short*>(        \
                                   v_cache),                                   \
                               block_table, seq_lens, p_buf, part_o, nkv, W,   \
                               bs, C);                                         \
            hipLaunchKernelGGL(attn_decode_combine_kernel<GG>, cgrid,          \
                               dim3(DEC_BLOCK), 0, stream, part_o, part_ml,    \
                               seq_lens, out, out_ml, nkv, hd, C);             \
        } else if (Z == 1) {                                                   \
            hipLaunchKernelGGL((attn_pv_kernel<GG, true, KVT>), grid,          \
                               dim3(DEC_BLOCK), smem_pv, stream, vc,           \
                               block_table, seq_lens, p_buf, part_o, part_ml,  \
                               out, out_ml, nkv, W, bs, hd, C);                \
        } else {                                                               \
            hipLaunchKernelGGL((attn_pv_kernel<GG, false, KVT>), grid,         \
                               dim3(DEC_BLOCK), smem_pv, stream, vc,           \
                               block_table, seq_lens, p_buf, part_o,           \
                               part_ml, out, nullptr, nkv, W, bs, hd, C);      \
            hipLaunchKernelGGL(attn_decode_combine_kernel<GG>, cgrid,          \
                               dim3(DEC_BLOCK), 0, stream, part_o, part_ml,    \
                               seq_lens, out, out_ml, nkv, hd, C);             \
        }                                                                      \
    } while (0)
#define LAUNCH(GG)                                                             \
    do {                                                                       \
        if (fp8) LAUNCH_T(GG, unsigned char);                                  \
        else LAUNCH_T(GG, unsigned short);                                     \
    } while (0)
    switch (G) {
        case 1: LAUNCH(1); break;
        case 2: LAUNCH(2); break;
        case 3: LAUNCH(3); break;
        case 4: LAUNCH(4); break;
        case 5: LAUNCH(5); break;
        case 6: LAUNCH(6); break;
        case 7: LAUNCH(7); break;  // Qwen2.5-7B: 28 q / 4 kv heads
        case 8: LAUNCH(8); break;
        default: LAUNCH(16); break;
    }
#undef LAUNCH
#undef LAUNCH_T
}
