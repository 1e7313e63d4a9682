// Paged GQA decode attention — flash-decoding with key-chunk splitting.
//
// The decode step is memory-bound on KV reads (per key: 2*hd*2 B moved,
// ~O(G*hd) VALU — far under the VALU roof). The first version used one
// workgroup per (seq, kv-head): at B=64, nkv=8 that is 512 workgroups = 2
// per CU = 2 waves/SIMD — not enough latency hiding (measured 178 us/layer
// vs the ~43 us HBM roofline). This version splits the key axis into
// 256-key chunks, one workgroup per (seq, kv-head, chunk), each writing an
// unnormalized partial (m, l, o) that a small combine kernel merges — the
// MI355X needs >>256 workgroups to fill its 8 XCDs.
//
// Phase 1 (scores): lanes-over-keys — lane l owns one key; it streams the
//   key row in 16 B pieces and accumulates G dot products against q held
//   in LDS (no per-key cross-lane reduction).
// Phase 2 (PV): lanes-over-dims — lane owns 2 output dims; V rows read as
//   4 B/lane x 64 lanes = one coalesced 256 B row; P broadcast from LDS.
//
// Replaces: the reference's decode path inside transformers.generate()
// (bee2bee/hf.py:84-108). Numerics reference: ops/reference.py attn_decode.
#include "common.h"

#define DEC_BLOCK 256
#define DEC_WAVES 4
#define DEC_CHUNK 256  // keys per workgroup (DEC_WAVES x 64)

template <int G>
__global__ __launch_bounds__(DEC_BLOCK) void attn_decode_chunk_kernel(
    const unsigned short* __restrict__ q,        // [B, nq, hd] (strided)
    const unsigned short* __restrict__ k_cache,  // [nb, nkv, bs, hd]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_table,         // [B, W]
    const int* __restrict__ seq_lens,            // [B]
    float* __restrict__ part_o,                  // [B, nkv, C, G, hd]
    float* __restrict__ part_ml,                 // [B, nkv, C, G, 2]
    int nkv, int W, int bs, int hd, int C, long q_stride, float scale) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int chunk = blockIdx.z;
    const int L = seq_lens[b];
    const int start = chunk * DEC_CHUNK;
    if (start >= L) return;  // inactive chunk: combine never reads it
    const int nq = nkv * G;
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    float* q_s = reinterpret_cast<float*>(smem_raw);  // [G * hd]
    float* p_s = q_s + G * hd;                        // [DEC_WAVES][G][WAVE]
    float* merge = p_s + DEC_WAVES * G * WAVE;        // [DEC_WAVES][G][hd+2]

    for (int i = threadIdx.x; i < G * hd; i += DEC_BLOCK) {
        const int g = i / hd, d = i % hd;
        q_s[i] = bf2f(q[(long)b * q_stride + (kvh * G + g) * (long)hd + d]) * scale;
    }
    __syncthreads();

    const int* bt = block_table + (long)b * W;
    const int key = start + wid * WAVE + lane;  // one key per lane
    const bool valid = key < L;

    float s[G];
#pragma unroll
    for (int g = 0; g < G; ++g) s[g] = -1e30f;
    if (valid) {
        const int page = bt[key / bs];
        const unsigned short* kr =
            k_cache + (((long)page * nkv + kvh) * bs + key % bs) * hd;
#pragma unroll
        for (int g = 0; g < G; ++g) s[g] = 0.f;
        for (int d = 0; d < hd; d += 8) {
            float kv[8];
            load_bf16x8(kr + d, kv);
#pragma unroll
            for (int g = 0; g < G; ++g) {
                const float* qg = q_s + g * hd + d;
#pragma unroll
                for (int j = 0; j < 8; ++j) s[g] = fmaf(kv[j], qg[j], s[g]);
            }
        }
    }

    // per-wave softmax over this wave's 64 keys
    float m[G], lsum[G];
    float* my_p = p_s + (wid * G) * WAVE;
#pragma unroll
    for (int g = 0; g < G; ++g) {
        m[g] = wave_max(s[g]);
        const float p = valid ? __expf(s[g] - m[g]) : 0.f;
        lsum[g] = wave_sum(p);
        my_p[g * WAVE + lane] = p;
    }

    // phase 2: lane owns dims (d0, d0+1); iterate this wave's keys
    const int d0 = lane * 2;
    const int wave_start = start + wid * WAVE;
    const int nkeys = min(WAVE, L - wave_start);
    float o0[G], o1[G];
#pragma unroll
    for (int g = 0; g < G; ++g) o0[g] = o1[g] = 0.f;
    if (d0 < hd && nkeys > 0) {
        for (int t = 0; t < nkeys; ++t) {
            const int tkey = wave_start + t;
            const int page = bt[tkey / bs];
            const unsigned short* vr =
                v_cache + (((long)page * nkv + kvh) * bs + tkey % bs) * hd;
            const short2v vv = *reinterpret_cast<const short2v*>(vr + d0);
            const float v0 = bf2f((unsigned short)vv[0]);
            const float v1 = bf2f((unsigned short)vv[1]);
#pragma unroll
            for (int g = 0; g < G; ++g) {
                const float p = my_p[g * WAVE + t];
                o0[g] = fmaf(p, v0, o0[g]);
                o1[g] = fmaf(p, v1, o1[g]);
            }
        }
    }

    // cross-wave merge through LDS -> one partial per chunk
    float* mw = merge + wid * G * (hd + 2);
    if (d0 < hd) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            mw[g * (hd + 2) + d0] = o0[g];
            mw[g * (hd + 2) + d0 + 1] = o1[g];
        }
    }
    if (lane == 0) {
#pragma unroll
        for (int g = 0; g < G; ++g) {
            mw[g * (hd + 2) + hd] = m[g];
            mw[g * (hd + 2) + hd + 1] = lsum[g];
        }
    }
    __syncthreads();
    if (wid == 0 && d0 < hd) {
        const long base = (((long)b * nkv + kvh) * C + chunk) * G;
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float M = -1e30f;
#pragma unroll
            for (int w2 = 0; w2 < DEC_WAVES; ++w2)
                M = fmaxf(M, merge[(w2 * G + g) * (hd + 2) + hd]);
            float Ltot = 0.f, acc0 = 0.f, acc1 = 0.f;
#pragma unroll
            for (int w2 = 0; w2 < DEC_WAVES; ++w2) {
                const float* row = merge + (w2 * G + g) * (hd + 2);
                const float f = __expf(row[hd] - M);
                Ltot += f * row[hd + 1];
                acc0 += f * row[d0];
                acc1 += f * row[d0 + 1];
            }
            float* po = part_o + (base + g) * hd;
            po[d0] = acc0;
            po[d0 + 1] = acc1;
            if (lane == 0) {
                float* pml = part_ml + (base + g) * 2;
                pml[0] = M;
                pml[1] = Ltot;
            }
        }
    }
}

// merge the per-chunk partials into the final normalized output
template <int G>
__global__ __launch_bounds__(DEC_BLOCK) void attn_decode_combine_kernel(
    const float* __restrict__ part_o,   // [B, nkv, C, G, hd]
    const float* __restrict__ part_ml,  // [B, nkv, C, G, 2]
    const int* __restrict__ seq_lens,
    unsigned short* __restrict__ out,   // [B, nq, hd]
    int nkv, int hd, int C) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int nq = nkv * G;
    const int L = seq_lens[b];
    const int nc = (L + DEC_CHUNK - 1) / DEC_CHUNK;
    const long base = ((long)b * nkv + kvh) * C;

    // threads cover (g, d) pairs
    for (int i = threadIdx.x; i < G * hd; i += DEC_BLOCK) {
        const int g = i / hd, d = i % hd;
        float M = -1e30f;
        for (int c = 0; c < nc; ++c)
            M = fmaxf(M, part_ml[((base + c) * G + g) * 2]);
        float Ltot = 0.f, acc = 0.f;
        for (int c = 0; c < nc; ++c) {
            const float* ml = part_ml + ((base + c) * G + g) * 2;
            const float f = __expf(ml[0] - M);
            Ltot += f * ml[1];
            acc += f * part_o[((base + c) * G + g) * hd + d];
        }
        const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
        out[((long)b * nq + kvh * G + g) * hd + d] = f2bf(acc * inv);
    }
}

extern "C" void launch_attn_decode(
    const unsigned short* q, const unsigned short* k_cache,
    const unsigned short* v_cache, const int* block_table,
    const int* seq_lens, float* part_o, float* part_ml,
    unsigned short* out, int B, int nkv, int G, int W, int bs, int hd, int C,
    long q_stride, float scale, hipStream_t stream) {
    dim3 grid(B, nkv, C);
    dim3 cgrid(B, nkv);
    const int smem =
        (G * hd + DEC_WAVES * G * WAVE + DEC_WAVES * G * (hd + 2)) * 4;
#define LAUNCH(GG)                                                             \
    do {                                                                       \
        hipLaunchKernelGGL(attn_decode_chunk_kernel<GG>, grid,                 \
                           dim3(DEC_BLOCK), smem, stream, q, k_cache,          \
                           v_cache, block_table, seq_lens, part_o, part_ml,    \
                           nkv, W, bs, hd, C, q_stride, scale);                \
        hipLaunchKernelGGL(attn_decode_combine_kernel<GG>, cgrid,              \
                           dim3(DEC_BLOCK), 0, stream, part_o, part_ml,        \
                           seq_lens, out, nkv, hd, C);                         \
    } while (0)
    switch (G) {
        case 1: LAUNCH(1); break;
        case 2: LAUNCH(2); break;
        case 4: LAUNCH(4); break;
        case 8: LAUNCH(8); break;
        default: LAUNCH(16); break;
    }
#undef LAUNCH
}
