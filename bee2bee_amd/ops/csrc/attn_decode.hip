// Paged GQA decode attention — flash-decoding style, CDNA4.
//
// One workgroup (4 waves, 256 threads) per (sequence, kv-head). The G query
// heads of the group are processed together so the K/V rows are read from
// HBM exactly once — this kernel is the decode-throughput hot spot and is
// memory-bound: per key it moves 2*hd*2 bytes and does ~O(G*hd) VALU work,
// far under the VALU roof, so the design goal is clean 256 B coalesced
// reads of cache rows and no wasted bytes.
//
// Phase 1 (scores): lanes-over-keys — lane l owns key (chunk + l); it
//   streams the key row in 16 B pieces and accumulates G dot products
//   against q held in LDS. No cross-lane reduction per key (the classic
//   per-key shuffle-reduce is the naive CUDA shape; lanes-over-keys removes
//   it entirely).
// Phase 2 (PV): lanes-over-dims — lane owns 2 output dims; V rows are read
//   as 4 B/lane x 64 lanes = one coalesced 256 B row; P broadcast from LDS.
// Each wave keeps a private online-softmax state (m, l, o) over its own
// key chunks; the 4 waves merge once at the end through LDS.
//
// Replaces: the reference's decode path inside transformers.generate()
// (bee2bee/hf.py:84-108). Numerics reference: ops/reference.py attn_decode.
#include "common.h"

#define DEC_BLOCK 256
#define DEC_WAVES 4
// max G (query heads per kv head) supported by the templated loop
template <int G>
__global__ __launch_bounds__(DEC_BLOCK) void attn_decode_kernel(
    const unsigned short* __restrict__ q,        // [B, nq, hd]
    const unsigned short* __restrict__ k_cache,  // [nb, nkv, bs, hd]
    const unsigned short* __restrict__ v_cache,
    const int* __restrict__ block_table,         // [B, W]
    const int* __restrict__ seq_lens,            // [B]
    unsigned short* __restrict__ out,            // [B, nq, hd]
    int nkv, int W, int bs, int hd, long q_stride, float scale) {
    const int b = blockIdx.x;
    const int kvh = blockIdx.y;
    const int nq = nkv * G;
    const int L = seq_lens[b];
    const int lane = threadIdx.x % WAVE;
    const int wid = threadIdx.x / WAVE;

    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    float* q_s = reinterpret_cast<float*>(smem_raw);       // [G * hd]
    float* p_s = q_s + G * hd;                             // [DEC_WAVES][G][WAVE]
    float* merge = p_s + DEC_WAVES * G * WAVE;             // [DEC_WAVES][G*(hd+2)]

    // stage q (pre-scaled) into LDS
    for (int i = threadIdx.x; i < G * hd; i += DEC_BLOCK) {
        int g = i / hd, d = i % hd;
        q_s[i] = bf2f(q[(long)b * q_stride + (kvh * G + g) * (long)hd + d]) * scale;
    }
    __syncthreads();

    const int* bt = block_table + (long)b * W;
    float m[G], lsum[G], o0[G], o1[G];
#pragma unroll
    for (int g = 0; g < G; ++g) {
        m[g] = -1e30f;
        lsum[g] = 0.f;
        o0[g] = o1[g] = 0.f;
    }
    const int d0 = lane * 2;  // this lane's output dims (hd <= 128)
    float* my_p = p_s + (wid * G) * WAVE;

    for (int base = wid * WAVE; base < L; base += DEC_BLOCK) {
        const int key = base + lane;
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
        // online softmax update (per wave, all G heads)
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float cmax = wave_max(s[g]);
            float mn = fmaxf(m[g], cmax);
            float p = valid ? __expf(s[g] - mn) : 0.f;
            float alpha = __expf(m[g] - mn);
            m[g] = mn;
            lsum[g] = lsum[g] * alpha + wave_sum(p);
            o0[g] *= alpha;
            o1[g] *= alpha;
            my_p[g * WAVE + lane] = p;
        }
        // phase 2: accumulate O over this chunk's keys; lane owns dims d0,d0+1
        const int nkeys = min(WAVE, L - base);
        if (d0 < hd) {
            for (int t = 0; t < nkeys; ++t) {
                const int tkey = base + t;
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
    }

    // cross-wave merge through LDS: layout per wave: [G][hd + 2]
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
#pragma unroll
        for (int g = 0; g < G; ++g) {
            float M = -1e30f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2)
                M = fmaxf(M, merge[(w2 * G + g) * (hd + 2) + hd]);
            float Ltot = 0.f, acc0 = 0.f, acc1 = 0.f;
            for (int w2 = 0; w2 < DEC_WAVES; ++w2) {
                const float* row = merge + (w2 * G + g) * (hd + 2);
                const float f = __expf(row[hd] - M);
                Ltot += f * row[hd + 1];
                acc0 += f * row[d0];
                acc1 += f * row[d0 + 1];
            }
            const float inv = (Ltot > 0.f) ? 1.f / Ltot : 0.f;
            unsigned short* orow = out + ((long)b * nq + kvh * G + g) * hd;
            short2v res;
            res[0] = (short)f2bf(acc0 * inv);
            res[1] = (short)f2bf(acc1 * inv);
            *reinterpret_cast<short2v*>(orow + d0) = res;
        }
    }
}

extern "C" void launch_attn_decode(
    const unsigned short* q, const unsigned short* k_cache,
    const unsigned short* v_cache, const int* block_table,
    const int* seq_lens, unsigned short* out, int B, int nkv, int G, int W,
    int bs, int hd, long q_stride, float scale, hipStream_t stream) {
    dim3 grid(B, nkv);
    const int smem =
        (G * hd + DEC_WAVES * G * WAVE + DEC_WAVES * G * (hd + 2)) * 4;
#define LAUNCH(GG)                                                          \
    hipLaunchKernelGGL(attn_decode_kernel<GG>, grid, dim3(DEC_BLOCK), smem, \
                       stream, q, k_cache, v_cache, block_table, seq_lens,  \
                       out, nkv, W, bs, hd, q_stride, scale)
    switch (G) {
        case 1: LAUNCH(1); break;
        case 2: LAUNCH(2); break;
        case 4: LAUNCH(4); break;
        case 8: LAUNCH(8); break;
        default: LAUNCH(16); break;  // covers exotic group sizes
    }
#undef LAUNCH
}
