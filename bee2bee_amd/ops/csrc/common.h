// Shared helpers for the bee2bee_amd CDNA4 (gfx950) kernels.
//
// Target: MI355X only — wave64, 256 CUs, 160 KiB LDS/CU, HBM3E ~6.3 TB/s
// achievable. No CUDA-compat paths, no multi-backend dispatch.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(2))) short short2v;
typedef __attribute__((ext_vector_type(2))) float float2v;

// bf16 (as ushort bits) <-> float
__device__ __forceinline__ float bf2f(unsigned short u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

__device__ __forceinline__ unsigned short f2bf(float f) {
    union { unsigned int i; float f; } v;
    v.f = f;
    // round-to-nearest-even
    unsigned int rounding = 0x7FFF + ((v.i >> 16) & 1);
    return (unsigned short)((v.i + rounding) >> 16);
}

// load 8 bf16 (16 B) and expand to 8 floats
__device__ __forceinline__ void load_bf16x8(const unsigned short* p, float* out) {
    short8 raw = *reinterpret_cast<const short8*>(p);
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = bf2f((unsigned short)raw[j]);
}

// store 8 floats as 8 bf16 (16 B)
__device__ __forceinline__ void store_bf16x8(unsigned short* p, const float* in) {
    short8 raw;
#pragma unroll
    for (int j = 0; j < 8; ++j) raw[j] = (short)f2bf(in[j]);
    *reinterpret_cast<short8*>(p) = raw;
}

// full-wave (64-lane) reductions via xor shuffles
__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
    return v;
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off);
    return v;
}

#define DEV_INLINE __device__ __forceinline__

#define HIP_CHECK_KERNEL()                                                    \
    do {                                                                      \
        hipError_t e_ = hipGetLastError();                                    \
        if (e_ != hipSuccess)                                                 \
            TORCH_CHECK(false, "HIP kernel launch failed: ",                  \
                        hipGetErrorString(e_));                               \
    } while (0)
