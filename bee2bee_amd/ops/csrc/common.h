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
typedef __attribute__((ext_vector_type(8))) unsigned char uchar8;
typedef __attribute__((ext_vector_type(2))) unsigned char u8x2;

// --- OCP fp8 (e4m3) helpers: gfx950's v_cvt_* work on packed pairs.
// pack two floats into the low word of an int (2 x fp8 bytes)
__device__ __forceinline__ unsigned short f2fp8x2(float a, float b) {
    return (unsigned short)(__builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false) &
                            0xffff);
}
// unpack two fp8 bytes (in the low word) to two floats
__device__ __forceinline__ float2v fp8x2_2f(unsigned short w) {
    return __builtin_amdgcn_cvt_pk_f32_fp8((int)w, false);
}
// 8 fp8 bytes -> 8 floats
__device__ __forceinline__ void load_fp8x8(const unsigned char* p,
                                           float* out) {
    const unsigned int* pi = reinterpret_cast<const unsigned int*>(p);
    const unsigned int w0 = pi[0], w1 = pi[1];
    const float2v a = __builtin_amdgcn_cvt_pk_f32_fp8((int)w0, false);
    const float2v b = __builtin_amdgcn_cvt_pk_f32_fp8((int)w0, true);
    const float2v c = __builtin_amdgcn_cvt_pk_f32_fp8((int)w1, false);
    const float2v d = __builtin_amdgcn_cvt_pk_f32_fp8((int)w1, true);
    out[0] = a.x; out[1] = a.y; out[2] = b.x; out[3] = b.y;
    out[4] = c.x; out[5] = c.y; out[6] = d.x; out[7] = d.y;
}

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

// Debug-assert build (BEE2BEE_DEBUG_KERNELS=1 at build time -> hipcc gets
// -DBEE2BEE_DEBUG): bounds/invariant checks compile into the kernels and
// trap with file:line on violation. The release build compiles them away
// entirely (no branch, no register cost). This is the kernel-side
// sanitizer story: HIP has no compute-sanitizer equivalent on this stack,
// so invariants are asserted at the source level instead.
#ifdef BEE2BEE_DEBUG
#define BB_KASSERT(cond)                                                      \
    do {                                                                      \
        if (!(cond)) {                                                        \
            printf("BB_KASSERT failed %s:%d: %s (block %d,%d,%d thread %d)\n",\
                   __FILE__, __LINE__, #cond, (int)blockIdx.x,                \
                   (int)blockIdx.y, (int)blockIdx.z, (int)threadIdx.x);       \
            __builtin_trap();                                                 \
        }                                                                     \
    } while (0)
#else
#define BB_KASSERT(cond) do { } while (0)
#endif

#define HIP_CHECK_KERNEL()                                                    \
    do {                                                                      \
        hipError_t e_ = hipGetLastError();                                    \
        if (e_ != hipSuccess)                                                 \
            TORCH_CHECK(false, "HIP kernel launch failed: ",                  \
                        hipGetErrorString(e_));                               \
    } while (0)
