// Common device helpers for gfx950 (CDNA4) kernels.
// Wave size is 64 on CDNA — hard-coded per the CDNA HIP guide.
#pragma once
#include <hip/hip_runtime.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef unsigned short ushort_t;
typedef ushort_t ushort8_t __attribute__((ext_vector_type(8)));
typedef ushort_t ushort2_t __attribute__((ext_vector_type(2)));
typedef ushort_t ushort4_t __attribute__((ext_vector_type(4)));
typedef float float2_t __attribute__((ext_vector_type(2)));
typedef float float4_t __attribute__((ext_vector_type(4)));

// bf16 <-> f32 (bf16 = top 16 bits of f32; RNE on pack)
DEVINL float bf2f(ushort_t u) {
    unsigned int x = ((unsigned int)u) << 16;
    return __uint_as_float(x);
}
DEVINL ushort_t f2bf(float f) {
    unsigned int x = __float_as_uint(f);
    unsigned int r = (x + 0x7FFFu + ((x >> 16) & 1u)) >> 16;
    return (ushort_t)r;
}

// full-wave (64-lane) reductions via xor shuffles
DEVINL float wave_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    return v;
}
DEVINL float wave_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    return v;
}

#define HIP_CHECK_LAST()                                                              \
    do {                                                                              \
        hipError_t e_ = hipGetLastError();                                            \
        if (e_ != hipSuccess) {                                                       \
            TORCH_CHECK(false, "HIP kernel launch failed: ", hipGetErrorString(e_));  \
        }                                                                             \
    } while (0)
