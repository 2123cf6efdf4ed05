// Common device helpers for the glom_pytorch_amd CDNA4 (gfx950) kernels.
// Target: AMD Instinct MI355X only — wave64, MFMA bf16 16x16x32, 160 KiB LDS.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef short short8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef unsigned int uint4v __attribute__((ext_vector_type(4)));

typedef unsigned short ushort_t;

__device__ __forceinline__ float bf2f(ushort_t u) {
    union { unsigned int i; float f; } v;
    v.i = ((unsigned int)u) << 16;
    return v.f;
}

// round-to-nearest-even f32 -> bf16, matching PyTorch's conversion
__device__ __forceinline__ ushort_t f2bf(float f) {
    union { float f; unsigned int i; } v;
    v.f = f;
    if ((v.i & 0x7fffffffu) > 0x7f800000u) return (ushort_t)0x7fc0; // NaN
    unsigned int lsb = (v.i >> 16) & 1u;
    v.i += 0x7fffu + lsb;
    return (ushort_t)(v.i >> 16);
}

// exact (erf) GELU, as used by nn.GELU() default
__device__ __forceinline__ float gelu_f(float x) {
    return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

// d/dx gelu(x) = Phi(x) + x * phi(x)
__device__ __forceinline__ float gelu_grad_f(float x) {
    const float inv_sqrt2 = 0.70710678118654752440f;
    const float inv_sqrt2pi = 0.39894228040143267794f;
    float cdf = 0.5f * (1.0f + erff(x * inv_sqrt2));
    float pdf = inv_sqrt2pi * __expf(-0.5f * x * x);
    return cdf + x * pdf;
}

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, WAVE);
    return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, WAVE));
    return v;
}

