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

// Fast erf (Abramowitz & Stegun 7.1.26, max abs err 1.5e-7 — below bf16
// resolution): one v_rcp + one v_exp + 5 fma instead of the branchy libm
// erff, which dominated the GEMM epilogues at 200M gelu evals per launch.
__device__ __forceinline__ float fast_erf(float x) {
    float ax = fabsf(x);
    float t = __builtin_amdgcn_rcpf(1.0f + 0.3275911f * ax);
    float poly = t * (0.254829592f
                      + t * (-0.284496736f
                             + t * (1.421413741f
                                    + t * (-1.453152027f
                                           + t * 1.061405429f))));
    float y = 1.0f - poly * __expf(-ax * ax);
    return copysignf(y, x);
}

// exact-erf GELU, matching nn.GELU() default to ~1e-7
__device__ __forceinline__ float gelu_f(float x) {
    return 0.5f * x * (1.0f + fast_erf(x * 0.70710678118654752440f));
}

// d/dx gelu(x) = Phi(x) + x * phi(x)
__device__ __forceinline__ float gelu_grad_f(float x) {
    const float inv_sqrt2 = 0.70710678118654752440f;
    const float inv_sqrt2pi = 0.39894228040143267794f;
    float cdf = 0.5f * (1.0f + fast_erf(x * inv_sqrt2));
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

