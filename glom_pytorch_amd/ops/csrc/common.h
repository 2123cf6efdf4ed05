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

// f32 -> bf16 via the hardware converter (v_cvt_pk_bf16_f32, RNE —
// identical rounding to PyTorch). The software RNE bit-twiddle this
// replaces cost ~7 VALU ops per element and dominated the epilogues.
__device__ __forceinline__ ushort_t f2bf(float f) {
    __hip_bfloat16 h(f);
    return *reinterpret_cast<ushort_t*>(&h);
}

// Transcendental-free GELU / GELU' (erf-accurate to ~1e-3 abs in fp32,
// below the bf16 output resolution of the GEMM epilogues). Measured: the
// v_exp/v_rcp chain of an exp-based erf cost ~270 us per 200M-element
// epilogue on gfx950; these are pure fma Horner chains at the full VALU
// rate. Both use that erf and gelu'(x)-0.5 are odd and saturated outside
// [-4,4] / [-5,5]: clamp, then x * P((x/X)^2).
__device__ __forceinline__ float fast_erf(float x) {
    const float C[11] = {
        1.128355365e+00f, -6.011598717e+00f, 2.859643773e+01f,
        -1.048367491e+02f, 2.930073200e+02f, -6.110858998e+02f,
        9.211419265e+02f, -9.628073236e+02f, 6.565095731e+02f,
        -2.612567565e+02f, 4.586479609e+01f};
    float xc = fminf(fmaxf(x, -4.0f), 4.0f);
    float w = xc * xc * 0.0625f;
    float p = C[10];
#pragma unroll
    for (int k = 9; k >= 0; k--) p = fmaf(p, w, C[k]);
    return xc * p;
}

// exact-erf GELU, matching nn.GELU() default within bf16 rounding
__device__ __forceinline__ float gelu_f(float x) {
    return 0.5f * x * (1.0f + fast_erf(x * 0.70710678118654752440f));
}

// W-wide GELU: the W independent Horner chains are interleaved so the
// scheduler never stalls on the dependent fma chain (the scalar form
// compiled to packed fmas separated by hazard nops).
template <int W>
__device__ __forceinline__ void gelu_f_vec(const float* __restrict__ x,
                                           float* __restrict__ y) {
    const float C[11] = {
        1.128355365e+00f, -6.011598717e+00f, 2.859643773e+01f,
        -1.048367491e+02f, 2.930073200e+02f, -6.110858998e+02f,
        9.211419265e+02f, -9.628073236e+02f, 6.565095731e+02f,
        -2.612567565e+02f, 4.586479609e+01f};
    float xc[W], w[W], p[W];
#pragma unroll
    for (int e = 0; e < W; e++) {
        float a = x[e] * 0.70710678118654752440f;
        a = fminf(fmaxf(a, -4.0f), 4.0f);
        xc[e] = a;
        w[e] = a * a * 0.0625f;
        p[e] = C[10];
    }
#pragma unroll
    for (int k = 9; k >= 0; k--)
#pragma unroll
        for (int e = 0; e < W; e++) p[e] = fmaf(p[e], w[e], C[k]);
#pragma unroll
    for (int e = 0; e < W; e++)
        y[e] = 0.5f * x[e] * (1.0f + xc[e] * p[e]);
}

template <int W>
__device__ __forceinline__ void gelu_grad_vec(const float* __restrict__ x,
                                              float* __restrict__ y) {
    const float C[13] = {
        7.978831877e-01f, -6.648536156e+00f, 3.737029049e+01f,
        -1.476795118e+02f, 4.415801975e+02f, -1.035120216e+03f,
        1.913875597e+03f, -2.750035158e+03f, 2.981664476e+03f,
        -2.336216301e+03f, 1.239185943e+03f, -3.960376139e+02f,
        5.736295785e+01f};
    float xc[W], w[W], p[W];
#pragma unroll
    for (int e = 0; e < W; e++) {
        float a = fminf(fmaxf(x[e], -5.0f), 5.0f);
        xc[e] = a;
        w[e] = a * a * 0.04f;
        p[e] = C[12];
    }
#pragma unroll
    for (int k = 11; k >= 0; k--)
#pragma unroll
        for (int e = 0; e < W; e++) p[e] = fmaf(p[e], w[e], C[k]);
#pragma unroll
    for (int e = 0; e < W; e++) y[e] = 0.5f + xc[e] * p[e];
}

// d/dx gelu(x) = Phi(x) + x * phi(x); g'(x) - 0.5 is odd in x
__device__ __forceinline__ float gelu_grad_f(float x) {
    const float C[13] = {
        7.978831877e-01f, -6.648536156e+00f, 3.737029049e+01f,
        -1.476795118e+02f, 4.415801975e+02f, -1.035120216e+03f,
        1.913875597e+03f, -2.750035158e+03f, 2.981664476e+03f,
        -2.336216301e+03f, 1.239185943e+03f, -3.960376139e+02f,
        5.736295785e+01f};
    float xc = fminf(fmaxf(x, -5.0f), 5.0f);
    float w = xc * xc * 0.04f;
    float p = C[12];
#pragma unroll
    for (int k = 11; k >= 0; k--) p = fmaf(p, w, C[k]);
    return 0.5f + xc * p;
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

