// Row-wise / elementwise CDNA4 kernels for the GLOM hot loop (gfx950):
//   - k_rnorm:        per-(b,n,l) inverse L2 norm of the level rows
//                     (fuses F.normalize of reference glom_pytorch.py:58)
//   - k_softmax_fwd:  masked row softmax over consensus scores
//                     (reference glom_pytorch.py:62-71: -5e-4 diagonal fill
//                     and -max non-local fill happen BEFORE the softmax)
//   - k_softmax_bwd:  row softmax backward + mask-zeroing + k-norm column
//                     scaling (emits both dS and dS*rnorm[j])
//   - k_knorm_combine: backward of k = x / max(||x||, eps) plus the final
//                     sum dlevels = dv + dq + dk of the attention backward
//   - k_mix_fwd/bwd:  level mixing (prev + bu + pad(td) + cons) / c_l with
//                     c = [4,..,4,3] (reference glom_pytorch.py:128-144)
// All inputs/outputs bf16; accumulation in f32. One wave owns one row for
// the row-wise kernels (scalar per-lane loads are coalesced across lanes).

#include "common.h"
#include "aux_kernels.h"

#define NTHREADS 256
#define WPB (NTHREADS / WAVE)

// levels (B,N,L,d) -> rnorm (B,L,N) f32, rnorm = 1 / max(||row||_2, 1e-12)
__global__ __launch_bounds__(NTHREADS) void k_rnorm(
        const ushort_t* __restrict__ levels, float* __restrict__ out,
        int B, int N, int L, int d) {
    long row = (long)blockIdx.x * WPB + threadIdx.x / WAVE;
    long total = (long)B * N * L;
    if (row >= total) return;
    int lane = threadIdx.x % WAVE;
    int l = row % L;
    long n = (row / L) % N;
    long b = row / ((long)N * L);
    const ushort_t* x = levels + row * d;
    float ss = 0.f;
    if (d % 8 == 0) {
        for (int q0 = lane * 8; q0 < d; q0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } t;
            t.v = *(const uint4v*)(x + q0);
#pragma unroll
            for (int e = 0; e < 8; e++) {
                float v = bf2f(t.u[e]);
                ss += v * v;
            }
        }
    } else {
        for (int q = lane; q < d; q += WAVE) {
            float v = bf2f(x[q]);
            ss += v * v;
        }
    }
    ss = wave_reduce_sum(ss);
    if (lane == 0)
        out[(b * L + l) * N + n] = 1.0f / fmaxf(sqrtf(ss), 1e-12f);
}

// scores/P: (P, N, N) bf16; one wave per row i of problem p.
__global__ __launch_bounds__(NTHREADS) void k_softmax_fwd(
        const ushort_t* __restrict__ scores, ushort_t* __restrict__ probs,
        const bool* __restrict__ nonlocal_mask, int nprob, int N,
        int self_mask) {
    long row = (long)blockIdx.x * WPB + threadIdx.x / WAVE;
    if (row >= (long)nprob * N) return;
    int lane = threadIdx.x % WAVE;
    int i = row % N;
    const ushort_t* s = scores + row * N;
    ushort_t* o = probs + row * N;
    const bool* m = nonlocal_mask ? nonlocal_mask + (long)i * N : nullptr;
    const float SELF = bf2f(f2bf(-5e-4f));
    const float NEG = -3.3895314e38f;  // -finfo(bf16).max

    if (N % 8 == 0) {
        // vectorized: each lane owns contiguous 8-element chunks; the row
        // (< 1024 values at the stretch config) is re-read from L1/L2
        float mx = -INFINITY, sum = 0.f;
#pragma unroll 1
        for (int j0 = lane * 8; j0 < N; j0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } t;
            t.v = *(const uint4v*)(s + j0);
#pragma unroll
            for (int e = 0; e < 8; e++) {
                float v = bf2f(t.u[e]);
                if (self_mask && j0 + e == i) v = SELF;
                if (m && m[j0 + e]) v = NEG;
                mx = fmaxf(mx, v);
            }
        }
        mx = wave_reduce_max(mx);
        mx = __shfl(mx, 0, WAVE);
#pragma unroll 1
        for (int j0 = lane * 8; j0 < N; j0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } t, ov;
            t.v = *(const uint4v*)(s + j0);
#pragma unroll
            for (int e = 0; e < 8; e++) {
                float v = bf2f(t.u[e]);
                if (self_mask && j0 + e == i) v = SELF;
                if (m && m[j0 + e]) v = NEG;
                float ex = __expf(v - mx);
                sum += ex;
                ov.u[e] = f2bf(ex);
            }
            *(uint4v*)(o + j0) = ov.v;   // unnormalized, rescaled below
        }
        sum = wave_reduce_sum(sum);
        sum = __shfl(sum, 0, WAVE);
        float inv = 1.0f / sum;
#pragma unroll 1
        for (int j0 = lane * 8; j0 < N; j0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } t;
            t.v = *(const uint4v*)(o + j0);
#pragma unroll
            for (int e = 0; e < 8; e++) t.u[e] = f2bf(bf2f(t.u[e]) * inv);
            *(uint4v*)(o + j0) = t.v;
        }
        return;
    }
    float mx = -INFINITY;
    for (int j = lane; j < N; j += WAVE) {
        float v = bf2f(s[j]);
        if (self_mask && j == i) v = SELF;
        if (m && m[j]) v = NEG;
        mx = fmaxf(mx, v);
    }
    mx = wave_reduce_max(mx);
    mx = __shfl(mx, 0, WAVE);
    float sum = 0.f;
    for (int j = lane; j < N; j += WAVE) {
        float v = bf2f(s[j]);
        if (self_mask && j == i) v = SELF;
        if (m && m[j]) v = NEG;
        sum += expf(v - mx);
    }
    sum = wave_reduce_sum(sum);
    sum = __shfl(sum, 0, WAVE);
    float inv = 1.0f / sum;
    for (int j = lane; j < N; j += WAVE) {
        float v = bf2f(s[j]);
        if (self_mask && j == i) v = SELF;
        if (m && m[j]) v = NEG;
        o[j] = f2bf(expf(v - mx) * inv);
    }
}

// dS[i,j] = alpha * P[i,j] * (dP[i,j] - sum_k P[i,k] dP[i,k]), masked -> 0;
// dSr[i,j] = dS[i,j] * rnorm[p,j].
__global__ __launch_bounds__(NTHREADS) void k_softmax_bwd(
        const ushort_t* __restrict__ P, const ushort_t* __restrict__ dP,
        const float* __restrict__ rnorm, ushort_t* __restrict__ dS,
        ushort_t* __restrict__ dSr, const bool* __restrict__ nonlocal_mask,
        int nprob, int N, int self_mask, float alpha) {
    long row = (long)blockIdx.x * WPB + threadIdx.x / WAVE;
    if (row >= (long)nprob * N) return;
    int lane = threadIdx.x % WAVE;
    int i = row % N;
    long p = row / N;
    const ushort_t* pr = P + row * N;
    const ushort_t* dpr = dP + row * N;
    const float* rn = rnorm + p * N;
    const bool* m = nonlocal_mask ? nonlocal_mask + (long)i * N : nullptr;

    float t = 0.f;
    if (N % 8 == 0) {
#pragma unroll 1
        for (int j0 = lane * 8; j0 < N; j0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } a, bb;
            a.v = *(const uint4v*)(pr + j0);
            bb.v = *(const uint4v*)(dpr + j0);
#pragma unroll
            for (int e = 0; e < 8; e++) t += bf2f(a.u[e]) * bf2f(bb.u[e]);
        }
        t = wave_reduce_sum(t);
        t = __shfl(t, 0, WAVE);
#pragma unroll 1
        for (int j0 = lane * 8; j0 < N; j0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } a, bb, o1, o2;
            a.v = *(const uint4v*)(pr + j0);
            bb.v = *(const uint4v*)(dpr + j0);
#pragma unroll
            for (int e = 0; e < 8; e++) {
                float v = alpha * bf2f(a.u[e]) * (bf2f(bb.u[e]) - t);
                if (self_mask && j0 + e == i) v = 0.f;
                if (m && m[j0 + e]) v = 0.f;
                o1.u[e] = f2bf(v);
                o2.u[e] = f2bf(v * rn[j0 + e]);
            }
            *(uint4v*)(dS + row * N + j0) = o1.v;
            *(uint4v*)(dSr + row * N + j0) = o2.v;
        }
        return;
    }
    for (int j = lane; j < N; j += WAVE) t += bf2f(pr[j]) * bf2f(dpr[j]);
    t = wave_reduce_sum(t);
    t = __shfl(t, 0, WAVE);
    for (int j = lane; j < N; j += WAVE) {
        float v = alpha * bf2f(pr[j]) * (bf2f(dpr[j]) - t);
        if (self_mask && j == i) v = 0.f;
        if (m && m[j]) v = 0.f;
        dS[row * N + j] = f2bf(v);
        dSr[row * N + j] = f2bf(v * rn[j]);
    }
}

// Backward of k = x * r, r = 1/max(||x||, eps), fused with the attention
// gradient sum: out(B,N,L,d) = dv + dq + r*(dkhat - x*r^2*(dkhat . x))
// (when the norm clamps at eps, the projection term vanishes).
// dkhat layout (B,L,N,d); dv,dq,out (B,N,L,d); rnorm (B,L,N).
__global__ __launch_bounds__(NTHREADS) void k_knorm_combine(
        const ushort_t* __restrict__ dkhat, const ushort_t* __restrict__ lev,
        const float* __restrict__ rnorm, const ushort_t* __restrict__ dv,
        const ushort_t* __restrict__ dq, ushort_t* __restrict__ out,
        int B, int N, int L, int d) {
    long row = (long)blockIdx.x * WPB + threadIdx.x / WAVE;
    long total = (long)B * N * L;
    if (row >= total) return;
    int lane = threadIdx.x % WAVE;
    int l = row % L;
    long n = (row / L) % N;
    long b = row / ((long)N * L);
    const ushort_t* dkh = dkhat + ((b * L + l) * N + n) * (long)d;
    const ushort_t* x = lev + row * d;
    float r = rnorm[(b * L + l) * N + n];
    bool clamped = r >= 0.999e12f;

    const bool vec = (d % 8 == 0);
    float dot = 0.f;
    if (!clamped) {
        if (vec) {
            for (int q0 = lane * 8; q0 < d; q0 += WAVE * 8) {
                union { uint4v v; ushort_t u[8]; } a, b;
                a.v = *(const uint4v*)(dkh + q0);
                b.v = *(const uint4v*)(x + q0);
#pragma unroll
                for (int e = 0; e < 8; e++)
                    dot += bf2f(a.u[e]) * bf2f(b.u[e]);
            }
        } else {
            for (int q = lane; q < d; q += WAVE)
                dot += bf2f(dkh[q]) * bf2f(x[q]);
        }
        dot = wave_reduce_sum(dot);
        dot = __shfl(dot, 0, WAVE);
    }
    float c = clamped ? 0.f : r * r * dot;
    if (vec) {
        for (int q0 = lane * 8; q0 < d; q0 += WAVE * 8) {
            union { uint4v v; ushort_t u[8]; } a, b, dvv, dqv, o;
            a.v = *(const uint4v*)(dkh + q0);
            b.v = *(const uint4v*)(x + q0);
            dvv.v = *(const uint4v*)(dv + row * d + q0);
            dqv.v = *(const uint4v*)(dq + row * d + q0);
#pragma unroll
            for (int e = 0; e < 8; e++) {
                float dk = r * (bf2f(a.u[e]) - bf2f(b.u[e]) * c);
                o.u[e] = f2bf(dk + bf2f(dvv.u[e]) + bf2f(dqv.u[e]));
            }
            *(uint4v*)(out + row * d + q0) = o.v;
        }
    } else {
        for (int q = lane; q < d; q += WAVE) {
            float dk = r * (bf2f(dkh[q]) - bf2f(x[q]) * c);
            out[row * d + q] =
                f2bf(dk + bf2f(dv[row * d + q]) + bf2f(dq[row * d + q]));
        }
    }
}

// out = (prev + bu + td_padded + cons) / c_l, elementwise over (B,N,L,d);
// td has L-1 level slots, the top slot contributes zero.
__global__ __launch_bounds__(NTHREADS) void k_mix_fwd(
        const ushort_t* __restrict__ prev, const ushort_t* __restrict__ bu,
        const ushort_t* __restrict__ td, const ushort_t* __restrict__ cons,
        ushort_t* __restrict__ out, long total, int L, int d) {
    long idx = ((long)blockIdx.x * NTHREADS + threadIdx.x) * 8;
    if (idx >= total) return;
    int q = idx % d;
    int l = (idx / d) % L;
    long bn = idx / ((long)d * L);
    union { uint4v v; ushort_t u[8]; } a, b, c4, t, o;
    a.v = *(const uint4v*)(prev + idx);
    b.v = *(const uint4v*)(bu + idx);
    c4.v = *(const uint4v*)(cons + idx);
    bool has_td = l < L - 1;
    if (has_td) t.v = *(const uint4v*)(td + (bn * (L - 1) + l) * (long)d + q);
    float c = has_td ? 4.0f : 3.0f;
#pragma unroll
    for (int e = 0; e < 8; e++) {
        float v = bf2f(a.u[e]) + bf2f(b.u[e]) + bf2f(c4.u[e]);
        if (has_td) v += bf2f(t.u[e]);
        o.u[e] = f2bf(v / c);
    }
    *(uint4v*)(out + idx) = o.v;
}

// dmix = dout / c_l (shared by prev/bu/cons); dtd = dout[..., :L-1, :] / 4
__global__ __launch_bounds__(NTHREADS) void k_mix_bwd(
        const ushort_t* __restrict__ dout, ushort_t* __restrict__ dmix,
        ushort_t* __restrict__ dtd, long total, int L, int d) {
    long idx = ((long)blockIdx.x * NTHREADS + threadIdx.x) * 8;
    if (idx >= total) return;
    int q = idx % d;
    int l = (idx / d) % L;
    long bn = idx / ((long)d * L);
    union { uint4v v; ushort_t u[8]; } g, o;
    g.v = *(const uint4v*)(dout + idx);
    bool has_td = l < L - 1;
    float c = has_td ? 4.0f : 3.0f;
#pragma unroll
    for (int e = 0; e < 8; e++) o.u[e] = f2bf(bf2f(g.u[e]) / c);
    *(uint4v*)(dmix + idx) = o.v;
    if (has_td) *(uint4v*)(dtd + (bn * (L - 1) + l) * (long)d + q) = o.v;
}

// td_in[b,n,g,:] = levels[b,n,g+1,:] + pos[n,:]  for g in 0..L-2
// (the top-down input of reference glom_pytorch.py:136, materialized once
// per iteration so the GEMM can stream it with plain 16B loads)
__global__ __launch_bounds__(NTHREADS) void k_add_pos(
        const ushort_t* __restrict__ levels, const ushort_t* __restrict__ pos,
        ushort_t* __restrict__ out, long total, int N, int L, int d) {
    long i8 = ((long)blockIdx.x * NTHREADS + threadIdx.x) * 8;
    if (i8 >= total) return;
    int G = L - 1;
    int q = i8 % d;
    int g = (i8 / d) % G;
    long bn = i8 / ((long)d * G);
    long n = bn % N;
    const ushort_t* lv = levels + (bn * L + g + 1) * (long)d + q;
    const ushort_t* pp = pos + n * (long)d + q;
    union { uint4v v; ushort_t u[8]; } a, b;
    a.v = *(const uint4v*)lv;
    b.v = *(const uint4v*)pp;
#pragma unroll
    for (int e = 0; e < 8; e++) a.u[e] = f2bf(bf2f(a.u[e]) + bf2f(b.u[e]));
    *(uint4v*)(out + i8) = a.v;
}

// ---------------- host launchers ----------------

static inline long cdiv(long a, long b) { return (a + b - 1) / b; }

void launch_add_pos(const void* levels, const void* pos, void* out,
                    long total, int N, int L, int d, hipStream_t s) {
    hipLaunchKernelGGL(k_add_pos, dim3(cdiv(total / 8, NTHREADS)),
                       dim3(NTHREADS), 0, s, (const ushort_t*)levels,
                       (const ushort_t*)pos, (ushort_t*)out, total, N, L, d);
}

void launch_rnorm(const void* levels, float* out, int B, int N, int L, int d,
                  hipStream_t s) {
    long rows = (long)B * N * L;
    hipLaunchKernelGGL(k_rnorm, dim3(cdiv(rows, WPB)), dim3(NTHREADS), 0, s,
                       (const ushort_t*)levels, out, B, N, L, d);
}

void launch_softmax_fwd(const void* scores, void* probs, const bool* mask,
                        int nprob, int N, int self_mask, hipStream_t s) {
    long rows = (long)nprob * N;
    hipLaunchKernelGGL(k_softmax_fwd, dim3(cdiv(rows, WPB)), dim3(NTHREADS),
                       0, s, (const ushort_t*)scores, (ushort_t*)probs, mask,
                       nprob, N, self_mask);
}

void launch_softmax_bwd(const void* P, const void* dP, const float* rnorm,
                        void* dS, void* dSr, const bool* mask, int nprob,
                        int N, int self_mask, float alpha, hipStream_t s) {
    long rows = (long)nprob * N;
    hipLaunchKernelGGL(k_softmax_bwd, dim3(cdiv(rows, WPB)), dim3(NTHREADS),
                       0, s, (const ushort_t*)P, (const ushort_t*)dP, rnorm,
                       (ushort_t*)dS, (ushort_t*)dSr, mask, nprob, N,
                       self_mask, alpha);
}

void launch_knorm_combine(const void* dkhat, const void* lev,
                          const float* rnorm, const void* dv, const void* dq,
                          void* out, int B, int N, int L, int d,
                          hipStream_t s) {
    long rows = (long)B * N * L;
    hipLaunchKernelGGL(k_knorm_combine, dim3(cdiv(rows, WPB)), dim3(NTHREADS),
                       0, s, (const ushort_t*)dkhat, (const ushort_t*)lev,
                       rnorm, (const ushort_t*)dv, (const ushort_t*)dq,
                       (ushort_t*)out, B, N, L, d);
}

void launch_mix_fwd(const void* prev, const void* bu, const void* td,
                    const void* cons, void* out, long total, int L, int d,
                    hipStream_t s) {
    hipLaunchKernelGGL(k_mix_fwd, dim3(cdiv(total / 8, NTHREADS)), dim3(NTHREADS),
                       0, s, (const ushort_t*)prev, (const ushort_t*)bu,
                       (const ushort_t*)td, (const ushort_t*)cons,
                       (ushort_t*)out, total, L, d);
}

void launch_mix_bwd(const void* dout, void* dmix, void* dtd, long total,
                    int L, int d, hipStream_t s) {
    hipLaunchKernelGGL(k_mix_bwd, dim3(cdiv(total / 8, NTHREADS)), dim3(NTHREADS),
                       0, s, (const ushort_t*)dout, (ushort_t*)dmix,
                       (ushort_t*)dtd, total, L, d);
}

// out = a + b + c + d, elementwise bf16 (the four levels-gradient
// contributions of one GLOM iteration summed in one pass)
__global__ __launch_bounds__(NTHREADS) void k_add4(
        const ushort_t* __restrict__ a, const ushort_t* __restrict__ b,
        const ushort_t* __restrict__ c, const ushort_t* __restrict__ d,
        ushort_t* __restrict__ out, long total) {
    long i8 = ((long)blockIdx.x * NTHREADS + threadIdx.x) * 8;
    if (i8 >= total) return;
    union { uint4v v; ushort_t u[8]; } va, vb, vc, vd, o;
    va.v = *(const uint4v*)(a + i8);
    vb.v = *(const uint4v*)(b + i8);
    vc.v = *(const uint4v*)(c + i8);
    vd.v = *(const uint4v*)(d + i8);
#pragma unroll
    for (int e = 0; e < 8; e++)
        o.u[e] = f2bf(bf2f(va.u[e]) + bf2f(vb.u[e]) + bf2f(vc.u[e])
                      + bf2f(vd.u[e]));
    *(uint4v*)(out + i8) = o.v;
}

void launch_add4(const void* a, const void* b, const void* c, const void* d,
                 void* out, long total, hipStream_t s) {
    hipLaunchKernelGGL(k_add4, dim3(cdiv(total / 8, NTHREADS)),
                       dim3(NTHREADS), 0, s, (const ushort_t*)a,
                       (const ushort_t*)b, (const ushort_t*)c,
                       (const ushort_t*)d, (ushort_t*)out, total);
}

// out = gelu(in), elementwise bf16 (the up-projection's activation pass;
// running this as its own bandwidth-bound kernel beats computing it in the
// GEMM epilogue, which runs at 1 block/CU with nothing to overlap)
__global__ __launch_bounds__(NTHREADS) void k_gelu(
        const ushort_t* __restrict__ in, ushort_t* __restrict__ out,
        long total) {
    long i8 = ((long)blockIdx.x * NTHREADS + threadIdx.x) * 8;
    if (i8 >= total) return;
    union { uint4v v; ushort_t u[8]; } x, o;
    x.v = *(const uint4v*)(in + i8);
    float xin[8], yv[8];
#pragma unroll
    for (int e = 0; e < 8; e++) xin[e] = bf2f(x.u[e]);
    gelu_f_vec<8>(xin, yv);
#pragma unroll
    for (int e = 0; e < 8; e++) o.u[e] = f2bf(yv[e]);
    *(uint4v*)(out + i8) = o.v;
}

void launch_gelu(const void* in, void* out, long total, hipStream_t s) {
    hipLaunchKernelGGL(k_gelu, dim3(cdiv(total / 8, NTHREADS)),
                       dim3(NTHREADS), 0, s, (const ushort_t*)in,
                       (ushort_t*)out, total);
}
