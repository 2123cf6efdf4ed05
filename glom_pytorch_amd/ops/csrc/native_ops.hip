// Native-mandate kernels (gfx950):
//   - k_patchify / k_unpatchify: the reference's
//     Rearrange('b c (h p1) (w p2) -> b (h w) (p1 p2 c)') and its
//     transpose (reference glom_pytorch.py:95), K-padded so the patch
//     embedding GEMM runs on the fast NT kernels (zero columns contribute
//     zero to the products).
//   - k_pad_cols / k_slice_cols: row-wise K-padding of the (dim, p^2*3)
//     embed weight and the inverse slice of its gradient.
//   - k_colsum: deterministic two-stage column sum over the row axis
//     (bias gradients dB1/dB2 - replaces the ATen .sum() reductions).
//   - k_dpos: positional-embedding gradient, sum of the top-down input
//     grads over batch and levels 1..L-1 (reference glom_pytorch.py:136).
//   - k_zero_slice: zero one level slice of a (B,N,L,d) tensor (the one
//     slice the scattered grouped-GEMM backward never writes).
//   - k_grad_norm_* / k_adamw: fused bf16-grad -> fp32-master AdamW with
//     global-norm clipping; replaces the per-step Python master-weight
//     copy loop + foreach-AdamW + copy-back (torch semantics preserved).
// All tensor I/O bf16 unless noted; accumulation f32.

#include "common.h"
#include "native_ops.h"

#define NT256 256

__host__ __device__ static inline long cdivl(long a, long b) {
    return (a + b - 1) / b;
}

// --------------------------------------------------------------------- //
// patch embedding data movement

// X[b, n, k] = img[b, c, h*P+p1, w*P+p2] for k = (p1*P + p2)*C + c < P*P*C,
// else 0 (K padded to Kp for /8-aligned GEMM row strides).
__global__ __launch_bounds__(NT256) void k_patchify(
        const ushort_t* __restrict__ img, ushort_t* __restrict__ X,
        int B, int C, int H, int W, int P, int Kp) {
    const int S = W / P;               // patches per row
    const int K0 = P * P * C;
    long idx = (long)blockIdx.x * NT256 + threadIdx.x;
    long total = (long)B * (H / P) * S * Kp;
    if (idx >= total) return;
    int k = idx % Kp;
    long n = (idx / Kp) % ((long)(H / P) * S);
    long b = idx / ((long)Kp * (H / P) * S);
    ushort_t v = 0;
    if (k < K0) {
        int c = k % C;
        int p2 = (k / C) % P;
        int p1 = k / (C * P);
        int h = n / S, w = n % S;
        v = img[((b * C + c) * (long)H + h * P + p1) * W + w * P + p2];
    }
    X[idx] = v;
}

// dImg[b, c, y, x] = dX[b, (y/P)*S + x/P, ((y%P)*P + x%P)*C + c]
__global__ __launch_bounds__(NT256) void k_unpatchify(
        const ushort_t* __restrict__ dX, ushort_t* __restrict__ dImg,
        int B, int C, int H, int W, int P, int Kp) {
    const int S = W / P;
    long idx = (long)blockIdx.x * NT256 + threadIdx.x;
    long total = (long)B * C * H * W;
    if (idx >= total) return;
    int x = idx % W;
    int y = (idx / W) % H;
    int c = (idx / ((long)W * H)) % C;
    long b = idx / ((long)W * H * C);
    long n = (long)(y / P) * S + x / P;
    int k = ((y % P) * P + x % P) * C + c;
    dImg[idx] = dX[(b * ((long)(H / P) * S) + n) * Kp + k];
}

// dst (rows, Kp) = src (rows, Kin) zero-padded on the column axis
__global__ __launch_bounds__(NT256) void k_pad_cols(
        const ushort_t* __restrict__ src, ushort_t* __restrict__ dst,
        long rows, int Kin, int Kp) {
    long idx = (long)blockIdx.x * NT256 + threadIdx.x;
    if (idx >= rows * Kp) return;
    int k = idx % Kp;
    long r = idx / Kp;
    dst[idx] = (k < Kin) ? src[r * Kin + k] : (ushort_t)0;
}

// dst (rows, Kin) = src (rows, Kp)[:, :Kin]
__global__ __launch_bounds__(NT256) void k_slice_cols(
        const ushort_t* __restrict__ src, ushort_t* __restrict__ dst,
        long rows, int Kp, int Kin) {
    long idx = (long)blockIdx.x * NT256 + threadIdx.x;
    if (idx >= rows * Kin) return;
    int k = idx % Kin;
    long r = idx / Kin;
    dst[idx] = src[r * Kp + k];
}

void launch_patchify(const void* img, void* X, int B, int C, int H, int W,
                     int P, int Kp, hipStream_t s) {
    long total = (long)B * (H / P) * (W / P) * Kp;
    hipLaunchKernelGGL(k_patchify, dim3(cdivl(total, NT256)), dim3(NT256),
                       0, s, (const ushort_t*)img, (ushort_t*)X, B, C, H, W,
                       P, Kp);
}

void launch_unpatchify(const void* dX, void* dImg, int B, int C, int H,
                       int W, int P, int Kp, hipStream_t s) {
    long total = (long)B * C * H * W;
    hipLaunchKernelGGL(k_unpatchify, dim3(cdivl(total, NT256)), dim3(NT256),
                       0, s, (const ushort_t*)dX, (ushort_t*)dImg, B, C, H,
                       W, P, Kp);
}

void launch_pad_cols(const void* src, void* dst, long rows, int Kin, int Kp,
                     hipStream_t s) {
    hipLaunchKernelGGL(k_pad_cols, dim3(cdivl(rows * Kp, NT256)),
                       dim3(NT256), 0, s, (const ushort_t*)src,
                       (ushort_t*)dst, rows, Kin, Kp);
}

void launch_slice_cols(const void* src, void* dst, long rows, int Kp,
                       int Kin, hipStream_t s) {
    hipLaunchKernelGGL(k_slice_cols, dim3(cdivl(rows * Kin, NT256)),
                       dim3(NT256), 0, s, (const ushort_t*)src,
                       (ushort_t*)dst, rows, Kp, Kin);
}

// --------------------------------------------------------------------- //
// deterministic column sum (P, M, C) -> (P, C)

int colsum_rb(long M) {
    long rb = cdivl(M, 64);
    return (int)(rb < 256 ? rb : 256);
}

// stage 1: each block owns (problem p, row-chunk rc, column tile) and
// sums its chunk's rows into partials[(p*RB + rc)*C + c]. 8-wide 16B
// loads when C % 8 == 0 (every model shape); scalar fallback otherwise.
__global__ __launch_bounds__(NT256) void k_colsum_part(
        const ushort_t* __restrict__ in, float* __restrict__ partials,
        long M, long C, int RB) {
    int rc = blockIdx.y;
    int p = blockIdx.z;
    long chunk = cdivl(M, RB);
    long r0 = rc * chunk, r1 = min((long)M, r0 + chunk);
    const ushort_t* base = in + (long)p * M * C;
    float* pout = partials + ((long)p * RB + rc) * C;
    if (C % 8 == 0) {
        long c8 = ((long)blockIdx.x * NT256 + threadIdx.x) * 8;
        if (c8 >= C) return;
        float acc[8] = {};
        for (long r = r0; r < r1; r++) {
            union { uint4v v; ushort_t u[8]; } t;
            t.v = *(const uint4v*)(base + r * C + c8);
#pragma unroll
            for (int e = 0; e < 8; e++) acc[e] += bf2f(t.u[e]);
        }
#pragma unroll
        for (int e = 0; e < 8; e++) pout[c8 + e] = acc[e];
        return;
    }
    long c = (long)blockIdx.x * NT256 + threadIdx.x;
    if (c >= C) return;
    float acc = 0.f;
    for (long r = r0; r < r1; r++) acc += bf2f(base[r * C + c]);
    pout[c] = acc;
}

// stage 2: out[p][c] = bf16(sum_rc partials[p][rc][c])
__global__ __launch_bounds__(NT256) void k_colsum_fin(
        const float* __restrict__ partials, ushort_t* __restrict__ out,
        long C, int RB) {
    int c = blockIdx.x * NT256 + threadIdx.x;
    int p = blockIdx.y;
    if (c >= C) return;
    const float* base = partials + (long)p * RB * C + c;
    float acc = 0.f;
    for (int rc = 0; rc < RB; rc++) acc += base[(long)rc * C];
    out[(long)p * C + c] = f2bf(acc);
}

void launch_colsum(const void* in, float* partials, void* out, int nprob,
                   long M, long C, hipStream_t s) {
    int RB = colsum_rb(M);
    long ctiles = (C % 8 == 0) ? cdivl(C / 8, NT256) : cdivl(C, NT256);
    hipLaunchKernelGGL(k_colsum_part,
                       dim3(ctiles, RB, nprob), dim3(NT256), 0, s,
                       (const ushort_t*)in, partials, M, C, RB);
    hipLaunchKernelGGL(k_colsum_fin, dim3(cdivl(C, NT256), nprob),
                       dim3(NT256), 0, s, partials, (ushort_t*)out, C, RB);
}

// --------------------------------------------------------------------- //
// positional-embedding gradient: dPos[n, q] = sum_b sum_{l=1..L-1}
// dLevels[b, n, l, q]  (deterministic: fixed b-then-l order per thread)

// two-stage when B is large: stage A sums a batch chunk into f32
// partials (grid.z = BCH chunks -> fills the chip), stage B folds the
// chunks and writes bf16. d % 8 == 0 on every native config.
__global__ __launch_bounds__(NT256) void k_dpos_part(
        const ushort_t* __restrict__ dlev, float* __restrict__ partials,
        int B, int N, int L, int d, int BCH) {
    long q8 = ((long)blockIdx.x * NT256 + threadIdx.x) * 8;
    int n = blockIdx.y;
    int bc = blockIdx.z;
    if (q8 >= d) return;
    int bchunk = (B + BCH - 1) / BCH;
    int b0 = bc * bchunk, b1 = min(B, b0 + bchunk);
    float acc[8] = {};
    const long bstride = (long)N * L * d;
    const ushort_t* base = dlev + ((long)n * L + 1) * d + q8;
    for (int b = b0; b < b1; b++) {
        const ushort_t* p = base + b * bstride;
        for (int l = 0; l < L - 1; l++) {
            union { uint4v v; ushort_t u[8]; } t;
            t.v = *(const uint4v*)(p + (long)l * d);
#pragma unroll
            for (int e = 0; e < 8; e++) acc[e] += bf2f(t.u[e]);
        }
    }
    float* pout = partials + ((long)bc * N + n) * d + q8;
#pragma unroll
    for (int e = 0; e < 8; e++) pout[e] = acc[e];
}

__global__ __launch_bounds__(NT256) void k_dpos_fin(
        const float* __restrict__ partials, ushort_t* __restrict__ out,
        long Nd, int BCH) {
    long i = (long)blockIdx.x * NT256 + threadIdx.x;
    if (i >= Nd) return;
    float acc = 0.f;
    for (int bc = 0; bc < BCH; bc++) acc += partials[(long)bc * Nd + i];
    out[i] = f2bf(acc);
}

// scalar fallback for d % 8 != 0 (not reachable from the native path)
__global__ __launch_bounds__(NT256) void k_dpos(
        const ushort_t* __restrict__ dlev, ushort_t* __restrict__ out,
        int B, int N, int L, int d) {
    int q = blockIdx.x * NT256 + threadIdx.x;
    int n = blockIdx.y;
    if (q >= d) return;
    float acc = 0.f;
    const long bstride = (long)N * L * d;
    const ushort_t* base = dlev + ((long)n * L + 1) * d + q;
    for (int b = 0; b < B; b++) {
        const ushort_t* p = base + b * bstride;
        for (int l = 0; l < L - 1; l++) acc += bf2f(p[(long)l * d]);
    }
    out[(long)n * d + q] = f2bf(acc);
}

void launch_dpos(const void* dlev, void* out, float* partials, int BCH,
                 int B, int N, int L, int d, hipStream_t s) {
    if (d % 8 == 0 && partials != nullptr && BCH > 0) {
        hipLaunchKernelGGL(k_dpos_part,
                           dim3(cdivl(d / 8, NT256), N, BCH), dim3(NT256),
                           0, s, (const ushort_t*)dlev, partials, B, N, L,
                           d, BCH);
        hipLaunchKernelGGL(k_dpos_fin,
                           dim3(cdivl((long)N * d, NT256)), dim3(NT256), 0,
                           s, partials, (ushort_t*)out, (long)N * d, BCH);
        return;
    }
    hipLaunchKernelGGL(k_dpos, dim3(cdivl(d, NT256), N), dim3(NT256), 0, s,
                       (const ushort_t*)dlev, (ushort_t*)out, B, N, L, d);
}

// --------------------------------------------------------------------- //
// zero one level slice of (B,N,L,d): out[bn, l0, :] = 0  (d % 8 == 0)

__global__ __launch_bounds__(NT256) void k_zero_slice(
        ushort_t* __restrict__ out, long BN, int L, int d, int l0) {
    long i8 = ((long)blockIdx.x * NT256 + threadIdx.x) * 8;
    if (i8 >= BN * d) return;
    int q = i8 % d;
    long bn = i8 / d;
    uint4v z = {0, 0, 0, 0};
    *(uint4v*)(out + (bn * L + l0) * (long)d + q) = z;
}

void launch_zero_slice(void* out, long BN, int L, int d, int l0,
                       hipStream_t s) {
    hipLaunchKernelGGL(k_zero_slice, dim3(cdivl(BN * d / 8, NT256)),
                       dim3(NT256), 0, s, (ushort_t*)out, BN, L, d, l0);
}

// --------------------------------------------------------------------- //
// fused AdamW

// stage 1: deterministic per-block sum of squared gradients. Fixed grid of
// OPT_NPART blocks; block b's partial covers a fixed (tensor-order,
// grid-stride) subset, so the final sum order is launch-invariant.
__global__ __launch_bounds__(NT256) void k_grad_norm_part(
        OptTable t, float* __restrict__ partials) {
    long gid = (long)blockIdx.x * NT256 + threadIdx.x;
    const long gstride = (long)OPT_NPART * NT256;
    float acc = 0.f;
    for (int ti = 0; ti < t.nt; ti++) {
        const ushort_t* g = t.g[ti];
        long n = t.cum[ti + 1] - t.cum[ti];
        for (long i = gid; i < n; i += gstride) {
            float v = bf2f(g[i]);
            acc = fmaf(v, v, acc);
        }
    }
    __shared__ float red[NT256 / WAVE];
    acc = wave_reduce_sum(acc);
    if (threadIdx.x % WAVE == 0) red[threadIdx.x / WAVE] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int w = 0; w < NT256 / WAVE; w++) s += red[w];
        partials[blockIdx.x] = s;
    }
}

// stage 2 (single block): norm[0] = sqrt(sum partials); also advances the
// device step counter (graph-replay-safe: the count lives on device).
__global__ __launch_bounds__(NT256) void k_grad_norm_fin(
        const float* __restrict__ partials, float* __restrict__ norm,
        float* __restrict__ step_dev) {
    __shared__ float red[NT256 / WAVE];
    float acc = 0.f;
    for (int i = threadIdx.x; i < OPT_NPART; i += NT256) acc += partials[i];
    acc = wave_reduce_sum(acc);
    if (threadIdx.x % WAVE == 0) red[threadIdx.x / WAVE] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
        float s = 0.f;
        for (int w = 0; w < NT256 / WAVE; w++) s += red[w];
        norm[0] = sqrtf(s);
        step_dev[0] += 1.0f;
    }
}

// torch.optim.AdamW, one fused pass: clip-scaled bf16 grad -> fp32 m/v ->
// decoupled weight decay -> bias-corrected update of the fp32 master ->
// bf16 param write-back.
__global__ __launch_bounds__(NT256) void k_adamw(
        OptTable t, float lr, float b1, float b2, float eps, float wd,
        float max_norm, const float* __restrict__ norm,
        const float* __restrict__ step_dev) {
    const float tstep = step_dev[0];
    float scale = 1.0f;
    if (max_norm > 0.f) {
        float c = max_norm / (norm[0] + 1e-6f);
        scale = fminf(1.0f, c);
    }
    const float bc1 = 1.0f - __powf(b1, tstep);
    const float bc2 = 1.0f - __powf(b2, tstep);
    const float step_size = lr / bc1;
    const float inv_bc2 = 1.0f / bc2;
    const float decay = 1.0f - lr * wd;

    long gid = (long)blockIdx.x * NT256 + threadIdx.x;
    const long gstride = (long)gridDim.x * NT256;
    for (int ti = 0; ti < t.nt; ti++) {
        const ushort_t* gp = t.g[ti];
        float* mw = t.mw[ti];
        float* m1 = t.m1[ti];
        float* m2 = t.m2[ti];
        ushort_t* pw = t.pw[ti];
        long n = t.cum[ti + 1] - t.cum[ti];
        for (long i = gid; i < n; i += gstride) {
            float g = bf2f(gp[i]) * scale;
            float m = fmaf(b1, m1[i], (1.0f - b1) * g);
            float v = fmaf(b2, m2[i], (1.0f - b2) * g * g);
            m1[i] = m;
            m2[i] = v;
            float p = mw[i] * decay;
            p -= step_size * m / (sqrtf(v * inv_bc2) + eps);
            mw[i] = p;
            pw[i] = f2bf(p);
        }
    }
}

void launch_grad_norm(const OptTable& t, float* partials, float* norm,
                      float* step_dev, hipStream_t s) {
    hipLaunchKernelGGL(k_grad_norm_part, dim3(OPT_NPART), dim3(NT256), 0, s,
                       t, partials);
    hipLaunchKernelGGL(k_grad_norm_fin, dim3(1), dim3(NT256), 0, s,
                       partials, norm, step_dev);
}

void launch_adamw(const OptTable& t, float lr, float b1, float b2, float eps,
                  float wd, float max_norm, const float* norm,
                  const float* step_dev, hipStream_t s) {
    // ~2048 blocks: plenty to fill 256 CUs, few enough that per-tensor
    // grid-stride loops stay short
    hipLaunchKernelGGL(k_adamw, dim3(2048), dim3(NT256), 0, s, t, lr, b1,
                       b2, eps, wd, max_norm, norm, step_dev);
}
