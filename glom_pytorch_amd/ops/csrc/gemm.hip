// Generic grouped / batched bf16 MFMA GEMM for the GLOM hot path (gfx950).
//
// Computes, for every problem p in a batch:
//     C_p[i,j] = alpha * sum_k Aop_p[i,k] * Bop_p[j,k]   (+ bias_p[j])
// with layout selecting how Aop/Bop map onto the row-major source tensors:
//     NT: Aop[i,k] = A[i,k]        Bop[j,k] = B[j,k]   (B = packed weights)
//     NN: Aop[i,k] = A[i,k]        Bop[j,k] = B[k,j]
//     TN: Aop[i,k] = A[k,i]        Bop[j,k] = B[k,j]   (k = token index)
//
// One kernel serves all GLOM GEMMs (SURVEY.md §2.3 K1-K7 and their
// backwards): per-problem addressing is either strided (two-level:
// p%nInner, p/nInner) or an explicit <=16 entry pointer table (used for the
// bottom-up groups where group 0 reads the patch tokens and groups 1..L-1
// read level slices — avoids the reference's per-iteration torch.cat,
// glom_pytorch.py:132). Optional source transforms: exact-erf GELU on load
// (down-projection consumes gelu(H_pre) without materializing it) and
// positional-embedding add on load (top-down input = levels + pos,
// glom_pytorch.py:136). Optional epilogues: column scale (consensus k-row
// L2 norms), gelu'(aux) multiply (GELU backward), bias add.
//
// v1 structure: 128x128x32 tiles, 4 waves x (64x64), mfma_f32_16x16x32_bf16,
// register-staged LDS with +8 element row padding. Correctness-first; the
// tuned glds/8-phase structure comes later.

#include "common.h"
#include "gemm.h"
#include "gemm_device.h"

#define BM 128
#define BN 128
#define BK 32
#define BKP (BK + 8)
#define NTHREADS 256

// Stage a [R x BK] tile of a row-major (rows x cols, leading dim ld) source
// into LDS dst[R][BKP]. Rows r0.., cols k0..; out-of-range -> 0.
// Fast path: 16-byte vector load + 16-byte LDS store when the row is fully
// in range and 16B-aligned (ld % 8 == 0); transforms (pos-add / GELU) are
// applied in registers between load and store.
template <int R>
__device__ __forceinline__ void stage_normal(
        ushort_t* dst, const ushort_t* src, long ld, int r0, int k0,
        int maxR, int maxK, int flags, const ushort_t* pos, long pos_ld,
        int npatch) {
    constexpr int CH = R * (BK / 8);
    const bool xform = (flags & (OP_GELU | OP_POS)) != 0;
    const bool aligned = (ld % 8 == 0) && (((uintptr_t)src & 15) == 0);
#pragma unroll
    for (int c0 = 0; c0 < CH; c0 += NTHREADS) {
        int c = c0 + threadIdx.x;
        if (CH % NTHREADS != 0 && c >= CH) break;
        int row = c / (BK / 8);
        int kp = c % (BK / 8);
        int gr = r0 + row, gk = k0 + kp * 8;
        union { uint4v v; ushort_t u[8]; } t;
        bool full = gr < maxR && gk + 7 < maxK;
        if (full && aligned) {
            t.v = *(const uint4v*)(src + (long)gr * ld + gk);
        } else if (gr < maxR) {
#pragma unroll
            for (int e = 0; e < 8; e++)
                t.u[e] = (gk + e < maxK) ? src[(long)gr * ld + gk + e]
                                         : (ushort_t)0;
        } else {
            t.v = 0;
        }
        if (xform && gr < maxR) {
            const ushort_t* prow =
                (flags & OP_POS) ? pos + (long)(gr % npatch) * pos_ld + gk
                                 : nullptr;
#pragma unroll
            for (int e = 0; e < 8; e++) {
                if (gk + e < maxK) {
                    float v = bf2f(t.u[e]);
                    if (flags & OP_POS) v += bf2f(prow[e]);
                    if (flags & OP_GELU) v = gelu_f(v);
                    t.u[e] = f2bf(v);
                }
            }
        }
        *(uint4v*)(dst + row * BKP + kp * 8) = t.v;
    }
}

// Stage a transposed tile: source is row-major with the REDUCTION index as
// the row (rows t0..t0+BK-1, cols c0..c0+R-1); LDS image is dst[R][BKP]
// with dst[col][trow].
template <int R>
__device__ __forceinline__ void stage_transposed(
        ushort_t* dst, const ushort_t* src, long ld, int t0, int c0,
        int maxT, int maxC, int flags, const ushort_t* pos, long pos_ld,
        int npatch) {
    constexpr int CH = BK * (R / 8);
    const bool xform = (flags & (OP_GELU | OP_POS)) != 0;
    const bool aligned = (ld % 8 == 0) && (((uintptr_t)src & 15) == 0);
#pragma unroll
    for (int cc0 = 0; cc0 < CH; cc0 += NTHREADS) {
        int c = cc0 + threadIdx.x;
        if (CH % NTHREADS != 0 && c >= CH) break;
        int trow = c / (R / 8);
        int cp = c % (R / 8);
        int gt = t0 + trow, gc = c0 + cp * 8;
        union { uint4v v; ushort_t u[8]; } tt;
        ushort_t* u = tt.u;
        if (gt < maxT && gc + 7 < maxC && aligned) {
            tt.v = *(const uint4v*)(src + (long)gt * ld + gc);
        } else if (gt < maxT) {
#pragma unroll
            for (int e = 0; e < 8; e++)
                u[e] = (gc + e < maxC) ? src[(long)gt * ld + gc + e]
                                       : (ushort_t)0;
        } else {
            tt.v = 0;
        }
        if (xform && gt < maxT) {
            const ushort_t* prow =
                (flags & OP_POS) ? pos + (long)(gt % npatch) * pos_ld + gc
                                 : nullptr;
#pragma unroll
            for (int e = 0; e < 8; e++) {
                if (gc + e < maxC) {
                    float v = bf2f(u[e]);
                    if (flags & OP_POS) v += bf2f(prow[e]);
                    if (flags & OP_GELU) v = gelu_f(v);
                    u[e] = f2bf(v);
                }
            }
        }
#pragma unroll
        for (int e = 0; e < 8; e++) dst[(cp * 8 + e) * BKP + trow] = u[e];
    }
}

__global__ __launch_bounds__(NTHREADS) void gemm_kernel(GemmParams p) {
    __shared__ ushort_t As[BM * BKP];
    __shared__ ushort_t Bs[BN * BKP];

    int pid = blockIdx.z;
    int slice = 0, k_begin = 0, k_end = p.K;
    if (p.splitk > 1) {
        pid = blockIdx.z % p.nproblems;
        slice = blockIdx.z / p.nproblems;
        int per = ((p.K + BK - 1) / BK + p.splitk - 1) / p.splitk * BK;
        k_begin = slice * per;
        k_end = min(p.K, k_begin + per);
    }
    const int m0 = blockIdx.y * BM;
    const int n0 = blockIdx.x * BN;

    const ushort_t* Ap;
    const ushort_t* Bp;
    long lda, ldb;
    resolve_ptr2(p.A, p.Atab, p.Atabld, pid, p.nInner, &Ap, &lda);
    resolve_ptr2(p.B, p.Btab, p.Btabld, pid, p.nInner, &Bp, &ldb);

    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int wm = (wid >> 1) * 64;
    const int wn = (wid & 1) * 64;
    const int lrow = lane & 15;
    const int kq = lane >> 4;

    f32x4 acc[4][4] = {};

    const ushort_t* pos = (const ushort_t*)p.pos;

    for (int k0 = k_begin; k0 < k_end; k0 += BK) {
        if (p.layout == LAYOUT_TN)
            stage_transposed<BM>(As, Ap, lda, k0, m0, p.K, p.M, p.A.flags,
                                 pos, p.pos_ld, p.npatch);
        else
            stage_normal<BM>(As, Ap, lda, m0, k0, p.M, p.K, p.A.flags, pos,
                             p.pos_ld, p.npatch);
        if (p.layout == LAYOUT_NT)
            stage_normal<BN>(Bs, Bp, ldb, n0, k0, p.N, p.K, p.B.flags, pos,
                             p.pos_ld, p.npatch);
        else
            stage_transposed<BN>(Bs, Bp, ldb, k0, n0, p.K, p.N, p.B.flags,
                                 pos, p.pos_ld, p.npatch);
        __syncthreads();

        short8 af[4], bfr[4];
#pragma unroll
        for (int i = 0; i < 4; i++)
            af[i] = *(const short8*)&As[(wm + i * 16 + lrow) * BKP + kq * 8];
#pragma unroll
        for (int j = 0; j < 4; j++)
            bfr[j] = *(const short8*)&Bs[(wn + j * 16 + lrow) * BKP + kq * 8];
#pragma unroll
        for (int i = 0; i < 4; i++)
#pragma unroll
            for (int j = 0; j < 4; j++)
                acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af[i], bfr[j], acc[i][j], 0, 0, 0);
        __syncthreads();
    }

    // ---- split-K epilogue: raw f32 partials into this slice's slab ----
    if (p.splitk > 1) {
        float* ws = p.ws + ((long)slice * p.nproblems + pid) * p.M * p.N;
#pragma unroll
        for (int i16 = 0; i16 < 4; i16++) {
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                int j = n0 + wn + j16 * 16 + lrow;
                if (j >= p.N) continue;
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    int i = m0 + wm + i16 * 16 + kq * 4 + r;
                    if (i >= p.M) continue;
                    ws[(long)i * p.N + j] = acc[i16][j16][r];
                }
            }
        }
        return;
    }

    gemm_epilogue(p, pid, m0, n0, wm, wn, lrow, kq, acc);
}

// Sum the split-K slices and write bf16 C (alpha applied once here).
__global__ __launch_bounds__(NTHREADS) void gemm_finish_kernel(GemmParams p) {
    long mn = (long)p.M * p.N;
    long idx = (long)blockIdx.x * NTHREADS + threadIdx.x;
    if (idx >= (long)p.nproblems * mn) return;
    int pid = idx / mn;
    long rem = idx % mn;
    float s = 0.f;
    for (int sl = 0; sl < p.splitk; sl++)
        s += p.ws[((long)sl * p.nproblems + pid) * mn + rem];
    ushort_t* Cp;
    long ldc;
    {
        const ushort_t* tmp;
        OpArg ca;
        ca.base = p.Cbase; ca.sin = p.Csin; ca.sout = p.Csout; ca.ld = p.Cld;
        ca.flags = p.Cflags;
        resolve_ptr2(ca, (const void* const*)p.Ctab, p.Ctabld, pid, p.nInner,
                     &tmp, &ldc);
        Cp = (ushort_t*)tmp;
    }
    long i = rem / p.N, j = rem % p.N;
    Cp[i * ldc + j] = f2bf(s * p.alpha);
}

void launch_gemm(const GemmParams& p, hipStream_t stream) {
    int sk = p.splitk > 1 ? p.splitk : 1;
    dim3 grid((p.N + BN - 1) / BN, (p.M + BM - 1) / BM, p.nproblems * sk);
    hipLaunchKernelGGL(gemm_kernel, grid, dim3(NTHREADS), 0, stream, p);
}

void launch_gemm_finish(const GemmParams& p, hipStream_t stream) {
    long total = (long)p.nproblems * p.M * p.N;
    hipLaunchKernelGGL(gemm_finish_kernel,
                       dim3((total + NTHREADS - 1) / NTHREADS),
                       dim3(NTHREADS), 0, stream, p);
}
