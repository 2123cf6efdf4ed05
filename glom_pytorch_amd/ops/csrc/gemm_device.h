// Shared device-side helpers for the GLOM GEMM kernels.
#pragma once
#include "common.h"
#include "gemm.h"

__device__ __forceinline__ void resolve_ptr2(const OpArg& a,
                                             const void* const* tab,
                                             const long* tabld, int pid,
                                             int nInner, const ushort_t** ptr,
                                             long* ld) {
    if (a.flags & OP_TABLE) {
        *ptr = (const ushort_t*)tab[pid];
        *ld = tabld[pid];
    } else {
        *ptr = (const ushort_t*)a.base + (long)(pid % nInner) * a.sin
               + (long)(pid / nInner) * a.sout;
        *ld = a.ld;
    }
}

// Shared epilogue: alpha, colscale, gelu-grad, bias, optional gelu-pair.
__device__ __forceinline__ void gemm_epilogue(
        const GemmParams& p, int pid, int m0, int n0, int wm, int wn,
        int lrow, int kq, const f32x4 acc[4][4]) {
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
    const ushort_t* biasp = nullptr;
    if (p.has_bias)
        biasp = (const ushort_t*)p.bias_base
                + (long)(pid % p.nInner) * p.bias_sin
                + (long)(pid / p.nInner) * p.bias_sout;
    const float* csp = nullptr;
    if (p.has_colscale)
        csp = (const float*)p.colscale_base
              + (long)(pid % p.nInner) * p.cs_sin
              + (long)(pid / p.nInner) * p.cs_sout;
    const ushort_t* auxp = nullptr;
    if (p.epilogue == EPI_GELUGRAD)
        auxp = (const ushort_t*)p.aux_base
               + (long)(pid % p.nInner) * p.aux_sin
               + (long)(pid / p.nInner) * p.aux_sout;
    ushort_t* out2p = nullptr;
    if (p.epilogue == EPI_GELU_PAIR)
        out2p = (ushort_t*)p.out2 + (long)(pid % p.nInner) * p.out2_sin
                + (long)(pid / p.nInner) * p.out2_sout;

#pragma unroll
    for (int i16 = 0; i16 < 4; i16++) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
            int i = m0 + wm + i16 * 16 + kq * 4 + r;
            if (i >= p.M) continue;
            ushort_t* crow = Cp + (long)i * ldc;
            ushort_t* o2row =
                out2p ? out2p + (long)i * p.out2_ld : nullptr;
            const ushort_t* auxrow =
                auxp ? auxp + (long)i * p.aux_ld : nullptr;
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                int j = n0 + wn + j16 * 16 + lrow;
                if (j >= p.N) continue;
                float v = acc[i16][j16][r] * p.alpha;
                if (csp) v *= csp[j];
                if (auxrow) v *= gelu_grad_f(bf2f(auxrow[j]));
                if (biasp) v += bf2f(biasp[j]);
                crow[j] = f2bf(v);
                if (o2row) o2row[j] = f2bf(gelu_f(v));
            }
        }
    }
}

// Full-tile epilogue with LDS-staged stores: the MFMA D-fragment layout
// (col = lane&15, rows scattered) makes direct global stores 2-byte
// partial-line writes (read-modify-write amplified in L2). Instead the
// 128x128 output tile is packed to bf16 in LDS (padded rows), then
// streamed out as full 128B lines. EPI_GELU_PAIR computes gelu() during
// the linear pass (on the rounded bf16 pre-activation, exactly what the
// eager reference's gelu(conv_out_bf16) computes).
// Requires: full tile (no predicates), scratch >= 128*136*2 bytes.
#define EPI_LDS_ROW 136   // 128 cols + 8 pad elements (16B-aligned rows)

__device__ __forceinline__ void gemm_epilogue_lds(
        const GemmParams& p, int pid, int m0, int n0, int wm, int wn,
        int lrow, int kq, const f32x4 acc[4][4], ushort_t* scratch) {
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
    const ushort_t* biasp = nullptr;
    if (p.has_bias)
        biasp = (const ushort_t*)p.bias_base
                + (long)(pid % p.nInner) * p.bias_sin
                + (long)(pid / p.nInner) * p.bias_sout;
    const float* csp = nullptr;
    if (p.has_colscale)
        csp = (const float*)p.colscale_base
              + (long)(pid % p.nInner) * p.cs_sin
              + (long)(pid / p.nInner) * p.cs_sout;
    const ushort_t* auxp = nullptr;
    if (p.epilogue == EPI_GELUGRAD)
        auxp = (const ushort_t*)p.aux_base
               + (long)(pid % p.nInner) * p.aux_sin
               + (long)(pid / p.nInner) * p.aux_sout;
    ushort_t* out2p = nullptr;
    if (p.epilogue == EPI_GELU_PAIR || p.epilogue == 10)
        out2p = (ushort_t*)p.out2 + (long)(pid % p.nInner) * p.out2_sin
                + (long)(pid / p.nInner) * p.out2_sout;

    // phase 1: acc -> bf16 into LDS [128][EPI_LDS_ROW], local coords.
    // Column factors are pre-loaded under ONE branch: a per-element
    // `ptr ? load : skip` select makes hipcc branch around every load and
    // wait vmcnt(0) per element (serialized L2 round trips).
    float csv[4] = {1.f, 1.f, 1.f, 1.f};
    float bvv[4] = {0.f, 0.f, 0.f, 0.f};
    if (csp) {
#pragma unroll
        for (int j16 = 0; j16 < 4; j16++)
            csv[j16] = csp[n0 + wn + j16 * 16 + lrow];
    }
    if (biasp) {
#pragma unroll
        for (int j16 = 0; j16 < 4; j16++)
            bvv[j16] = bf2f(biasp[n0 + wn + j16 * 16 + lrow]);
    }
#pragma unroll
    for (int i16 = 0; i16 < 4; i16++) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
            int li = wm + i16 * 16 + kq * 4 + r;           // 0..127
            long gi = m0 + li;
            float vv[4];
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++)
                vv[j16] = acc[i16][j16][r] * p.alpha * csv[j16];
            if (auxp) {
                const ushort_t* auxrow = auxp + gi * p.aux_ld;
                ushort_t av[4];
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++)
                    av[j16] = auxrow[n0 + wn + j16 * 16 + lrow];
                float gx[4], gy[4];
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++) gx[j16] = bf2f(av[j16]);
                gelu_grad_vec<4>(gx, gy);
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++) vv[j16] *= gy[j16];
            }
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                int lj = wn + j16 * 16 + lrow;             // 0..127
                scratch[li * EPI_LDS_ROW + lj] = f2bf(vv[j16] + bvv[j16]);
            }
        }
    }
    __syncthreads();
    // phase 2: linear sweep, 2 threads per row, full 16B chunks
    {
        int t = threadIdx.x;           // 256 threads, 128 rows x 2 halves
        int li = t >> 1;
        int half = (t & 1) * 64;
        long gi = m0 + li;
        ushort_t* crow = Cp + gi * ldc + n0 + half;
        ushort_t* orow =
            out2p ? out2p + gi * p.out2_ld + n0 + half : nullptr;
        const ushort_t* srow = scratch + li * EPI_LDS_ROW + half;
#pragma unroll
        for (int c = 0; c < 8; c++)
            *(uint4v*)(crow + c * 8) = *(const uint4v*)(srow + c * 8);
        if (orow) {
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x, g;
                x.v = *(const uint4v*)(srow + c * 8);
                if (p.epilogue == 10) {       // debug: pair without gelu
                    g.v = x.v;
                } else {
                    float xin[8], yv[8];
#pragma unroll
                    for (int e = 0; e < 8; e++) xin[e] = bf2f(x.u[e]);
                    gelu_f_vec<8>(xin, yv);
#pragma unroll
                    for (int e = 0; e < 8; e++) g.u[e] = f2bf(yv[e]);
                }
                *(uint4v*)(orow + c * 8) = g.v;
            }
        }
    }
}
