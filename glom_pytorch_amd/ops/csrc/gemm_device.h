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
