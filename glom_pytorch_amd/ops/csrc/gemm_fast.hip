// Tuned full-tile GEMM kernels for the GLOM hot path (gfx950).
//
// gemm_nt_fast: 128x128x64 tiles, 4 waves x (64x64), double-buffered LDS
// filled by direct global->LDS DMA (`global_load_lds` width 16) with the
// T2 XOR swizzle applied on the per-lane SOURCE address and again on the
// ds_read_b128 fragment reads (linear lane-order LDS destination, as
// rule 21 requires). One barrier per K-tile; the barrier's implicit
// vmcnt(0) drains the in-flight DMA of the next tile.
//
// gemm_tn_fast: both operands have the reduction (token) index as the
// SOURCE row, so tiles are transpose-staged: each thread loads an 8x8
// block with 16B vector loads, repacks to column vectors in registers,
// and writes 8 x ds_write_b128 into the swizzled [out-dim][k] image —
// replacing the v1 per-element LDS scatter (8x fewer, 8x wider writes).
//
// Preconditions (host-checked, fast_ok): M%128==0, N%128==0, K%64==0
// (TN: K%8==0 with row predicates), all row strides %8==0, no on-load
// transforms. Everything else falls back to the generic gemm_kernel.

#include <cstdlib>

#include "common.h"
#include "gemm.h"
#include "gemm_device.h"

#define BM 128
#define BN 128
#define FBK 64
#define NTHREADS 256

// LDS bank swizzle: 16B-granule index in a 128B row is XORed with
// swz_row(row); includes row bit 3 so writes/reads 8 rows apart do not
// collide on the same 16B slot of the 256B bank row.
__device__ __forceinline__ int swz_row(int row) {
    return (row ^ (row >> 3)) & 7;
}

// Issue this wave's share of one operand tile (128 rows x 64 cols bf16 =
// 16 KB = 16 DMA chunks of 1 KB; 4 chunks per wave) into a LINEAR LDS
// image, pre-swizzling the source address so a swizzled ds_read_b128 is
// conflict-free (byte_in_row ^= (row&7)<<4).
__device__ __forceinline__ void stage_glds(
        ushort_t* lds, const ushort_t* src, long ld, int r0, int k0,
        int wid, int lane) {
#pragma unroll
    for (int c = 0; c < 4; c++) {
        int chunk = wid * 4 + c;
        int row = chunk * 8 + (lane >> 3);
        int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
        const ushort_t* gaddr = src + (long)(r0 + row) * ld + k0 + swz8;
        ushort_t* laddr = lds + chunk * 512;  // wave-uniform chunk base
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gaddr,
            (__attribute__((address_space(3))) unsigned int*)laddr, 16, 0, 0);
    }
}

__global__ __launch_bounds__(NTHREADS) void gemm_nt_fast_kernel(GemmParams p) {
    // ONE shared array: a second __shared__ object makes hipcc emit
    // s_waitcnt vmcnt(0) before the first ds_read of every k-step of a
    // glds pipeline, draining the prefetch (guide §5 ".s-level traps" (a)).
    __shared__ ushort_t smem[4 * BM * FBK];
    ushort_t* As0 = smem;
    ushort_t* Bs0 = smem + 2 * BM * FBK;

    const int pid = blockIdx.z;
    // XCD-aware block remap (T1), column-major: each XCD die owns a
    // contiguous run of N-columns (all M-tiles of a few n-tiles), so the
    // shared B panel of a column stays resident in that XCD's private L2
    // while the M sweep streams A.
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN;   // column-major: n outer
    const int m0 = (bid % gridDim.y) * BM;

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

    const int nk = p.K / FBK;
    int cur = 0;
    stage_glds(As0, Ap, lda, m0, 0, wid, lane);
    stage_glds(Bs0, Bp, ldb, n0, 0, wid, lane);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    for (int kt = 0; kt < nk; kt++) {
        if (kt + 1 < nk) {
            stage_glds(As0 + (cur ^ 1) * BM * FBK, Ap, lda, m0,
                       (kt + 1) * FBK, wid, lane);
            stage_glds(Bs0 + (cur ^ 1) * BN * FBK, Bp, ldb, n0,
                       (kt + 1) * FBK, wid, lane);
        }
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&As0[cur * BM * FBK + row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bs0[cur * BN * FBK + row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        __syncthreads();   // drains the in-flight DMA (vmcnt0) + buffer reuse
        cur ^= 1;
    }
    gemm_epilogue_lds(p, pid, m0, n0, wm, wn, lrow, kq, acc, smem);
}

// Transpose-stage one operand tile: source rows = reduction slice
// [t0, t0+64) (row stride ld), cols = output-dim slice [c0, c0+128).
// 128 threads per operand; thread (mb, cb) loads an 8(m) x 8(c) block and
// writes 8 column-vectors of 8 bf16 via ds_write_b128 into the swizzled
// [col][m] image (row length 64 elements = 128 B).
__device__ __forceinline__ void stage_repack(
        ushort_t* lds, const ushort_t* src, long ld, int t0, int c0,
        int kend, int tid128) {
    int cb = tid128 & 15;   // 16 col-blocks of 8
    int mb = tid128 >> 4;   // 8 m-blocks of 8
    union { uint4v v; ushort_t u[8]; } rowv[8];
#pragma unroll
    for (int r = 0; r < 8; r++) {
        int m = t0 + mb * 8 + r;
        if (m < kend)
            rowv[r].v = *(const uint4v*)(src + (long)m * ld + c0 + cb * 8);
        else
            rowv[r].v = 0;
    }
#pragma unroll
    for (int e = 0; e < 8; e++) {
        union { uint4v v; ushort_t u[8]; } col;
#pragma unroll
        for (int r = 0; r < 8; r++) col.u[r] = rowv[r].u[e];
        int row = cb * 8 + e;                       // output-dim index
        int off = (mb * 8) ^ (swz_row(row) << 3);   // swizzled m offset
        *(uint4v*)&lds[row * FBK + off] = col.v;
    }
}

__global__ __launch_bounds__(NTHREADS) void gemm_tn_fast_kernel(GemmParams p) {
    __shared__ ushort_t smem[128 * EPI_LDS_ROW];   // >= As+Bs (2*8192)
    ushort_t* As = smem;
    ushort_t* Bs = smem + BM * FBK;

    int pid = blockIdx.z;
    int slice = 0, k_begin = 0, k_end = p.K;
    if (p.splitk > 1) {
        pid = blockIdx.z % p.nproblems;
        slice = blockIdx.z / p.nproblems;
        int per = ((p.K + FBK - 1) / FBK + p.splitk - 1) / p.splitk * FBK;
        k_begin = slice * per;
        k_end = min(p.K, k_begin + per);
    }
    // XCD-aware block remap (T1), column-major: each XCD die owns a
    // contiguous run of N-columns (all M-tiles of a few n-tiles), so the
    // shared B panel of a column stays resident in that XCD's private L2
    // while the M sweep streams A.
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN;   // column-major: n outer
    const int m0 = (bid % gridDim.y) * BM;

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

    for (int k0 = k_begin; k0 < k_end; k0 += FBK) {
        if (threadIdx.x < 128)
            stage_repack(As, Ap, lda, k0, m0, k_end, threadIdx.x);
        else
            stage_repack(Bs, Bp, ldb, k0, n0, k_end, threadIdx.x - 128);
        __syncthreads();
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&As[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bs[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        __syncthreads();
    }

    if (p.splitk > 1) {
        float* ws = p.ws + ((long)slice * p.nproblems + pid) * p.M * p.N;
#pragma unroll
        for (int i16 = 0; i16 < 4; i16++)
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                int j = n0 + wn + j16 * 16 + lrow;
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    int i = m0 + wm + i16 * 16 + kq * 4 + r;
                    ws[(long)i * p.N + j] = acc[i16][j16][r];
                }
            }
        return;
    }
    __syncthreads();   // all waves done with As/Bs before epilogue staging
    gemm_epilogue_lds(p, pid, m0, n0, wm, wn, lrow, kq, acc, smem);
}

void launch_gemm_nt_fast(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN, p.M / BM, p.nproblems);
    hipLaunchKernelGGL(gemm_nt_fast_kernel, grid, dim3(NTHREADS), 0, stream,
                       p);
}

void launch_gemm_tn_fast(const GemmParams& p, hipStream_t stream) {
    int sk = p.splitk > 1 ? p.splitk : 1;
    dim3 grid(p.N / BN, p.M / BM, p.nproblems * sk);
    hipLaunchKernelGGL(gemm_tn_fast_kernel, grid, dim3(NTHREADS), 0, stream,
                       p);
}

// ---------------------------------------------------------------- //
// NN fast kernel (attention AV / dq): A row-major [M][K] staged as a
// plain swizzled copy; B row-major [K][N] transpose-staged with the
// 8x8 register repack. Single LDS buffer, two barriers per K-step.

__device__ __forceinline__ void stage_copy128(
        ushort_t* lds, const ushort_t* src, long ld, int r0, int k0,
        int maxK, int tid128) {
    // 128 threads stage a 128x64 tile: thread t handles rows t, t+... no:
    // 1024 chunks of 8 elements / 128 threads = 8 chunks each
#pragma unroll
    for (int cc = 0; cc < 8; cc++) {
        int c = tid128 * 8 + cc;
        int row = c >> 3;
        int g = c & 7;
        int gk = k0 + g * 8;
        union { uint4v v; ushort_t u[8]; } t;
        if (gk + 7 < maxK)
            t.v = *(const uint4v*)(src + (long)(r0 + row) * ld + gk);
        else
            t.v = 0;
        int off = (g * 8) ^ (swz_row(row) << 3);
        *(uint4v*)&lds[row * FBK + off] = t.v;
    }
}

__global__ __launch_bounds__(NTHREADS) void gemm_nn_fast_kernel(GemmParams p) {
    __shared__ ushort_t smem[128 * EPI_LDS_ROW];   // >= As+Bs (2*8192)
    ushort_t* As = smem;
    ushort_t* Bs = smem + BM * FBK;

    const int pid = blockIdx.z;
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN;
    const int m0 = (bid % gridDim.y) * BM;

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

    for (int k0 = 0; k0 < p.K; k0 += FBK) {
        if (threadIdx.x < 128)
            stage_copy128(As, Ap, lda, m0, k0, p.K, threadIdx.x);
        else
            stage_repack(Bs, Bp, ldb, k0, n0, p.K, threadIdx.x - 128);
        __syncthreads();
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&As[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bs[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        __syncthreads();
    }
    __syncthreads();
    gemm_epilogue_lds(p, pid, m0, n0, wm, wn, lrow, kq, acc, smem);
}

void launch_gemm_nn_fast(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN, p.M / BM, p.nproblems);
    hipLaunchKernelGGL(gemm_nn_fast_kernel, grid, dim3(NTHREADS), 0, stream,
                       p);
}




// ---------------------------------------------------------------- //
// 128x256-tile NT kernel: 8 waves (512 threads), per-wave 64x64, BK=64
// double-buffered glds, ONE barrier per K-step (the proven 2-phase loop).
// 2x the per-block work of the 128^2 kernel at 96 KiB LDS — occupancy
// stays at 2 waves/SIMD while the per-block prologue/epilogue halves
// relative to the K=512 GLOM shapes that dominate the step.

#define NT3 512
#define BN3 256
#define EPI2_ROW 264   // 256 cols + 8 pad (16B-aligned LDS epilogue rows)

__global__ __launch_bounds__(NT3) void gemm_nt_fast3_kernel(GemmParams p) {
    __shared__ ushort_t smem[2 * (128 + 256) * FBK];   // 96 KiB
    const int abuf = 128 * FBK, bbuf = 256 * FBK, stride = abuf + bbuf;

    const int pid = blockIdx.z;
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN3;
    const int m0 = (bid % gridDim.y) * BM;

    const ushort_t* Ap;
    const ushort_t* Bp;
    long lda, ldb;
    resolve_ptr2(p.A, p.Atab, p.Atabld, pid, p.nInner, &Ap, &lda);
    resolve_ptr2(p.B, p.Btab, p.Btabld, pid, p.nInner, &Bp, &ldb);

    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int wm = (wid >> 2) * 64;
    const int wn = (wid & 3) * 64;
    const int lrow = lane & 15;
    const int kq = lane >> 4;

    f32x4 acc[4][4] = {};

    // A: 16 chunks over 8 waves (2 each); B: 32 chunks (4 each)
    auto stage = [&](int buf, int k0) {
        ushort_t* Al = smem + buf * stride;
        ushort_t* Bl = Al + abuf;
#pragma unroll
        for (int c = 0; c < 2; c++) {
            int chunk = wid * 2 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g = Ap + (long)(m0 + row) * lda + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g,
                (__attribute__((address_space(3))) unsigned int*)
                    (Al + chunk * 512), 16, 0, 0);
        }
#pragma unroll
        for (int c = 0; c < 4; c++) {
            int chunk = wid * 4 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g = Bp + (long)(n0 + row) * ldb + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g,
                (__attribute__((address_space(3))) unsigned int*)
                    (Bl + chunk * 512), 16, 0, 0);
        }
    };

    const int nk = p.K / FBK;
    int cur = 0;
    stage(0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    for (int kt = 0; kt < nk; kt++) {
        if (kt + 1 < nk) stage(cur ^ 1, (kt + 1) * FBK);
        const ushort_t* Al = smem + cur * stride;
        const ushort_t* Bl = Al + abuf;
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&Al[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bl[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        __syncthreads();
        cur ^= 1;
    }

    // ---- epilogue: 128x256 tile via LDS, full-line stores ----
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

    float csv[4] = {1.f, 1.f, 1.f, 1.f};
    float bvv[4] = {0.f, 0.f, 0.f, 0.f};
    if (csp && p.epilogue != EPI_SMBWD) {
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
                int lj = wn + j16 * 16 + lrow;             // 0..255
                smem[li * EPI2_ROW + lj] = f2bf(vv[j16] + bvv[j16]);
            }
        }
    }
    __syncthreads();

    if (p.epilogue == EPI_SOFTMAX || p.epilogue == EPI_SMBWD) {
        // Full-row fusions: this tile holds complete rows (N == 256), so
        // the masked row softmax (fwd) / softmax backward (bwd) runs here
        // instead of a separate kernel + a global-memory round trip.
        // 4 consecutive threads (one quartet, same wave) own one row.
        int t = threadIdx.x;
        int li = t >> 2;
        int qt = (t & 3) * 64;
        long gi = m0 + li;
        const bool* mrow =
            p.nlmask ? (const bool*)p.nlmask + gi * p.N + n0 + qt : nullptr;
        ushort_t* srow = smem + li * EPI2_ROW + qt;
        const float SELF = bf2f(f2bf(-5e-4f));
        const float NEG = -3.3895314e38f;

        if (p.epilogue == EPI_SOFTMAX) {
            // pass 1: masks + quartet max
            float mx = -INFINITY;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    int gj = n0 + qt + c * 8 + e;
                    float v = bf2f(x.u[e]);
                    if (p.self_mask && gj == gi) v = SELF;
                    if (mrow && mrow[c * 8 + e]) v = NEG;
                    x.u[e] = f2bf(v);
                    mx = fmaxf(mx, v);
                }
                *(uint4v*)(srow + c * 8) = x.v;   // masked values back
            }
            mx = fmaxf(mx, __shfl_xor(mx, 1));
            mx = fmaxf(mx, __shfl_xor(mx, 2));
            // pass 2: exp + quartet sum (store unnormalized bf16)
            float sum = 0.f;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    float ex = __expf(bf2f(x.u[e]) - mx);
                    sum += ex;
                    x.u[e] = f2bf(ex);
                }
                *(uint4v*)(srow + c * 8) = x.v;
            }
            sum += __shfl_xor(sum, 1);
            sum += __shfl_xor(sum, 2);
            float inv = 1.0f / sum;
            // pass 3: normalize + store P
            ushort_t* prow = Cp + gi * ldc + n0 + qt;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++)
                    x.u[e] = f2bf(bf2f(x.u[e]) * inv);
                *(uint4v*)(prow + c * 8) = x.v;
            }
        } else {
            // EPI_SMBWD: dS = alpha2 * P * (dP - sum(P*dP)), masked -> 0;
            // dSr = dS * rnorm[j]. P comes via aux, rnorm via colscale
            // pointers (phase 1 ran with alpha=1, no colscale).
            const ushort_t* Prow = (const ushort_t*)p.aux_base
                + (long)(pid % p.nInner) * p.aux_sin
                + (long)(pid / p.nInner) * p.aux_sout + gi * p.aux_ld
                + n0 + qt;
            const float* rn = (const float*)p.colscale_base
                + (long)(pid % p.nInner) * p.cs_sin
                + (long)(pid / p.nInner) * p.cs_sout + n0 + qt;
            float tsum = 0.f;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } dp, pp;
                dp.v = *(const uint4v*)(srow + c * 8);
                pp.v = *(const uint4v*)(Prow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++)
                    tsum += bf2f(pp.u[e]) * bf2f(dp.u[e]);
            }
            tsum += __shfl_xor(tsum, 1);
            tsum += __shfl_xor(tsum, 2);
            ushort_t* dsrow = Cp + gi * ldc + n0 + qt;
            ushort_t* dsr_row = (ushort_t*)p.out2
                + (long)(pid % p.nInner) * p.out2_sin
                + (long)(pid / p.nInner) * p.out2_sout + gi * p.out2_ld
                + n0 + qt;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } dp, pp, o1, o2;
                dp.v = *(const uint4v*)(srow + c * 8);
                pp.v = *(const uint4v*)(Prow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    int gj = n0 + qt + c * 8 + e;
                    float v = p.alpha2 * bf2f(pp.u[e])
                              * (bf2f(dp.u[e]) - tsum);
                    if (p.self_mask && gj == gi) v = 0.f;
                    if (mrow && mrow[c * 8 + e]) v = 0.f;
                    o1.u[e] = f2bf(v);
                    o2.u[e] = f2bf(v * rn[c * 8 + e]);
                }
                *(uint4v*)(dsrow + c * 8) = o1.v;
                *(uint4v*)(dsr_row + c * 8) = o2.v;
            }
        }
        return;
    }
    {
        int t = threadIdx.x;              // 512 threads: 128 rows x 4 qtrs
        int li = t >> 2;
        int qt = (t & 3) * 64;
        long gi = m0 + li;
        ushort_t* crow = Cp + gi * ldc + n0 + qt;
        ushort_t* orow = out2p ? out2p + gi * p.out2_ld + n0 + qt : nullptr;
        const ushort_t* srow = smem + li * EPI2_ROW + qt;
#pragma unroll
        for (int c = 0; c < 8; c++)
            *(uint4v*)(crow + c * 8) = *(const uint4v*)(srow + c * 8);
        if (orow) {
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x, g;
                x.v = *(const uint4v*)(srow + c * 8);
                if (p.epilogue == 10) {
                    g.v = x.v;
                } else if (p.epilogue == 12) {      // debug: poly, no trans
#pragma unroll
                    for (int e = 0; e < 8; e++) {
                        float xv = bf2f(x.u[e]);
                        g.u[e] = f2bf(xv * (0.5f + 0.1f * xv));
                    }
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
    if (p.colsum_out) {
        // fused bias-grad: this tile's column sums (C values still live in
        // the LDS staging image) accumulated into the f32 output
        float* outp = p.colsum_out + (long)(pid % p.nInner) * p.colsum_sin;
        int c = threadIdx.x;
        if (c < BN3) {
            float ssum = 0.f;
            for (int r = 0; r < 128; r++)
                ssum += bf2f(smem[r * EPI2_ROW + c]);
            atomicAdd(&outp[n0 + c], ssum);
        }
    }
}

void launch_gemm_nt_fast3(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN3, p.M / BM, p.nproblems);
    hipLaunchKernelGGL(gemm_nt_fast3_kernel, grid, dim3(NT3), 0, stream, p);
}



// ------- 3-ring variant of nt_fast3 (counted vmcnt, raw barriers) ------
__global__ __launch_bounds__(NT3) void gemm_nt_fast4_kernel(GemmParams p) {
    __shared__ ushort_t smem[3 * (128 + 256) * FBK];   // 144 KiB, 3-ring
    const int abuf = 128 * FBK, bbuf = 256 * FBK, stride = abuf + bbuf;

    const int pid = blockIdx.z;
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN3;
    const int m0 = (bid % gridDim.y) * BM;

    const ushort_t* Ap;
    const ushort_t* Bp;
    long lda, ldb;
    resolve_ptr2(p.A, p.Atab, p.Atabld, pid, p.nInner, &Ap, &lda);
    resolve_ptr2(p.B, p.Btab, p.Btabld, pid, p.nInner, &Bp, &ldb);

    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int wm = (wid >> 2) * 64;
    const int wn = (wid & 3) * 64;
    const int lrow = lane & 15;
    const int kq = lane >> 4;

    f32x4 acc[4][4] = {};

    // A: 16 chunks over 8 waves (2 each); B: 32 chunks (4 each)
    auto stage = [&](int buf, int k0) {
        ushort_t* Al = smem + buf * stride;
        ushort_t* Bl = Al + abuf;
#pragma unroll
        for (int c = 0; c < 2; c++) {
            int chunk = wid * 2 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g = Ap + (long)(m0 + row) * lda + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g,
                (__attribute__((address_space(3))) unsigned int*)
                    (Al + chunk * 512), 16, 0, 0);
        }
#pragma unroll
        for (int c = 0; c < 4; c++) {
            int chunk = wid * 4 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g = Bp + (long)(n0 + row) * ldb + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g,
                (__attribute__((address_space(3))) unsigned int*)
                    (Bl + chunk * 512), 16, 0, 0);
        }
    };

    const int nk = p.K / FBK;
    // 3-deep ring with counted vmcnt (T3+T4): tile t computes while t+1 is
    // landed/landing and t+2 streams; ONE raw barrier + ONE counted wait
    // per K-tile — the vmcnt(0) drain of the 2-buffer loop never happens.
    stage(0, 0);
    if (nk > 1) stage(1, FBK);
    if (nk > 1)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    for (int kt = 0; kt < nk; kt++) {
        if (kt + 2 < nk) stage((kt + 2) % 3, (kt + 2) * FBK);
        const ushort_t* Al = smem + (kt % 3) * stride;
        const ushort_t* Bl = Al + abuf;
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&Al[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bl[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        if (kt + 1 < nk) {
            if (kt + 2 < nk)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }
    }
    __syncthreads();   // all waves done computing before epilogue staging

    // ---- epilogue: 128x256 tile via LDS, full-line stores ----
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

    float csv[4] = {1.f, 1.f, 1.f, 1.f};
    float bvv[4] = {0.f, 0.f, 0.f, 0.f};
    if (csp && p.epilogue != EPI_SMBWD) {
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
                int lj = wn + j16 * 16 + lrow;             // 0..255
                smem[li * EPI2_ROW + lj] = f2bf(vv[j16] + bvv[j16]);
            }
        }
    }
    __syncthreads();

    if (p.epilogue == EPI_SOFTMAX || p.epilogue == EPI_SMBWD) {
        // Full-row fusions: this tile holds complete rows (N == 256), so
        // the masked row softmax (fwd) / softmax backward (bwd) runs here
        // instead of a separate kernel + a global-memory round trip.
        // 4 consecutive threads (one quartet, same wave) own one row.
        int t = threadIdx.x;
        int li = t >> 2;
        int qt = (t & 3) * 64;
        long gi = m0 + li;
        const bool* mrow =
            p.nlmask ? (const bool*)p.nlmask + gi * p.N + n0 + qt : nullptr;
        ushort_t* srow = smem + li * EPI2_ROW + qt;
        const float SELF = bf2f(f2bf(-5e-4f));
        const float NEG = -3.3895314e38f;

        if (p.epilogue == EPI_SOFTMAX) {
            // pass 1: masks + quartet max
            float mx = -INFINITY;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    int gj = n0 + qt + c * 8 + e;
                    float v = bf2f(x.u[e]);
                    if (p.self_mask && gj == gi) v = SELF;
                    if (mrow && mrow[c * 8 + e]) v = NEG;
                    x.u[e] = f2bf(v);
                    mx = fmaxf(mx, v);
                }
                *(uint4v*)(srow + c * 8) = x.v;   // masked values back
            }
            mx = fmaxf(mx, __shfl_xor(mx, 1));
            mx = fmaxf(mx, __shfl_xor(mx, 2));
            // pass 2: exp + quartet sum (store unnormalized bf16)
            float sum = 0.f;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    float ex = __expf(bf2f(x.u[e]) - mx);
                    sum += ex;
                    x.u[e] = f2bf(ex);
                }
                *(uint4v*)(srow + c * 8) = x.v;
            }
            sum += __shfl_xor(sum, 1);
            sum += __shfl_xor(sum, 2);
            float inv = 1.0f / sum;
            // pass 3: normalize; P goes to global AND stays in LDS for the
            // fused AV product below
            ushort_t* prow = Cp + gi * ldc + n0 + qt;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x;
                x.v = *(const uint4v*)(srow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++)
                    x.u[e] = f2bf(bf2f(x.u[e]) * inv);
                *(uint4v*)(prow + c * 8) = x.v;
                *(uint4v*)(srow + c * 8) = x.v;
            }
            if (p.out2) {
                // ---- fused AV: O[i,:] = sum_j P[i,j] * V[j,:] ----
                // P: LDS [128][EPI2_ROW]; V: global rows via aux fields;
                // O: out2 fields; d passed in p.npatch. K = N = 256.
                __syncthreads();
                const ushort_t* Vp = (const ushort_t*)p.aux_base
                    + (long)(pid % p.nInner) * p.aux_sin
                    + (long)(pid / p.nInner) * p.aux_sout;
                const long ldv = p.aux_ld;
                ushort_t* Op = (ushort_t*)p.out2
                    + (long)(pid % p.nInner) * p.out2_sin
                    + (long)(pid / p.nInner) * p.out2_sout;
                const int dtot = p.npatch;
                // double-buffered transpose-staged V: stage s = (dc, jt)
                // loads the NEXT piece before this stage's MFMAs (VMEM in
                // flight under compute) and transpose-writes it into the
                // other buffer after them — one barrier per stage instead
                // of the single-buffered {stage, sync, mfma, sync} chain.
                ushort_t* Vt0 = smem + 128 * EPI2_ROW;  // [256][FBK] x2
                ushort_t* Vt1 = Vt0 + 256 * FBK;
                const int wrow = (wid >> 2) * 64;
                const int wcol64 = (wid & 3) * 64;
                const int nstage = ((dtot + 255) / 256) * 4;
                const int tt = threadIdx.x, cb = tt & 31, mb = tt >> 5;
                union Pc { uint4v v; ushort_t u[8]; };
                Pc rv[8];
                auto load_piece = [&](int st) {
                    if (tt < 256) {
                        int jt = st & 3, dc = (st >> 2) * 256;
#pragma unroll
                        for (int r = 0; r < 8; r++) {
                            int j = jt * 64 + mb * 8 + r;
                            int gc = dc + cb * 8;
                            if (gc + 7 < dtot)
                                rv[r].v = *(const uint4v*)(Vp
                                    + (long)j * ldv + gc);
                            else
                                rv[r].v = 0;
                        }
                    }
                };
                auto write_piece = [&](int st) {
                    if (tt < 256) {
                        ushort_t* Vt = (st & 1) ? Vt1 : Vt0;
#pragma unroll
                        for (int e = 0; e < 8; e++) {
                            Pc col;
#pragma unroll
                            for (int r = 0; r < 8; r++)
                                col.u[r] = rv[r].u[e];
                            int row = cb * 8 + e;
                            int off = (mb * 8) ^ (swz_row(row) << 3);
                            *(uint4v*)&Vt[row * FBK + off] = col.v;
                        }
                    }
                };
                f32x4 acc2[4][4] = {};
                load_piece(0);
                write_piece(0);
                __syncthreads();
                for (int st = 0; st < nstage; st++) {
                    if (st + 1 < nstage) load_piece(st + 1);
                    const ushort_t* Vt = (st & 1) ? Vt1 : Vt0;
                    const int jt = st & 3, dc = (st >> 2) * 256;
                    short8 paf[2][4], vbf[2][4];
#pragma unroll
                    for (int ss = 0; ss < 2; ss++) {
#pragma unroll
                        for (int i = 0; i < 4; i++) {
                            int row = wrow + i * 16 + lrow;
                            int koff = jt * 64 + ss * 32 + kq * 8;
                            paf[ss][i] = *(const short8*)&smem[
                                row * EPI2_ROW + koff];
                        }
#pragma unroll
                        for (int j = 0; j < 4; j++) {
                            int row = wcol64 + j * 16 + lrow;
                            int off = (ss * 32 + kq * 8)
                                      ^ (swz_row(row) << 3);
                            vbf[ss][j] = *(const short8*)&Vt[
                                row * FBK + off];
                        }
                    }
#pragma unroll
                    for (int ss = 0; ss < 2; ss++)
#pragma unroll
                        for (int i = 0; i < 4; i++)
#pragma unroll
                            for (int j = 0; j < 4; j++)
                                acc2[i][j] =
                                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                        paf[ss][i], vbf[ss][j],
                                        acc2[i][j], 0, 0, 0);
                    if (st + 1 < nstage) write_piece(st + 1);
                    if ((st & 3) == 3) {
                        // dc chunk complete: store O from registers
#pragma unroll
                        for (int i16 = 0; i16 < 4; i16++)
#pragma unroll
                            for (int r = 0; r < 4; r++) {
                                long oi = m0 + wrow + i16 * 16 + kq * 4 + r;
                                ushort_t* orow2 = Op + oi * p.out2_ld;
#pragma unroll
                                for (int j16 = 0; j16 < 4; j16++) {
                                    int oc = dc + wcol64 + j16 * 16 + lrow;
                                    if (oc < dtot)
                                        orow2[oc] = f2bf(acc2[i16][j16][r]);
                                    acc2[i16][j16][r] = 0.f;
                                }
                            }
                    }
                    __syncthreads();
                }
            }
        } else {
            // EPI_SMBWD: dS = alpha2 * P * (dP - sum(P*dP)), masked -> 0;
            // dSr = dS * rnorm[j]. P comes via aux, rnorm via colscale
            // pointers (phase 1 ran with alpha=1, no colscale).
            const ushort_t* Prow = (const ushort_t*)p.aux_base
                + (long)(pid % p.nInner) * p.aux_sin
                + (long)(pid / p.nInner) * p.aux_sout + gi * p.aux_ld
                + n0 + qt;
            const float* rn = (const float*)p.colscale_base
                + (long)(pid % p.nInner) * p.cs_sin
                + (long)(pid / p.nInner) * p.cs_sout + n0 + qt;
            float tsum = 0.f;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } dp, pp;
                dp.v = *(const uint4v*)(srow + c * 8);
                pp.v = *(const uint4v*)(Prow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++)
                    tsum += bf2f(pp.u[e]) * bf2f(dp.u[e]);
            }
            tsum += __shfl_xor(tsum, 1);
            tsum += __shfl_xor(tsum, 2);
            ushort_t* dsrow = Cp + gi * ldc + n0 + qt;
            ushort_t* dsr_row = (ushort_t*)p.out2
                + (long)(pid % p.nInner) * p.out2_sin
                + (long)(pid / p.nInner) * p.out2_sout + gi * p.out2_ld
                + n0 + qt;
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } dp, pp, o1, o2;
                dp.v = *(const uint4v*)(srow + c * 8);
                pp.v = *(const uint4v*)(Prow + c * 8);
#pragma unroll
                for (int e = 0; e < 8; e++) {
                    int gj = n0 + qt + c * 8 + e;
                    float v = p.alpha2 * bf2f(pp.u[e])
                              * (bf2f(dp.u[e]) - tsum);
                    if (p.self_mask && gj == gi) v = 0.f;
                    if (mrow && mrow[c * 8 + e]) v = 0.f;
                    o1.u[e] = f2bf(v);
                    o2.u[e] = f2bf(v * rn[c * 8 + e]);
                }
                *(uint4v*)(dsrow + c * 8) = o1.v;
                *(uint4v*)(dsr_row + c * 8) = o2.v;
            }
        }
        return;
    }
    {
        int t = threadIdx.x;              // 512 threads: 128 rows x 4 qtrs
        int li = t >> 2;
        int qt = (t & 3) * 64;
        long gi = m0 + li;
        ushort_t* crow = Cp + gi * ldc + n0 + qt;
        ushort_t* orow = out2p ? out2p + gi * p.out2_ld + n0 + qt : nullptr;
        const ushort_t* srow = smem + li * EPI2_ROW + qt;
#pragma unroll
        for (int c = 0; c < 8; c++)
            *(uint4v*)(crow + c * 8) = *(const uint4v*)(srow + c * 8);
        if (orow) {
#pragma unroll
            for (int c = 0; c < 8; c++) {
                union { uint4v v; ushort_t u[8]; } x, g;
                x.v = *(const uint4v*)(srow + c * 8);
                if (p.epilogue == 10) {
                    g.v = x.v;
                } else if (p.epilogue == 12) {      // debug: poly, no trans
#pragma unroll
                    for (int e = 0; e < 8; e++) {
                        float xv = bf2f(x.u[e]);
                        g.u[e] = f2bf(xv * (0.5f + 0.1f * xv));
                    }
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
    if (p.colsum_out) {
        // fused bias-grad: this tile's column sums (C values still live in
        // the LDS staging image) accumulated into the f32 output
        float* outp = p.colsum_out + (long)(pid % p.nInner) * p.colsum_sin;
        int c = threadIdx.x;
        if (c < BN3) {
            float ssum = 0.f;
            for (int r = 0; r < 128; r++)
                ssum += bf2f(smem[r * EPI2_ROW + c]);
            atomicAdd(&outp[n0 + c], ssum);
        }
    }
}


void launch_gemm_nt_fast4(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN3, p.M / BM, p.nproblems);
    hipLaunchKernelGGL(gemm_nt_fast4_kernel, grid, dim3(NT3), 0, stream, p);
}


// ------- persistent continuous-ring NT variant (nt5p) -------
// Each block sweeps PERSIST consecutive 128-row M-tiles of ONE 256-col
// N-column. The 3-ring never drains at tile boundaries (the global step
// stream g = tile*nk + kt staggers straight across tiles), so the per-tile
// prologue cost — ~50% of a K=512 block in nt4 (profiles/README
// K-amortization) — is paid once per PERSIST tiles. The epilogue runs from
// registers (scalar stores; no LDS image — the ring owns the arena) in the
// shadow of the next tile's MFMAs. Supports alpha/bias/colscale/GELUGRAD
// and the fused f32 colsum (per-wave atomics, 2/column/tile); no
// softmax/pair/table-C epilogues — the dispatcher routes those to nt4.
template <int PERSIST>
__global__ __launch_bounds__(NT3) void gemm_nt_fast5p_kernel(GemmParams p) {
    __shared__ ushort_t smem[3 * (128 + 256) * FBK];   // 144 KiB, 3-ring
    const int abuf = 128 * FBK, bbuf = 256 * FBK, stride = abuf + bbuf;

    const int pid = blockIdx.z;
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN3;
    const int m0 = (bid % gridDim.y) * (BM * PERSIST);

    const ushort_t* Ap;
    const ushort_t* Bp;
    long lda, ldb;
    resolve_ptr2(p.A, p.Atab, p.Atabld, pid, p.nInner, &Ap, &lda);
    resolve_ptr2(p.B, p.Btab, p.Btabld, pid, p.nInner, &Bp, &ldb);

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
    float* colp = nullptr;
    if (p.colsum_out)
        colp = p.colsum_out + (long)(pid % p.nInner) * p.colsum_sin;
    ushort_t* o2p = nullptr;   // EPI_GELU_PAIR: gelu(pre-act) output
    if (p.epilogue == EPI_GELU_PAIR && p.out2)
        o2p = (ushort_t*)p.out2 + (long)(pid % p.nInner) * p.out2_sin
              + (long)(pid / p.nInner) * p.out2_sout;

    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int wm = (wid >> 2) * 64;
    const int wn = (wid & 3) * 64;
    const int lrow = lane & 15;
    const int kq = lane >> 4;

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

    f32x4 acc[4][4] = {};

    const int nk = p.K / FBK;
    const int TOT = PERSIST * nk;
    auto stage_g = [&](int buf, int g) {
        const int mt0 = m0 + (g / nk) * BM;
        const int k0 = (g % nk) * FBK;
        ushort_t* Al = smem + buf * stride;
        ushort_t* Bl = Al + abuf;
#pragma unroll
        for (int c = 0; c < 2; c++) {
            int chunk = wid * 2 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g_ = Ap + (long)(mt0 + row) * lda + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g_,
                (__attribute__((address_space(3))) unsigned int*)
                    (Al + chunk * 512), 16, 0, 0);
        }
#pragma unroll
        for (int c = 0; c < 4; c++) {
            int chunk = wid * 4 + c;
            int row = chunk * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g_ = Bp + (long)(n0 + row) * ldb + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g_,
                (__attribute__((address_space(3))) unsigned int*)
                    (Bl + chunk * 512), 16, 0, 0);
        }
    };

    stage_g(0, 0);
    if (TOT > 1) stage_g(1, 1);
    if (TOT > 1)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    for (int g = 0; g < TOT; g++) {
        const bool tile_end = (g % nk) == nk - 1;
        // vmcnt retires IN ISSUE ORDER, so at tile-end steps the next
        // stage's glds must be issued AFTER the epilogue stores: the
        // bottom vmcnt(6) then covers [g+1 glds, stores] and leaves g+2's
        // glds in flight. Stores retire once their data leaves the VGPRs
        // (no memory round trip), so that wait stays cheap.
        if (!tile_end && g + 2 < TOT) stage_g((g + 2) % 3, g + 2);
        const ushort_t* Al = smem + (g % 3) * stride;
        const ushort_t* Bl = Al + abuf;
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&Al[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bl[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);

        if (tile_end) {
            // tile done: drain it from registers while the ring streams the
            // next tile
            const long mt0 = m0 + (long)(g / nk) * BM;
            float colacc[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
            for (int i16 = 0; i16 < 4; i16++) {
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    long gi = mt0 + wm + i16 * 16 + kq * 4 + r;
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
                        for (int j16 = 0; j16 < 4; j16++)
                            gx[j16] = bf2f(av[j16]);
                        gelu_grad_vec<4>(gx, gy);
#pragma unroll
                        for (int j16 = 0; j16 < 4; j16++) vv[j16] *= gy[j16];
                    }
                    ushort_t* crow = Cp + gi * ldc + n0 + wn + lrow;
                    ushort_t cbv[4];
#pragma unroll
                    for (int j16 = 0; j16 < 4; j16++) {
                        ushort_t cb = f2bf(vv[j16] + bvv[j16]);
                        // colsum over the bf16-ROUNDED stored values,
                        // matching the LDS-epilogue colsum bit for bit
                        // per summand (only the add order differs, and
                        // that was already atomic-nondeterministic)
                        colacc[j16] += bf2f(cb);
                        cbv[j16] = cb;
                        crow[j16 * 16] = cb;
                    }
                    if (o2p) {
                        // gelu of the bf16-rounded pre-activation:
                        // bitwise the standalone k_gelu pass
                        float gx[4], gy[4];
#pragma unroll
                        for (int j16 = 0; j16 < 4; j16++)
                            gx[j16] = bf2f(cbv[j16]);
                        gelu_f_vec<4>(gx, gy);
                        ushort_t* orow = o2p + gi * p.out2_ld + n0 + wn
                                         + lrow;
#pragma unroll
                        for (int j16 = 0; j16 < 4; j16++)
                            orow[j16 * 16] = f2bf(gy[j16]);
                    }
                    acc[i16][0][r] = 0.f; acc[i16][1][r] = 0.f;
                    acc[i16][2][r] = 0.f; acc[i16][3][r] = 0.f;
                }
            }
            if (colp) {
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++)
                    atomicAdd(&colp[n0 + wn + j16 * 16 + lrow], colacc[j16]);
            }
            if (g + 2 < TOT) stage_g((g + 2) % 3, g + 2);
        }

        if (g + 1 < TOT) {
            if (g + 2 < TOT)
                asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
            else
                asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            __builtin_amdgcn_s_barrier();
        }
    }
}

void launch_gemm_nt_fast5p(const GemmParams& p, hipStream_t stream) {
    // sweep depth: 4 by default (dispatcher guarantees M % 512 == 0);
    // GLOM_NT5P_P=8 selects the deeper sweep when M allows it
    static const int pref = []() {
        const char* e = getenv("GLOM_NT5P_P");
        return e ? atoi(e) : 0;
    }();
    const int P = (pref == 8 && p.M % (BM * 8) == 0) ? 8
                : (pref == 2) ? 2 : 4;
    dim3 grid(p.N / BN3, p.M / (BM * P), p.nproblems);
    if (P == 8)
        hipLaunchKernelGGL(gemm_nt_fast5p_kernel<8>, grid, dim3(NT3), 0,
                           stream, p);
    else if (P == 2)
        hipLaunchKernelGGL(gemm_nt_fast5p_kernel<2>, grid, dim3(NT3), 0,
                           stream, p);
    else
        hipLaunchKernelGGL(gemm_nt_fast5p_kernel<4>, grid, dim3(NT3), 0,
                           stream, p);
}



// --------------------------------------------------------------------- //
// gemm_nt_fast8p: 256x256x64 tile, 8 waves (2M x 4N, 128x64 per wave),
// guide §5 "256² 8-phase template" schedule: the K-step is split into 4
// phases of {ds_read subtile || 2 global_load_lds -> barrier ->
// setprio(1) 16xMFMA setprio(0) -> counted per-wave vmcnt -> barrier}.
// 2 full-tile LDS buffers (128 KiB, 1 block/CU); each wave stages ONLY
// the data it will itself read (its B quarter + the A mfrag-pair of its
// wn slot), ordered B-first, so the counted waits leave later-needed
// halves in flight across phase barriers (T3+T4) and the phase split
// gives setprio something to arbitrate (T5).
//
// Per-wave wait schedule (steady state; derivation in-line):
//   end of phase 0: wave wn==1 waits vmcnt(2)   (its A(t) share)
//   end of phase 1: wave wn==2 waits vmcnt(4)
//   end of phase 2: wave wn==3 waits vmcnt(6)
//   end of phase 3: wave wn==0 waits vmcnt(0), others vmcnt(4)
// When the next tile is not staged (last K-tile) the protecting counts
// drop to 0/0/0/0 (nothing newer in the queue to leave in flight).

#define BM8 256
#define BN8 256
#define NT8 512

__global__ __launch_bounds__(NT8) void gemm_nt_fast8p_kernel(GemmParams p) {
    __shared__ ushort_t smem[2 * (BM8 + BN8) * FBK];   // 128 KiB
    const int abuf = BM8 * FBK, stride = (BM8 + BN8) * FBK;

    const int pid = blockIdx.z;
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN8;   // column-major: n outer
    const int m0 = (bid % gridDim.y) * BM8;

    const ushort_t* Ap;
    const ushort_t* Bp;
    long lda, ldb;
    resolve_ptr2(p.A, p.Atab, p.Atabld, pid, p.nInner, &Ap, &lda);
    resolve_ptr2(p.B, p.Btab, p.Btabld, pid, p.nInner, &Bp, &ldb);

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
    float* colp = nullptr;
    if (p.colsum_out)
        colp = p.colsum_out + (long)(pid % p.nInner) * p.colsum_sin;

    const int wid = threadIdx.x / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int whalf = wid >> 2;          // A half (0: rows 0-127, 1: 128-255)
    const int wq = wid & 3;              // B quarter / A stage slot
    const int wm = whalf * 128;
    const int wn = wq * 64;
    const int lrow = lane & 15;
    const int kq = lane >> 4;

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

    f32x4 acc[8][4] = {};

    // Stage 2 chunks (16 rows, 2 glds/thread) of this wave's OWN data.
    // part 0/1: B quarter rows [wq*64 + whalf*32 + 16*part)
    // part 2/3: A mfrag-pair rows [whalf*128 + wq*32 + 16*(part-2))
    auto stage2 = [&](int buf, int k0, int part) {
        ushort_t* img;
        int row0;
        if (part < 2) {
            img = smem + buf * stride + abuf;            // B image
            row0 = wq * 64 + whalf * 32 + part * 16;
        } else {
            img = smem + buf * stride;                    // A image
            row0 = whalf * 128 + wq * 32 + (part - 2) * 16;
        }
        const ushort_t* src = (part < 2 ? Bp : Ap);
        const long ld = (part < 2 ? ldb : lda);
        const int g0 = (part < 2 ? n0 : m0);
#pragma unroll
        for (int c = 0; c < 2; c++) {
            int row = row0 + c * 8 + (lane >> 3);
            int swz8 = ((lane & 7) ^ swz_row(row)) * 8;
            const ushort_t* g = src + (long)(g0 + row) * ld + k0 + swz8;
            __builtin_amdgcn_global_load_lds(
                (const __attribute__((address_space(1))) unsigned int*)g,
                (__attribute__((address_space(3))) unsigned int*)
                    (img + (row >> 3) * 512), 16, 0, 0);
        }
    };

    const int nk = p.K / FBK;
    // prologue: stage tile 0 fully, drain, enter the loop (tile t's
    // phases stage tile t+1 into the buffer freed at the end of t-1)
#pragma unroll
    for (int part = 0; part < 4; part++) stage2(0, 0, part);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();

    for (int t = 0; t < nk; t++) {
        const ushort_t* Al = smem + (t & 1) * stride;
        const ushort_t* Bl = Al + abuf;
        const bool more = (t + 1) < nk;   // staging tile t+1 this pass
        short8 bfrag[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfrag[s][j] = *(const short8*)&Bl[row * FBK + off];
            }
#pragma unroll
        for (int ph = 0; ph < 4; ph++) {
            short8 af[2][2];
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int i = 0; i < 2; i++) {
                    int row = wm + (2 * ph + i) * 16 + lrow;
                    int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                    af[s][i] = *(const short8*)&Al[row * FBK + off];
                }
            if (more) stage2((t + 1) & 1, (t + 1) * FBK, ph);
            __builtin_amdgcn_s_barrier();
            __builtin_amdgcn_s_setprio(1);
#pragma unroll
            for (int s = 0; s < 2; s++)
#pragma unroll
                for (int i = 0; i < 2; i++)
#pragma unroll
                    for (int j = 0; j < 4; j++)
                        acc[2 * ph + i][j] =
                            __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                                af[s][i], bfrag[s][j], acc[2 * ph + i][j],
                                0, 0, 0);
            __builtin_amdgcn_s_setprio(0);
            // per-wave counted wait protecting the NEXT phase's ds_reads
            if (ph < 3) {
                if (__builtin_amdgcn_readfirstlane(wq) == ph + 1) {
                    if (!more)
                        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                    else if (ph == 0)
                        asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
                    else if (ph == 1)
                        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
                    else
                        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
                }
            } else if (t + 1 < nk) {
                // tile boundary: everyone needs B(t+1); wn==0 also its
                // A(t+1) mfrag 0-1 share (the last 4 it issued)
                if (__builtin_amdgcn_readfirstlane(wq) == 0)
                    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
                else
                    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
            }
            __builtin_amdgcn_s_barrier();
        }
    }

    // ------------- register epilogue (mirrors nt5p, 8 mfrags) ----------
    float colacc[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int i16 = 0; i16 < 8; i16++) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
            long gi = m0 + wm + i16 * 16 + kq * 4 + r;
            float vv[4];
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++)
                vv[j16] = acc[i16][j16][r] * p.alpha * csv[j16];
            if (auxp) {
                const ushort_t* auxrow = auxp + gi * p.aux_ld;
                float gx[4], gy[4];
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++)
                    gx[j16] = bf2f(auxrow[n0 + wn + j16 * 16 + lrow]);
                gelu_grad_vec<4>(gx, gy);
#pragma unroll
                for (int j16 = 0; j16 < 4; j16++) vv[j16] *= gy[j16];
            }
            ushort_t* crow = Cp + gi * ldc + n0 + wn + lrow;
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                ushort_t cb = f2bf(vv[j16] + bvv[j16]);
                colacc[j16] += bf2f(cb);
                crow[j16 * 16] = cb;
            }
        }
    }
    if (colp) {
#pragma unroll
        for (int j16 = 0; j16 < 4; j16++)
            atomicAdd(&colp[n0 + wn + j16 * 16 + lrow], colacc[j16]);
    }
}

void launch_gemm_nt_fast8p(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN8, p.M / BM8, p.nproblems);
    hipLaunchKernelGGL(gemm_nt_fast8p_kernel, grid, dim3(NT8), 0, stream, p);
}

// ------- split-K-only TN variant (32 KiB arena, 4 blocks/CU) -------
__global__ __launch_bounds__(NTHREADS) void gemm_tn_sk_kernel(GemmParams p) {
    __shared__ ushort_t smem[2 * BM * FBK];   // 32 KiB: 4 blocks/CU
    ushort_t* As = smem;
    ushort_t* Bs = smem + BM * FBK;

    int pid = blockIdx.z;
    int slice = 0, k_begin = 0, k_end = p.K;
    if (p.splitk > 1) {
        pid = blockIdx.z % p.nproblems;
        slice = blockIdx.z / p.nproblems;
        int per = ((p.K + FBK - 1) / FBK + p.splitk - 1) / p.splitk * FBK;
        k_begin = slice * per;
        k_end = min(p.K, k_begin + per);
    }
    // XCD-aware block remap (T1), column-major: each XCD die owns a
    // contiguous run of N-columns (all M-tiles of a few n-tiles), so the
    // shared B panel of a column stays resident in that XCD's private L2
    // while the M sweep streams A.
    int nwg = gridDim.x * gridDim.y;
    int bid = blockIdx.y * gridDim.x + blockIdx.x;
    {
        int q = nwg >> 3, r = nwg & 7, xcd = bid & 7, off = bid >> 3;
        bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + off;
    }
    const int n0 = (bid / gridDim.y) * BN;   // column-major: n outer
    const int m0 = (bid % gridDim.y) * BM;

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

    for (int k0 = k_begin; k0 < k_end; k0 += FBK) {
        if (threadIdx.x < 128)
            stage_repack(As, Ap, lda, k0, m0, k_end, threadIdx.x);
        else
            stage_repack(Bs, Bp, ldb, k0, n0, k_end, threadIdx.x - 128);
        __syncthreads();
        short8 af[2][4], bfr[2][4];
#pragma unroll
        for (int s = 0; s < 2; s++) {
#pragma unroll
            for (int i = 0; i < 4; i++) {
                int row = wm + i * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                af[s][i] = *(const short8*)&As[row * FBK + off];
            }
#pragma unroll
            for (int j = 0; j < 4; j++) {
                int row = wn + j * 16 + lrow;
                int off = (s * 32 + kq * 8) ^ (swz_row(row) << 3);
                bfr[s][j] = *(const short8*)&Bs[row * FBK + off];
            }
        }
#pragma unroll
        for (int s = 0; s < 2; s++)
#pragma unroll
            for (int i = 0; i < 4; i++)
#pragma unroll
                for (int j = 0; j < 4; j++)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        af[s][i], bfr[s][j], acc[i][j], 0, 0, 0);
        __syncthreads();
    }

    if (p.splitk > 1) {
        float* ws = p.ws + ((long)slice * p.nproblems + pid) * p.M * p.N;
#pragma unroll
        for (int i16 = 0; i16 < 4; i16++)
#pragma unroll
            for (int j16 = 0; j16 < 4; j16++) {
                int j = n0 + wn + j16 * 16 + lrow;
#pragma unroll
                for (int r = 0; r < 4; r++) {
                    int i = m0 + wm + i16 * 16 + kq * 4 + r;
                    ws[(long)i * p.N + j] = acc[i16][j16][r];
                }
            }
        return;
    }
    // non-split-K falls back to the generic-epilogue variant (dispatch
    // guarantees splitk > 1 here)
}




void launch_gemm_tn_sk(const GemmParams& p, hipStream_t stream) {
    dim3 grid(p.N / BN, p.M / BM, p.nproblems * p.splitk);
    hipLaunchKernelGGL(gemm_tn_sk_kernel, grid, dim3(NTHREADS), 0, stream, p);
}


// ---------------------------------------------------------------- //
// Measured-and-rejected variants (see git history + profiles/README.md):
//  - 256^2 2-phase and phase-pipelined kernels (1 block/CU sync stalls
//    beat the intensity gain: 670-774us vs 463us for the up shape)
//  - 128^2 3-ring (96 KiB LDS halves occupancy: down 292->357us)
//  - 128x256 TN (68 KiB arena halves tn_fast's 4 blocks/CU: dW1 +37%)
//  - 256^2 BK=32 3-ring (paired-row LDS; slower on K=512 and buggy)
//  - reg-pipelined TN 2-buf (issue-early/write-late, 64 KiB, 1 barrier
//    per step): loses to tn_fast/tn_sk on every dW shape (486->477,
//    560->542, 730->703, 775->756 TF; bench 962->943) — 4 blocks/CU
//    cross-block overlap hides the repack latency better than in-kernel
//    pipelining at 2 blocks/CU
//  - 128^2 persistent 3-ring (nt5p idea at BN=128, 96 KiB, 1 block/CU):
//    4-wave/64x64 cut hit 260 VGPRs (1 wave/SIMD) and halved throughput;
//    the 8-wave/64x32 cut (91 VGPRs) still lost on every target shape
//    (down 622->464 TF, dP 457->407; bench 949->908) — at 128-wide
//    tiles, 2 blocks/CU of the plain 2-buf kernel beats any 1-block/CU
//    in-ring overlap. The wide-tile nt5p wins come from intensity AND
//    persistence together, not persistence alone
// The shipping set: gemm_nt_fast (128^2 glds 2-buf), gemm_nt_fast3/4
// (128x256 8-wave, 2-buf / 3-ring), gemm_nt_fast5p (persistent
// continuous-ring, default for M%512==0 plain/GELUGRAD epilogues),
// gemm_tn_fast(+_sk), gemm_nn_fast.
