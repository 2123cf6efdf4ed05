// Shared host/device parameter block for the generic GLOM GEMM.
#pragma once

#include <hip/hip_runtime.h>

enum GemmLayout { LAYOUT_NT = 0, LAYOUT_NN = 1, LAYOUT_TN = 2 };
enum OpFlags { OP_GELU = 1, OP_POS = 2, OP_TABLE = 4 };
enum GemmEpilogue {
    EPI_NONE = 0,
    EPI_GELUGRAD = 1,
    EPI_GELU_PAIR = 2,
    // full-row fusions (tile spans the whole row, nt_fast3 only):
    EPI_SOFTMAX = 3,   // consensus scores -> masked row softmax -> P
    EPI_SMBWD = 4,     // dP -> row softmax backward -> dS (C), dSr (out2)
};

#define GEMM_MAX_TABLE 16

// Per-operand addressing: strided mode resolves problem p's base pointer as
//   base + (p % nInner) * sin + (p / nInner) * sout      (element strides)
// or, when OP_TABLE is set, from the explicit pointer table (<=16 problems).
struct OpArg {
    const void* base;
    long sin, sout;
    long ld;     // leading (row) stride in elements
    int flags;   // OP_GELU | OP_POS | OP_TABLE
};

struct GemmParams {
    int M, N, K;
    int layout;          // GemmLayout
    int nproblems, nInner;
    int npatch;          // pos-emb row period for OP_POS
    int epilogue;        // GemmEpilogue
    float alpha;

    OpArg A, B;

    void* Cbase; long Csin, Csout, Cld; int Cflags;

    const void* bias_base; long bias_sin, bias_sout; int has_bias;
    const void* colscale_base; long cs_sin, cs_sout; int has_colscale;
    const void* aux_base; long aux_sin, aux_sout, aux_ld;

    const void* pos; long pos_ld;

    // optional fused column sum of C over the row axis (bias gradient):
    // nt_fast3 adds its tile's column sums into colsum_out[p][col] (f32,
    // caller-zeroed) during the epilogue, saving a full re-read of C.
    float* colsum_out; long colsum_sin;

    // attention-fusion extras (EPI_SOFTMAX / EPI_SMBWD)
    const void* nlmask;     // optional (N,N) bool non-local mask
    int self_mask;          // fill diagonal with -5e-4 before softmax
    float alpha2;           // d^-0.5 applied inside EPI_SMBWD

    // split-K (reduction split): when > 1, the kernel writes per-slice f32
    // partials into ws[slice][problem][M][N] and gemm_finish sums them into
    // C (only EPI_NONE, no bias/colscale). Used for the weight-grad GEMMs
    // whose natural grids underfill the 256 CUs (K = B*N tokens is huge).
    int splitk; float* ws;

    // EPI_GELU_PAIR: C gets the pre-activation (acc*alpha + bias), out2
    // gets gelu(pre-activation). Used by the FF up-projection so the
    // down-projection streams plain activations.
    void* out2; long out2_sin, out2_sout, out2_ld;

    // host-verified: every A/B row stride is a multiple of 8 elements and
    // tiles are full -> the glds/repack fast kernels may run
    int fast_ok;

    const void* Atab[GEMM_MAX_TABLE]; long Atabld[GEMM_MAX_TABLE];
    const void* Btab[GEMM_MAX_TABLE]; long Btabld[GEMM_MAX_TABLE];
    void* Ctab[GEMM_MAX_TABLE]; long Ctabld[GEMM_MAX_TABLE];
};

void launch_gemm(const GemmParams& p, hipStream_t stream);
void launch_gemm_finish(const GemmParams& p, hipStream_t stream);
void launch_gemm_nt_fast(const GemmParams& p, hipStream_t stream);
void launch_gemm_tn_fast(const GemmParams& p, hipStream_t stream);
void launch_gemm_nn_fast(const GemmParams& p, hipStream_t stream);
void launch_gemm_nt_fast3(const GemmParams& p, hipStream_t stream);
void launch_gemm_nt_fast4(const GemmParams& p, hipStream_t stream);
void launch_gemm_nt_fast5p(const GemmParams& p, hipStream_t stream);
void launch_gemm_nt_fast8p(const GemmParams& p, hipStream_t stream);
void launch_gemm_tn_sk(const GemmParams& p, hipStream_t stream);
