// Native-mandate kernels: patch embedding data movement, deterministic
// column-sum / positional-embedding gradient reductions, slice zero-fill,
// and the fused AdamW optimizer update. See native_ops.hip.
#pragma once
#include <hip/hip_runtime.h>

void launch_patchify(const void* img, void* X, int B, int C, int H, int W,
                     int P, int Kp, hipStream_t s);
void launch_unpatchify(const void* dX, void* dImg, int B, int C, int H,
                       int W, int P, int Kp, hipStream_t s);
void launch_pad_cols(const void* src, void* dst, long rows, int Kin, int Kp,
                     hipStream_t s);
void launch_slice_cols(const void* src, void* dst, long rows, int Kp,
                       int Kin, hipStream_t s);
// in (P, M, C) bf16 -> out (P, C) bf16, f32 accumulation, deterministic
// two-stage reduction through the caller-provided f32 partials buffer of
// shape (P, RB, C) where RB = colsum_rb(M).
int colsum_rb(long M);
void launch_colsum(const void* in, float* partials, void* out, int nprob,
                   long M, long C, hipStream_t s);
// dLevels (B,N,L,d) -> dPos (N,d) = sum over b and l in [1,L).
// partials: caller-provided (BCH, N, d) f32 workspace (BCH batch chunks);
// pass nullptr/0 for the scalar single-stage fallback.
void launch_dpos(const void* dlev, void* out, float* partials, int BCH,
                 int B, int N, int L, int d, hipStream_t s);
// zero the single level slice out[:, :, l0, :] of a (B,N,L,d) tensor
void launch_zero_slice(void* out, long BN, int L, int d, int l0,
                       hipStream_t s);

// ---------------- fused AdamW ----------------
#define OPT_MAX_T 24

struct OptTable {
    int nt;
    const unsigned short* g[OPT_MAX_T];   // bf16 grads
    float* mw[OPT_MAX_T];                 // fp32 master weights
    float* m1[OPT_MAX_T];                 // exp_avg
    float* m2[OPT_MAX_T];                 // exp_avg_sq
    unsigned short* pw[OPT_MAX_T];        // bf16 params (written back)
    long cum[OPT_MAX_T + 1];              // prefix numel offsets
};

#define OPT_NPART 1024

// deterministic global grad L2 norm: partials (OPT_NPART f32) -> norm[0];
// the finalize kernel also increments step_dev[0] by 1 (single block).
void launch_grad_norm(const OptTable& t, float* partials, float* norm,
                      float* step_dev, hipStream_t s);
// AdamW update (torch.optim.AdamW semantics incl. decoupled weight decay
// and bias correction), with the global-norm gradient clip fused:
// scale = min(1, max_norm / (norm + 1e-6)) applied to every grad read.
void launch_adamw(const OptTable& t, float lr, float b1, float b2, float eps,
                  float wd, float max_norm, const float* norm,
                  const float* step_dev, hipStream_t s);
