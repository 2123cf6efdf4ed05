#pragma once
#include <hip/hip_runtime.h>

void launch_add_pos(const void* levels, const void* pos, void* out,
                    long total, int N, int L, int d, hipStream_t s);
void launch_rnorm(const void* levels, float* out, int B, int N, int L, int d,
                  hipStream_t s);
void launch_softmax_fwd(const void* scores, void* probs, const bool* mask,
                        int nprob, int N, int self_mask, hipStream_t s);
void launch_softmax_bwd(const void* P, const void* dP, const float* rnorm,
                        void* dS, void* dSr, const bool* mask, int nprob,
                        int N, int self_mask, float alpha, hipStream_t s);
void launch_knorm_combine(const void* dkhat, const void* lev,
                          const float* rnorm, const void* dv, const void* dq,
                          void* out, int B, int N, int L, int d,
                          hipStream_t s);
void launch_mix_fwd(const void* prev, const void* bu, const void* td,
                    const void* cons, void* out, long total, int L, int d,
                    hipStream_t s);
void launch_mix_bwd(const void* dout, void* dmix, void* dtd, long total,
                    int L, int d, hipStream_t s);
void launch_add4(const void* a, const void* b, const void* c, const void* d,
                 void* out, long total, hipStream_t s);
void launch_gelu(const void* in, void* out, long total, hipStream_t s);
