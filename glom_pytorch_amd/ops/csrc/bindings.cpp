// Torch bindings: compose the generic MFMA GEMM + row-wise kernels into the
// five GLOM hot-path ops (grouped FF fwd/bwd, consensus attention fwd/bwd,
// level mix fwd/bwd). Layout contracts follow SURVEY.md §2.2/§2.3; the
// grouped-conv weights (G*mult*d, d, 1)/(G*d, mult*d, 1) are consumed
// directly as G packed GEMM operands.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstring>
#include <vector>

#include "gemm.h"
#include "aux_kernels.h"
#include "native_ops.h"

namespace {

// Untracked alias of `base` (shares storage, keeps it alive, but has its
// own TensorImpl/version counter). Used for trajectory-slab slices: a
// custom Function may return such an alias even though later steps write
// other slices of the same slab — a torch view would trip the
// view+inplace guard.
torch::Tensor alias_slab_slice(const torch::Tensor& slab, int64_t idx,
                               c10::IntArrayRef sizes) {
    int64_t n = 1;
    for (auto s : sizes) n *= s;
    void* ptr = (char*)slab.data_ptr() + idx * n * slab.element_size();
    auto keep = slab;   // captured by the deleter -> storage stays alive
    return torch::from_blob(
        ptr, sizes, [keep](void*) mutable {}, slab.options());
}

#define CHECK_IN(x)                                                        \
    TORCH_CHECK(x.is_cuda(), #x " must be on GPU");                        \
    TORCH_CHECK(x.is_contiguous(), #x " must be contiguous");              \
    TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")

hipStream_t cur_stream() {
    return c10::hip::getCurrentHIPStream().stream();
}

// nt8p dispatch mode: default ON for K>=2048 shapes (GLOM_NT8P=0 opts
// out; set_nt8p() for within-process A/B, guide §5.4 rule 24). History:
// +15% on the down-projection microbench but -1% end-to-end under the
// pre-overlap stream mix; with the forward-tail/backward overlap the
// balance flipped and nt8p measures +1.8% e2e (1437 -> 1463 img/s,
// same box) — see profiles/README.md round-2 notes.
int g_nt8p_mode = []() {
    const char* e = getenv("GLOM_NT8P");
    return (e && e[0] == '0') ? 0 : 1;
}();

void set_nt8p(bool on) { g_nt8p_mode = on ? 1 : 0; }

// GELU placement: 0 = standalone bandwidth pass after the up-projection
// (round-1 measurement); 1 = EPI_GELU_PAIR fused into the up-projection
// epilogue (nt5p register epilogue writes both Hpre and gelu(Hpre),
// saving the standalone pass's full re-read of Hpre).
// Default ON: bitwise-identical to the standalone pass and measured
// 0.74 -> 0.71 ms on the headline ff-forward (saves the standalone
// pass's full Hpre re-read); e2e within noise. GLOM_GELU_PAIR=0 reverts.
int g_gelu_pair = []() {
    const char* e = getenv("GLOM_GELU_PAIR");
    return (e && e[0] == '0') ? 0 : 1;
}();
void set_gelu_pair(bool on) { g_gelu_pair = on ? 1 : 0; }

void check_launch() {
    hipError_t e = hipGetLastError();
    TORCH_CHECK(e == hipSuccess, "HIP launch failed: ", hipGetErrorString(e));
}

// Central dispatcher: picks the tuned full-tile kernels (glds NT / repack
// TN) when the host-verified preconditions hold, applies split-K when the
// natural grid underfills the 256 CUs (the weight-grad TN GEMMs: tiny
// output, K = B*N tokens; partial f32 slabs summed by gemm_finish), and
// otherwise falls back to the generic predicated kernel.
// `lds_ok` asserts every A/B row stride (incl. table entries) is %8 == 0.
void run_gemm(GemmParams& p, hipStream_t s, const torch::TensorOptions& opts,
              bool lds_ok) {
    const bool no_xform = !(p.A.flags & (OP_GELU | OP_POS))
                          && !(p.B.flags & (OP_GELU | OP_POS));
    const bool nt_fast = p.layout == LAYOUT_NT && lds_ok && no_xform
                         && p.M % 128 == 0 && p.N % 128 == 0
                         && p.K % 64 == 0 && p.K >= 64;
    const bool tn_fast = p.layout == LAYOUT_TN && lds_ok && no_xform
                         && p.M % 128 == 0 && p.N % 128 == 0 && p.K % 8 == 0;
    const bool nn_fast = p.layout == LAYOUT_NN && lds_ok && no_xform
                         && p.M % 128 == 0 && p.N % 128 == 0 && p.K % 8 == 0;

    p.splitk = 1;
    p.ws = nullptr;
    torch::Tensor ws;
    long blocks = ((p.N + 127) / 128) * ((p.M + 127) / 128) * p.nproblems;
    if (p.layout == LAYOUT_TN && p.epilogue == EPI_NONE && !p.has_bias
        && !p.has_colscale && blocks < 1024 && p.K >= 4096) {
        long sk = std::min<long>(16, std::max<long>(1, 2048 / blocks));
        sk = std::min<long>(sk, p.K / 64);
        if (sk > 1) {
            ws = torch::empty({sk * p.nproblems * (long)p.M * p.N},
                              opts.dtype(at::kFloat));
            p.splitk = (int)sk;
            p.ws = ws.data_ptr<float>();
        }
    }
    const bool nt3 = nt_fast && p.N % 256 == 0 && p.N >= 1024;
    // persistent continuous-ring variant (default ON; opt out with
    // GLOM_NT5P=0): one block sweeps 4 M-tiles of a column, register
    // epilogue in the shadow of the next tile's MFMAs, fused colsum kept
    static const bool nt5p_on = []() {
        const char* e = getenv("GLOM_NT5P");
        return !e || e[0] != '0';
    }();
    // N >= 1024 only: at N=512 (2 column panels) the persistent sweep
    // measured SLOWER than the 128^2 kernel on the down-projection
    // (637 -> 573 TF), so that shape keeps nt_fast
    // >= 96 (was 256): at the small-batch inference shapes (video B=8:
    // 192-block grids) the persistent sweep beats nt3 by ~20% (3172 ->
    // 3812 frames/s); training grids are >= 768 and unaffected.
    static const long nt5p_ming = []() {
        const char* e = getenv("GLOM_NT5P_MING");
        return e ? atol(e) : 96;
    }();
    const bool nt5p = nt5p_on && nt3 && p.M % 512 == 0
                      && (long)(p.N / 256) * (p.M / 512) * p.nproblems
                         >= nt5p_ming
                      && !(p.Cflags & OP_TABLE)
                      && (p.epilogue == EPI_NONE
                          || p.epilogue == EPI_GELUGRAD
                          || p.epilogue == EPI_GELU_PAIR);
    static const bool disp_dbg = []() {
        const char* e = getenv("GLOM_DISPATCH_DEBUG");
        return e && e[0] == '1';
    }();
    // 8-phase 256^2 schedule (guide §5 template): per-phase interleave of
    // ds_read / glds / MFMA with counted per-wave vmcnt. GLOM_NT8P=0
    // opts out; set_nt8p() lets microbenches A/B within one process.
    const bool nt8p_on = g_nt8p_mode == 1;
    // measured (profiles/README.md): the 256^2 8-phase tile wins at
    // K >= 2048 (down-projection 685 -> 789 TF, square parity+), while
    // the K=512 up-projection family stays on nt5p's persistent ring
    // (8-phase loses its prologue amortization at 8 K-steps/tile)
    static const long nt8p_mink = []() {
        const char* e = getenv("GLOM_NT8P_MINK");
        return e ? atol(e) : 2048;
    }();
    const bool nt8p = nt8p_on && p.layout == LAYOUT_NT && lds_ok && no_xform
                      && p.M % 256 == 0 && p.N % 256 == 0 && p.K % 64 == 0
                      && p.K >= nt8p_mink
                      && !(p.Cflags & OP_TABLE)
                      && (p.epilogue == EPI_NONE
                          || p.epilogue == EPI_GELUGRAD)
                      && (long)(p.N / 256) * (p.M / 256) * p.nproblems >= 256;
    if (disp_dbg)
        fprintf(stderr,
                "[dispatch] L%d M%ld N%ld K%ld epi%d nt3=%d nt5p=%d on=%d "
                "nt8p=%d ctabflag=%d m512=%d\n",
                p.layout, (long)p.M, (long)p.N, (long)p.K, p.epilogue,
                (int)nt3, (int)nt5p, (int)nt5p_on, (int)nt8p,
                (int)(p.Cflags & OP_TABLE), (int)(p.M % 512 == 0));
    if (nt8p)
        launch_gemm_nt_fast8p(p, s);
    else if (nt5p)
        launch_gemm_nt_fast5p(p, s);
    else if (nt3)
        launch_gemm_nt_fast4(p, s);   // 3-ring counted-vmcnt variant
    else if (nt_fast)
        launch_gemm_nt_fast(p, s);
    else if (tn_fast && p.splitk > 1)
        launch_gemm_tn_sk(p, s);     // 32KB arena: 4 blocks/CU for the
    else if (tn_fast)                // weight-grad split-K launches
        launch_gemm_tn_fast(p, s);
    else if (nn_fast)
        launch_gemm_nn_fast(p, s);
    else
        launch_gemm(p, s);
    check_launch();
    if (p.splitk > 1) {
        launch_gemm_finish(p, s);
        check_launch();
    }
}

GemmParams base_params(int64_t M, int64_t N, int64_t K, int layout,
                       int64_t nproblems, int64_t nInner, float alpha) {
    GemmParams p;
    std::memset(&p, 0, sizeof(p));
    p.M = (int)M; p.N = (int)N; p.K = (int)K;
    p.layout = layout;
    p.nproblems = (int)nproblems;
    p.nInner = (int)nInner;
    p.alpha = alpha;
    p.epilogue = EPI_NONE;
    return p;
}

// ------------------------------------------------------------------ //
// grouped feed-forward (bottom-up: mode 0, top-down: mode 1)
// reference glom_pytorch.py:23-36 (GroupedFeedForward) applied at :134/:136

std::vector<torch::Tensor> grouped_ff_fwd(
        c10::optional<torch::Tensor> tokens_opt, torch::Tensor levels,
        c10::optional<torch::Tensor> pos_opt, torch::Tensor w1,
        torch::Tensor b1, torch::Tensor w2, torch::Tensor b2, int64_t mode) {
    CHECK_IN(levels); CHECK_IN(w1); CHECK_IN(b1); CHECK_IN(w2); CHECK_IN(b2);
    const int64_t B = levels.size(0), N = levels.size(1),
                  L = levels.size(2), d = levels.size(3);
    const int64_t G = (mode == 0) ? L : L - 1;
    TORCH_CHECK(G <= GEMM_MAX_TABLE, "levels > 16 unsupported by table mode");
    const int64_t m4 = w1.size(0) / G;
    const int64_t M = B * N;
    TORCH_CHECK(d % 8 == 0 && m4 % 8 == 0, "dim and mult*dim must be /8");
    TORCH_CHECK(w1.size(0) == G * m4 && w1.size(1) == d);
    TORCH_CHECK(w2.size(0) == G * d && w2.size(1) == m4);

    auto opts = levels.options();
    auto Hpre = torch::empty({G, M, m4}, opts);
    auto Hact = torch::empty({G, M, m4}, opts);
    auto Y = torch::empty({B, N, G, d}, opts);
    hipStream_t s = cur_stream();

    torch::Tensor td_in;
    if (mode == 1) {
        // materialize td_in = levels[..., 1:, :] + pos once, so the GEMM
        // streams it with plain 16B loads / glds
        TORCH_CHECK(pos_opt.has_value(), "top-down needs pos");
        auto pos = pos_opt.value();
        CHECK_IN(pos);
        td_in = torch::empty({B, N, G, d}, opts);
        launch_add_pos(levels.data_ptr(), pos.data_ptr(), td_in.data_ptr(),
                       td_in.numel(), (int)N, (int)L, (int)d, s);
        check_launch();
    }

    // up-projection: Hpre_g = x_g @ W1_g^T + b1_g ; Hact_g = gelu(Hpre_g)
    {
        GemmParams p = base_params(M, m4, d, LAYOUT_NT, G, G, 1.0f);
        if (mode == 0) {
            TORCH_CHECK(tokens_opt.has_value(), "bottom-up needs tokens");
            auto tokens = tokens_opt.value();
            CHECK_IN(tokens);
            p.A.flags = OP_TABLE;
            p.Atab[0] = tokens.data_ptr();
            p.Atabld[0] = d;
            for (int64_t g = 1; g < G; g++) {
                p.Atab[g] = (const char*)levels.data_ptr() + (g - 1) * d * 2;
                p.Atabld[g] = L * d;
            }
        } else {
            p.A.base = td_in.data_ptr();
            p.A.sin = d; p.A.ld = G * d;
        }
        p.B.base = w1.data_ptr(); p.B.sin = m4 * d; p.B.ld = d;
        p.Cbase = Hpre.data_ptr(); p.Csin = M * m4; p.Cld = m4;
        p.bias_base = b1.data_ptr(); p.bias_sin = m4; p.has_bias = 1;
        const bool pair = g_gelu_pair
                          && (M % 512 == 0) && (m4 % 256 == 0)
                          && m4 >= 1024;   // nt5p-eligible shapes only
        if (pair) {
            p.epilogue = EPI_GELU_PAIR;
            p.out2 = Hact.data_ptr();
            p.out2_sin = M * m4; p.out2_ld = m4;
        }
        run_gemm(p, s, opts, true);
        if (!pair) {
            // activation as its own bandwidth-bound pass (round-1 default)
            launch_gelu(Hpre.data_ptr(), Hact.data_ptr(), Hpre.numel(), s);
            check_launch();
        }
    }
    // down-projection: Y_g = Hact_g @ W2_g^T + b2_g
    {
        GemmParams p = base_params(M, d, m4, LAYOUT_NT, G, G, 1.0f);
        p.A.base = Hact.data_ptr(); p.A.sin = M * m4; p.A.ld = m4;
        p.B.base = w2.data_ptr(); p.B.sin = d * m4; p.B.ld = m4;
        p.Cbase = (char*)Y.data_ptr() ; p.Csin = d; p.Cld = G * d;
        p.bias_base = b2.data_ptr(); p.bias_sin = d; p.has_bias = 1;
        run_gemm(p, s, opts, true);
    }
    if (mode != 1) td_in = torch::empty({0}, opts);
    // td_in returned so the backward reuses it (saves re-running
    // k_add_pos per iteration in the weight-grad GEMM)
    return {Y, Hpre, Hact, td_in};
}

std::vector<torch::Tensor> grouped_ff_bwd(
        torch::Tensor dY, c10::optional<torch::Tensor> tokens_opt,
        torch::Tensor levels, c10::optional<torch::Tensor> pos_opt,
        torch::Tensor w1, torch::Tensor w2, torch::Tensor Hpre,
        torch::Tensor Hact, int64_t mode,
        c10::optional<torch::Tensor> w1t_opt = c10::nullopt,
        c10::optional<torch::Tensor> w2t_opt = c10::nullopt,
        c10::optional<torch::Tensor> td_in_opt = c10::nullopt) {
    CHECK_IN(dY); CHECK_IN(levels); CHECK_IN(w1); CHECK_IN(w2); CHECK_IN(Hpre);
    CHECK_IN(Hact);
    const int64_t B = levels.size(0), N = levels.size(1),
                  L = levels.size(2), d = levels.size(3);
    const int64_t G = (mode == 0) ? L : L - 1;
    const int64_t m4 = w1.size(0) / G;
    const int64_t M = B * N;
    auto opts = levels.options();
    hipStream_t s = cur_stream();

    auto dHpre = torch::empty({G, M, m4}, opts);
    // the scattered dX GEMM writes every level slice except one (mode 0:
    // the top slice, which only the top-down path feeds; mode 1: slice 0);
    // zero just that slice instead of a full-tensor fill
    auto dLevels = torch::empty({B, N, L, d}, opts);
    launch_zero_slice(dLevels.data_ptr(), B * N, (int)L, (int)d,
                      mode == 0 ? (int)L - 1 : 0, s);
    check_launch();
    torch::Tensor dTokens;

    // per-group weight transposes turn the NN data grads into NT GEMMs
    // (callers that run many iterations pass them in, computed once)
    auto w1t = w1t_opt.has_value()
        ? w1t_opt.value()
        : w1.view({G, m4, d}).transpose(1, 2).contiguous();   // (G,d,m4)
    auto w2t = w2t_opt.has_value()
        ? w2t_opt.value()
        : w2.view({G, d, m4}).transpose(1, 2).contiguous();   // (G,m4,d)

    torch::Tensor td_in;
    if (mode == 1) {
        if (td_in_opt.has_value() && td_in_opt->numel() > 0) {
            td_in = td_in_opt.value();   // saved by the forward
        } else {
            auto pos = pos_opt.value();
            CHECK_IN(pos);
            td_in = torch::empty({B, N, G, d}, opts);
            launch_add_pos(levels.data_ptr(), pos.data_ptr(),
                           td_in.data_ptr(), td_in.numel(), (int)N, (int)L,
                           (int)d, s);
            check_launch();
        }
    }

    // dHpre = (dY_g @ W2_g) * gelu'(Hpre_g)  -- NT with W2^T; when the
    // 128x256 kernel serves it, the bias grad dB1 = colsum(dHpre) fuses
    // into its epilogue (saves a full re-read of the 400MB dHpre)
    const bool dh_nt3 = (M % 128 == 0) && (m4 % 256 == 0) && (m4 >= 1024)
                        && (d % 64 == 0);
    torch::Tensor db1f;
    {
        GemmParams p = base_params(M, m4, d, LAYOUT_NT, G, G, 1.0f);
        p.A.base = (const char*)dY.data_ptr(); p.A.sin = d; p.A.ld = G * d;
        p.B.base = w2t.data_ptr(); p.B.sin = m4 * d; p.B.ld = d;
        p.Cbase = dHpre.data_ptr(); p.Csin = M * m4; p.Cld = m4;
        p.epilogue = EPI_GELUGRAD;
        p.aux_base = Hpre.data_ptr(); p.aux_sin = M * m4; p.aux_ld = m4;
        if (dh_nt3) {
            db1f = torch::zeros({G, m4}, opts.dtype(at::kFloat));
            p.colsum_out = db1f.data_ptr<float>();
            p.colsum_sin = m4;
        }
        run_gemm(p, s, opts, true);
    }
    // dX_g = dHpre_g @ W1_g -- NT with W1^T, scattered into level slices
    {
        GemmParams p = base_params(M, d, m4, LAYOUT_NT, G, G, 1.0f);
        p.A.base = dHpre.data_ptr(); p.A.sin = M * m4; p.A.ld = m4;
        p.B.base = w1t.data_ptr(); p.B.sin = d * m4; p.B.ld = m4;
        if (mode == 0) {
            dTokens = torch::empty({B, N, d}, opts);
            p.Cflags = OP_TABLE;
            p.Ctab[0] = dTokens.data_ptr();
            p.Ctabld[0] = d;
            for (int64_t g = 1; g < G; g++) {
                p.Ctab[g] = (char*)dLevels.data_ptr() + (g - 1) * d * 2;
                p.Ctabld[g] = L * d;
            }
        } else {
            p.Cbase = (char*)dLevels.data_ptr() + d * 2;
            p.Csin = d; p.Cld = L * d;
        }
        run_gemm(p, s, opts, true);
    }
    // dW1_g[h, j] = sum_m dHpre_g[m, h] * x_g[m, j]
    auto dW1 = torch::empty({G * m4, d}, opts);
    {
        GemmParams p = base_params(m4, d, M, LAYOUT_TN, G, G, 1.0f);
        p.A.base = dHpre.data_ptr(); p.A.sin = M * m4; p.A.ld = m4;
        if (mode == 0) {
            auto tokens = tokens_opt.value();
            CHECK_IN(tokens);
            p.B.flags = OP_TABLE;
            p.Btab[0] = tokens.data_ptr();
            p.Btabld[0] = d;
            for (int64_t g = 1; g < G; g++) {
                p.Btab[g] = (const char*)levels.data_ptr() + (g - 1) * d * 2;
                p.Btabld[g] = L * d;
            }
        } else {
            p.B.base = td_in.data_ptr();
            p.B.sin = d; p.B.ld = G * d;
        }
        p.Cbase = dW1.data_ptr(); p.Csin = m4 * d; p.Cld = d;
        run_gemm(p, s, opts, true);
    }
    // dW2_g[o, h] = sum_m dY_g[m, o] * Hact_g[m, h]
    auto dW2 = torch::empty({G * d, m4}, opts);
    {
        GemmParams p = base_params(d, m4, M, LAYOUT_TN, G, G, 1.0f);
        p.A.base = (const char*)dY.data_ptr(); p.A.sin = d; p.A.ld = G * d;
        p.B.base = Hact.data_ptr(); p.B.sin = M * m4; p.B.ld = m4;
        p.Cbase = dW2.data_ptr(); p.Csin = d * m4; p.Cld = m4;
        run_gemm(p, s, opts, true);
    }
    torch::Tensor dB1;
    if (dh_nt3) {
        dB1 = db1f.flatten().to(at::kBFloat16);
    } else {
        // native deterministic column sum (replaces ATen .sum(1))
        dB1 = torch::empty({G * m4}, opts);
        auto ws = torch::empty({(long)G * colsum_rb(M) * m4},
                               opts.dtype(at::kFloat));
        launch_colsum(dHpre.data_ptr(), ws.data_ptr<float>(),
                      dB1.data_ptr(), (int)G, M, m4, s);
        check_launch();
    }
    // dB2 = colsum over the M token rows of dY viewed as (M, G*d)
    auto dB2 = torch::empty({G * d}, opts);
    {
        auto ws = torch::empty({(long)colsum_rb(M) * G * d},
                               opts.dtype(at::kFloat));
        launch_colsum(dY.data_ptr(), ws.data_ptr<float>(), dB2.data_ptr(),
                      1, M, G * d, s);
        check_launch();
    }
    if (mode != 0) dTokens = torch::empty({0}, opts);
    return {dTokens, dLevels, dW1, dB1, dW2, dB2};
}

// ------------------------------------------------------------------ //
// consensus attention (reference glom_pytorch.py:38-73)

std::vector<torch::Tensor> consensus_fwd(
        torch::Tensor levels, bool attend_self,
        c10::optional<torch::Tensor> mask_opt) {
    CHECK_IN(levels);
    const int64_t B = levels.size(0), N = levels.size(1),
                  L = levels.size(2), d = levels.size(3);
    const int64_t P = B * L;
    auto opts = levels.options();
    hipStream_t s = cur_stream();
    const bool* mask = nullptr;
    if (mask_opt.has_value()) {
        auto m = mask_opt.value();
        TORCH_CHECK(m.is_cuda() && m.is_contiguous()
                    && m.scalar_type() == at::kBool);
        mask = m.data_ptr<bool>();
    }

    const bool lds_ok = (N % 8 == 0) && (d % 8 == 0);
    auto rnorm = torch::empty({B, L, N}, opts.dtype(at::kFloat));
    launch_rnorm(levels.data_ptr(), rnorm.data_ptr<float>(),
                 (int)B, (int)N, (int)L, (int)d, s);
    check_launch();

    auto probs = torch::empty({B, L, N, N}, opts);
    const bool fused_sm = (N == 256) && (d % 64 == 0) && lds_ok;
    const bool fuse_av = fused_sm && !getenv("GLOM_NO_FUSE_AV");
    torch::Tensor out_av;
    if (fuse_av) out_av = torch::empty({B, N, L, d}, opts);
    // scores[i,j] = (q_i . k_j) * rnorm_j * d^-0.5, then masked row softmax
    {
        GemmParams p = base_params(N, N, d, LAYOUT_NT, P, L,
                                   (float)std::pow((double)d, -0.5));
        p.A.base = levels.data_ptr(); p.A.sin = d; p.A.sout = N * L * d;
        p.A.ld = L * d;
        p.B = p.A;
        p.Cbase = probs.data_ptr(); p.Csin = N * N; p.Csout = L * N * N;
        p.Cld = N;
        p.colscale_base = rnorm.data_ptr<float>();
        p.cs_sin = N; p.cs_sout = L * N; p.has_colscale = 1;
        if (fused_sm) {
            // the nt3 tile spans whole rows: softmax fuses into the epilogue
            p.epilogue = EPI_SOFTMAX;
            p.self_mask = attend_self ? 0 : 1;
            p.nlmask = mask;
            p.splitk = 1;
            if (fuse_av) {
                // AV product fused too: O computed from the LDS-resident P
                p.out2 = out_av.data_ptr();
                p.out2_sin = d; p.out2_sout = N * L * d; p.out2_ld = L * d;
                p.aux_base = levels.data_ptr();
                p.aux_sin = d; p.aux_sout = N * L * d; p.aux_ld = L * d;
                p.npatch = (int)d;
            }
            launch_gemm_nt_fast4(p, s);
            check_launch();
        } else {
            run_gemm(p, s, opts, lds_ok);
            launch_softmax_fwd(probs.data_ptr(), probs.data_ptr(), mask,
                               (int)P, (int)N, attend_self ? 0 : 1, s);
            check_launch();
        }
    }
    // out[i,:] = sum_j P[i,j] * levels[j,:]
    if (fuse_av)
        return {out_av, probs, rnorm};
    auto out = torch::empty({B, N, L, d}, opts);
    {
        GemmParams p = base_params(N, d, N, LAYOUT_NN, P, L, 1.0f);
        p.A.base = probs.data_ptr(); p.A.sin = N * N; p.A.sout = L * N * N;
        p.A.ld = N;
        p.B.base = levels.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = out.data_ptr(); p.Csin = d; p.Csout = N * L * d;
        p.Cld = L * d;
        run_gemm(p, s, opts, lds_ok);
    }
    return {out, probs, rnorm};
}

torch::Tensor consensus_bwd(torch::Tensor dOut, torch::Tensor levels,
                            torch::Tensor probs, torch::Tensor rnorm,
                            bool attend_self,
                            c10::optional<torch::Tensor> mask_opt) {
    CHECK_IN(dOut); CHECK_IN(levels); CHECK_IN(probs);
    const int64_t B = levels.size(0), N = levels.size(1),
                  L = levels.size(2), d = levels.size(3);
    const int64_t P = B * L;
    const bool lds_ok = (N % 8 == 0) && (d % 8 == 0);
    auto opts = levels.options();
    hipStream_t s = cur_stream();
    const bool* mask = nullptr;
    if (mask_opt.has_value()) mask = mask_opt.value().data_ptr<bool>();

    // dP[i,j] = dOut_i . v_j, then row softmax backward -> dS, dSr
    const bool fused_sm = (N == 256) && (d % 64 == 0) && lds_ok;
    auto dS = torch::empty({B, L, N, N}, opts);
    auto dSr = torch::empty({B, L, N, N}, opts);
    torch::Tensor dP;
    if (fused_sm) {
        GemmParams p = base_params(N, N, d, LAYOUT_NT, P, L, 1.0f);
        p.A.base = dOut.data_ptr(); p.A.sin = d; p.A.sout = N * L * d;
        p.A.ld = L * d;
        p.B.base = levels.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = dS.data_ptr(); p.Csin = N * N; p.Csout = L * N * N;
        p.Cld = N;
        p.epilogue = EPI_SMBWD;
        p.aux_base = probs.data_ptr();
        p.aux_sin = N * N; p.aux_sout = L * N * N; p.aux_ld = N;
        p.out2 = dSr.data_ptr();
        p.out2_sin = N * N; p.out2_sout = L * N * N; p.out2_ld = N;
        p.colscale_base = rnorm.data_ptr<float>();
        p.cs_sin = N; p.cs_sout = L * N; p.has_colscale = 1;
        p.self_mask = attend_self ? 0 : 1;
        p.nlmask = mask;
        p.alpha2 = (float)std::pow((double)d, -0.5);
        p.splitk = 1;
        launch_gemm_nt_fast4(p, s);
        check_launch();
    } else {
        dP = torch::empty({B, L, N, N}, opts);
        GemmParams p = base_params(N, N, d, LAYOUT_NT, P, L, 1.0f);
        p.A.base = dOut.data_ptr(); p.A.sin = d; p.A.sout = N * L * d;
        p.A.ld = L * d;
        p.B.base = levels.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = dP.data_ptr(); p.Csin = N * N; p.Csout = L * N * N;
        p.Cld = N;
        run_gemm(p, s, opts, lds_ok);
        launch_softmax_bwd(probs.data_ptr(), dP.data_ptr(),
                           rnorm.data_ptr<float>(), dS.data_ptr(),
                           dSr.data_ptr(), mask, (int)P, (int)N,
                           attend_self ? 0 : 1,
                           (float)std::pow((double)d, -0.5), s);
        check_launch();
    }

    // dv[j,:] = sum_i P[i,j] dOut[i,:]
    auto dv = torch::empty({B, N, L, d}, opts);
    {
        GemmParams p = base_params(N, d, N, LAYOUT_TN, P, L, 1.0f);
        p.A.base = probs.data_ptr(); p.A.sin = N * N; p.A.sout = L * N * N;
        p.A.ld = N;
        p.B.base = dOut.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = dv.data_ptr(); p.Csin = d; p.Csout = N * L * d;
        p.Cld = L * d;
        run_gemm(p, s, opts, lds_ok);
    }
    // dq[i,:] = sum_j dSr[i,j] levels[j,:]
    auto dq = torch::empty({B, N, L, d}, opts);
    {
        GemmParams p = base_params(N, d, N, LAYOUT_NN, P, L, 1.0f);
        p.A.base = dSr.data_ptr(); p.A.sin = N * N; p.A.sout = L * N * N;
        p.A.ld = N;
        p.B.base = levels.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = dq.data_ptr(); p.Csin = d; p.Csout = N * L * d;
        p.Cld = L * d;
        run_gemm(p, s, opts, lds_ok);
    }
    // dkhat[j,:] = sum_i dS[i,j] levels[i,:]   (layout (B,L,N,d))
    auto dkhat = torch::empty({B, L, N, d}, opts);
    {
        GemmParams p = base_params(N, d, N, LAYOUT_TN, P, L, 1.0f);
        p.A.base = dS.data_ptr(); p.A.sin = N * N; p.A.sout = L * N * N;
        p.A.ld = N;
        p.B.base = levels.data_ptr(); p.B.sin = d; p.B.sout = N * L * d;
        p.B.ld = L * d;
        p.Cbase = dkhat.data_ptr(); p.Csin = N * d; p.Csout = L * N * d;
        p.Cld = d;
        run_gemm(p, s, opts, lds_ok);
    }
    auto dLevels = torch::empty({B, N, L, d}, opts);
    launch_knorm_combine(dkhat.data_ptr(), levels.data_ptr(),
                         rnorm.data_ptr<float>(), dv.data_ptr(),
                         dq.data_ptr(), dLevels.data_ptr(),
                         (int)B, (int)N, (int)L, (int)d, s);
    check_launch();
    return dLevels;
}

// ------------------------------------------------------------------ //
// level mixing (reference glom_pytorch.py:128-144)

// When (slab, slab_idx) is given, the output is written straight into
// slab[slab_idx] and returned as an untracked alias — the return_all
// trajectory is then materialized in place, with no torch.stack copy
// (SURVEY.md §2.3 note on the trajectory slab).
torch::Tensor level_mix_fwd(torch::Tensor levels, torch::Tensor bu,
                            torch::Tensor td, torch::Tensor cons,
                            c10::optional<torch::Tensor> slab = c10::nullopt,
                            int64_t slab_idx = 0) {
    CHECK_IN(levels); CHECK_IN(bu); CHECK_IN(td); CHECK_IN(cons);
    const int64_t L = levels.size(2), d = levels.size(3);
    TORCH_CHECK(d % 8 == 0, "dim must be a multiple of 8");
    torch::Tensor out;
    if (slab.has_value()) {
        TORCH_CHECK(slab->is_contiguous()
                    && slab->scalar_type() == at::kBFloat16
                    && slab->numel() >= (slab_idx + 1) * levels.numel(),
                    "bad trajectory slab");
        out = alias_slab_slice(slab.value(), slab_idx, levels.sizes());
    } else {
        out = torch::empty_like(levels);
    }
    launch_mix_fwd(levels.data_ptr(), bu.data_ptr(), td.data_ptr(),
                   cons.data_ptr(), out.data_ptr(), levels.numel(), (int)L,
                   (int)d, cur_stream());
    check_launch();
    return out;
}

std::vector<torch::Tensor> level_mix_bwd(torch::Tensor dout) {
    CHECK_IN(dout);
    const int64_t B = dout.size(0), N = dout.size(1), L = dout.size(2),
                  d = dout.size(3);
    TORCH_CHECK(d % 8 == 0, "dim must be a multiple of 8");
    auto dmix = torch::empty_like(dout);
    auto dtd = torch::empty({B, N, L - 1, d}, dout.options());
    launch_mix_bwd(dout.data_ptr(), dmix.data_ptr(), dtd.data_ptr(),
                   dout.numel(), (int)L, (int)d, cur_stream());
    check_launch();
    return {dmix, dtd};
}



// ------------------------------------------------------------------ //
// One full GLOM iteration as a single op pair (reference
// glom_pytorch.py:131-145): forward chains bottom-up, top-down, consensus
// and the level mix in one extension call; backward additionally sums the
// four levels-gradient contributions in one fused pass instead of three
// autograd accumulations.

std::vector<torch::Tensor> glom_step_fwd(
        torch::Tensor tokens, torch::Tensor levels, torch::Tensor pos,
        torch::Tensor bw1, torch::Tensor bb1, torch::Tensor bw2,
        torch::Tensor bb2, torch::Tensor tw1, torch::Tensor tb1,
        torch::Tensor tw2, torch::Tensor tb2, bool attend_self,
        c10::optional<torch::Tensor> mask,
        c10::optional<torch::Tensor> slab = c10::nullopt,
        int64_t slab_idx = 0) {
    auto bu = grouped_ff_fwd(tokens, levels, c10::nullopt, bw1, bb1, bw2,
                             bb2, 0);
    auto td = grouped_ff_fwd(c10::nullopt, levels, pos, tw1, tb1, tw2,
                             tb2, 1);
    auto at = consensus_fwd(levels, attend_self, mask);
    auto out = level_mix_fwd(levels, bu[0], td[0], at[0], slab, slab_idx);
    return {out, bu[1], bu[2], td[1], td[2], at[1], at[2], td[3]};
}

std::vector<torch::Tensor> glom_step_bwd(
        torch::Tensor dnew, torch::Tensor tokens, torch::Tensor levels,
        torch::Tensor pos, torch::Tensor bw1, torch::Tensor bw2,
        torch::Tensor tw1, torch::Tensor tw2, torch::Tensor buHpre,
        torch::Tensor buHact, torch::Tensor tdHpre, torch::Tensor tdHact,
        torch::Tensor probs, torch::Tensor rnorm, bool attend_self,
        c10::optional<torch::Tensor> mask,
        c10::optional<torch::Tensor> bw1t, c10::optional<torch::Tensor> bw2t,
        c10::optional<torch::Tensor> tw1t,
        c10::optional<torch::Tensor> tw2t,
        c10::optional<torch::Tensor> td_in = c10::nullopt) {
    const int64_t L = levels.size(2);
    auto mix = level_mix_bwd(dnew);        // {dmix, dtd}
    auto bu = grouped_ff_bwd(mix[0], tokens, levels, c10::nullopt, bw1,
                             bw2, buHpre, buHact, 0, bw1t, bw2t);
    auto td = grouped_ff_bwd(mix[1], c10::nullopt, levels, pos, tw1, tw2,
                             tdHpre, tdHact, 1, tw1t, tw2t, td_in);
    auto dAttn = consensus_bwd(mix[0], levels, probs, rnorm, attend_self,
                               mask);
    auto dLevels = torch::empty_like(levels);
    launch_add4(mix[0].data_ptr(), bu[1].data_ptr(), td[1].data_ptr(),
                dAttn.data_ptr(), dLevels.data_ptr(), levels.numel(),
                cur_stream());
    check_launch();
    // dPos[n,:] = sum over batch and levels 1..L-1 of the top-down input
    // grads (native deterministic reduction, reference glom_pytorch.py:136)
    const int64_t B = levels.size(0), N = levels.size(1),
                  d = levels.size(3);
    auto dPos = torch::empty({N, d}, levels.options());
    const int BCH = (int)std::min<int64_t>(B, 8);
    auto ws = torch::empty({BCH * N * d}, levels.options().dtype(at::kFloat));
    launch_dpos(td[1].data_ptr(), dPos.data_ptr(), ws.data_ptr<float>(),
                BCH, (int)B, (int)N, (int)L, (int)d, cur_stream());
    check_launch();
    return {bu[0], dLevels, dPos, bu[2], bu[3], bu[4], bu[5],
            td[2], td[3], td[4], td[5]};
}

// Synthetic single-GEMM microbenchmark for kernel tuning (used by
// scripts/gemmbench.py; not part of the model path).
double bench_gemm(int64_t M, int64_t N, int64_t K, int64_t layout,
                  int64_t nproblems, int64_t epilogue, int64_t reps) {
    auto opts = torch::TensorOptions().dtype(at::kBFloat16)
                    .device(at::kCUDA);
    torch::Tensor A, Bt;
    if (layout == LAYOUT_TN)
        A = torch::randn({nproblems, K, M}, opts) * 0.05;
    else
        A = torch::randn({nproblems, M, K}, opts) * 0.05;
    if (layout == LAYOUT_NT)
        Bt = torch::randn({nproblems, N, K}, opts) * 0.05;
    else
        Bt = torch::randn({nproblems, K, N}, opts) * 0.05;
    auto C = torch::empty({nproblems, M, N}, opts);
    auto bias = torch::randn({nproblems, N}, opts);
    auto aux = torch::randn({nproblems, M, N}, opts);
    auto out2 = torch::empty({nproblems, M, N}, opts);
    hipStream_t s = cur_stream();

    auto run = [&]() {
        GemmParams p = base_params(M, N, K, (int)layout, nproblems,
                                   nproblems, 1.0f);
        p.A.base = A.data_ptr();
        p.A.sin = A.size(1) * A.size(2); p.A.ld = A.size(2);
        p.B.base = Bt.data_ptr();
        p.B.sin = Bt.size(1) * Bt.size(2); p.B.ld = Bt.size(2);
        p.Cbase = C.data_ptr(); p.Csin = M * N; p.Cld = N;
        p.epilogue = (int)epilogue;
        if (epilogue == EPI_GELUGRAD) {
            p.aux_base = aux.data_ptr(); p.aux_sin = M * N; p.aux_ld = N;
        }
        if (epilogue == EPI_GELU_PAIR || epilogue == 10 || epilogue == 12) {
            p.bias_base = bias.data_ptr(); p.bias_sin = N; p.has_bias = 1;
            p.out2 = out2.data_ptr(); p.out2_sin = M * N; p.out2_ld = N;
        }
        run_gemm(p, s, opts, true);
    };
    for (int i = 0; i < 3; i++) run();
    hipEvent_t e0, e1;
    hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0, s);
    for (int64_t i = 0; i < reps; i++) run();
    hipEventRecord(e1, s);
    hipEventSynchronize(e1);
    float ms = 0;
    hipEventElapsedTime(&ms, e0, e1);
    hipEventDestroy(e0); hipEventDestroy(e1);
    return ms / reps;
}

void add4_into(torch::Tensor a, torch::Tensor b, torch::Tensor c,
               torch::Tensor d, torch::Tensor out) {
    CHECK_IN(a); CHECK_IN(b); CHECK_IN(c); CHECK_IN(d); CHECK_IN(out);
    launch_add4(a.data_ptr(), b.data_ptr(), c.data_ptr(), d.data_ptr(),
                out.data_ptr(), out.numel(), cur_stream());
    check_launch();
}


// ------------------------------------------------------------------ //
// Fine-grained FF backward stages (stream-forked by the python layer):
// dH (with fused dB1), then dX and dW independently.

std::vector<torch::Tensor> ff_bwd_dh(torch::Tensor dY, torch::Tensor w2t,
                                     torch::Tensor Hpre) {
    CHECK_IN(dY); CHECK_IN(w2t); CHECK_IN(Hpre);
    const int64_t G = Hpre.size(0), M = Hpre.size(1), m4 = Hpre.size(2);
    const int64_t d = dY.size(3);
    auto opts = dY.options();
    hipStream_t s = cur_stream();
    auto dHpre = torch::empty({G, M, m4}, opts);
    const bool fused = (M % 128 == 0) && (m4 % 256 == 0) && (m4 >= 1024)
                       && (d % 64 == 0);
    torch::Tensor db1f;
    GemmParams p = base_params(M, m4, d, LAYOUT_NT, G, G, 1.0f);
    p.A.base = (const char*)dY.data_ptr(); p.A.sin = d; p.A.ld = G * d;
    p.B.base = w2t.data_ptr(); p.B.sin = m4 * d; p.B.ld = d;
    p.Cbase = dHpre.data_ptr(); p.Csin = M * m4; p.Cld = m4;
    p.epilogue = EPI_GELUGRAD;
    p.aux_base = Hpre.data_ptr(); p.aux_sin = M * m4; p.aux_ld = m4;
    if (fused) {
        db1f = torch::zeros({G, m4}, opts.dtype(at::kFloat));
        p.colsum_out = db1f.data_ptr<float>();
        p.colsum_sin = m4;
    }
    run_gemm(p, s, opts, true);
    torch::Tensor dB1;
    if (fused) {
        dB1 = db1f.flatten().to(at::kBFloat16);
    } else {
        dB1 = torch::empty({G * m4}, opts);
        auto cws = torch::empty({(long)G * colsum_rb(M) * m4},
                                opts.dtype(at::kFloat));
        launch_colsum(dHpre.data_ptr(), cws.data_ptr<float>(),
                      dB1.data_ptr(), (int)G, M, m4, s);
        check_launch();
    }
    return {dHpre, dB1};
}

std::vector<torch::Tensor> ff_bwd_dx(torch::Tensor dHpre, torch::Tensor w1t,
                                     c10::optional<torch::Tensor> tokens_opt,
                                     int64_t B, int64_t N, int64_t L,
                                     int64_t mode) {
    CHECK_IN(dHpre); CHECK_IN(w1t);
    const int64_t G = dHpre.size(0), M = dHpre.size(1), m4 = dHpre.size(2);
    const int64_t d = w1t.numel() / (G * m4);
    auto opts = dHpre.options();
    hipStream_t s = cur_stream();
    torch::Tensor dTokens;
    auto dLevels = torch::empty({B, N, L, d}, opts);
    launch_zero_slice(dLevels.data_ptr(), B * N, (int)L, (int)d,
                      mode == 0 ? (int)L - 1 : 0, s);
    check_launch();
    GemmParams p = base_params(M, d, m4, LAYOUT_NT, G, G, 1.0f);
    p.A.base = dHpre.data_ptr(); p.A.sin = M * m4; p.A.ld = m4;
    p.B.base = w1t.data_ptr(); p.B.sin = d * m4; p.B.ld = m4;
    if (mode == 0) {
        auto tokens = tokens_opt.value();
        CHECK_IN(tokens);
        dTokens = torch::empty({B, N, d}, opts);
        p.Cflags = OP_TABLE;
        p.Ctab[0] = dTokens.data_ptr();
        p.Ctabld[0] = d;
        for (int64_t g = 1; g < G; g++) {
            p.Ctab[g] = (char*)dLevels.data_ptr() + (g - 1) * d * 2;
            p.Ctabld[g] = L * d;
        }
    } else {
        p.Cbase = (char*)dLevels.data_ptr() + d * 2;
        p.Csin = d; p.Cld = L * d;
    }
    run_gemm(p, s, opts, true);
    if (mode != 0) dTokens = torch::empty({0}, opts);
    return {dTokens, dLevels};
}

std::vector<torch::Tensor> ff_bwd_dw(
        torch::Tensor dY, torch::Tensor dHpre,
        c10::optional<torch::Tensor> tokens_opt, torch::Tensor levels,
        c10::optional<torch::Tensor> pos_opt, torch::Tensor Hact,
        int64_t mode,
        c10::optional<torch::Tensor> td_in_opt = c10::nullopt) {
    CHECK_IN(dY); CHECK_IN(dHpre); CHECK_IN(levels); CHECK_IN(Hact);
    const int64_t B = levels.size(0), N = levels.size(1),
                  L = levels.size(2), d = levels.size(3);
    const int64_t G = dHpre.size(0), M = dHpre.size(1), m4 = dHpre.size(2);
    auto opts = levels.options();
    hipStream_t s = cur_stream();
    torch::Tensor td_in;
    if (mode == 1) {
        if (td_in_opt.has_value() && td_in_opt->numel() > 0) {
            td_in = td_in_opt.value();   // saved by the forward
        } else {
            auto pos = pos_opt.value();
            CHECK_IN(pos);
            td_in = torch::empty({B, N, G, d}, opts);
            launch_add_pos(levels.data_ptr(), pos.data_ptr(),
                           td_in.data_ptr(), td_in.numel(), (int)N, (int)L,
                           (int)d, s);
            check_launch();
        }
    }
    auto dW1 = torch::empty({G * m4, d}, opts);
    {
        GemmParams p = base_params(m4, d, M, LAYOUT_TN, G, G, 1.0f);
        p.A.base = dHpre.data_ptr(); p.A.sin = M * m4; p.A.ld = m4;
        if (mode == 0) {
            auto tokens = tokens_opt.value();
            CHECK_IN(tokens);
            p.B.flags = OP_TABLE;
            p.Btab[0] = tokens.data_ptr();
            p.Btabld[0] = d;
            for (int64_t g = 1; g < G; g++) {
                p.Btab[g] = (const char*)levels.data_ptr() + (g - 1) * d * 2;
                p.Btabld[g] = L * d;
            }
        } else {
            p.B.base = td_in.data_ptr();
            p.B.sin = d; p.B.ld = G * d;
        }
        p.Cbase = dW1.data_ptr(); p.Csin = m4 * d; p.Cld = d;
        run_gemm(p, s, opts, true);
    }
    auto dW2 = torch::empty({G * d, m4}, opts);
    {
        GemmParams p = base_params(d, m4, M, LAYOUT_TN, G, G, 1.0f);
        p.A.base = (const char*)dY.data_ptr(); p.A.sin = d; p.A.ld = G * d;
        p.B.base = Hact.data_ptr(); p.B.sin = M * m4; p.B.ld = m4;
        p.Cbase = dW2.data_ptr(); p.Csin = d * m4; p.Cld = m4;
        run_gemm(p, s, opts, true);
    }
    auto dB2 = torch::empty({G * d}, opts);
    {
        auto cws = torch::empty({(long)colsum_rb(M) * G * d},
                                opts.dtype(at::kFloat));
        launch_colsum(dY.data_ptr(), cws.data_ptr<float>(), dB2.data_ptr(),
                      1, M, G * d, s);
        check_launch();
    }
    return {dW1, dW2, dB2};
}

// ------------------------------------------------------------------ //
// patch embedding (reference glom_pytorch.py:94-97,114): patchify
// rearrange + Linear(p^2*3 -> dim), K-padded to a /64 boundary so the
// tuned NT kernel serves the GEMM (zero columns are exact no-ops).

static int64_t pad_k(int64_t k) { return (k + 63) / 64 * 64; }

std::vector<torch::Tensor> patch_embed_fwd(torch::Tensor img,
                                           torch::Tensor w, torch::Tensor b,
                                           int64_t P) {
    CHECK_IN(img); CHECK_IN(w); CHECK_IN(b);
    const int64_t B = img.size(0), C = img.size(1), H = img.size(2),
                  W = img.size(3);
    const int64_t S = W / P, N = (H / P) * S;
    const int64_t K0 = P * P * C, Kp = pad_k(K0), dout = w.size(0);
    TORCH_CHECK(w.size(1) == K0, "embed weight/patch mismatch");
    auto opts = img.options();
    hipStream_t s = cur_stream();

    auto X = torch::empty({B, N, Kp}, opts);
    launch_patchify(img.data_ptr(), X.data_ptr(), (int)B, (int)C, (int)H,
                    (int)W, (int)P, (int)Kp, s);
    check_launch();
    auto Wp = torch::empty({dout, Kp}, opts);
    launch_pad_cols(w.data_ptr(), Wp.data_ptr(), dout, (int)K0, (int)Kp, s);
    check_launch();

    auto tokens = torch::empty({B, N, dout}, opts);
    GemmParams p = base_params(B * N, dout, Kp, LAYOUT_NT, 1, 1, 1.0f);
    p.A.base = X.data_ptr(); p.A.sin = 0; p.A.ld = Kp;
    p.B.base = Wp.data_ptr(); p.B.sin = 0; p.B.ld = Kp;
    p.Cbase = tokens.data_ptr(); p.Csin = 0; p.Cld = dout;
    p.bias_base = b.data_ptr(); p.bias_sin = 0; p.has_bias = 1;
    run_gemm(p, s, opts, true);
    return {tokens, X};
}

std::vector<torch::Tensor> patch_embed_bwd(torch::Tensor dTokens,
                                           torch::Tensor X, torch::Tensor w,
                                           int64_t P, int64_t imgH,
                                           int64_t imgW, bool need_dimg) {
    CHECK_IN(dTokens); CHECK_IN(X); CHECK_IN(w);
    const int64_t B = dTokens.size(0), N = dTokens.size(1),
                  dout = dTokens.size(2);
    const int64_t Kp = X.size(2), K0 = w.size(1), C = K0 / (P * P);
    const int64_t M = B * N;
    auto opts = dTokens.options();
    hipStream_t s = cur_stream();

    // dWp[o, k] = sum_m dTokens[m, o] * X[m, k]   (TN, split-K eligible)
    auto dWp = torch::empty({dout, Kp}, opts);
    {
        GemmParams p = base_params(dout, Kp, M, LAYOUT_TN, 1, 1, 1.0f);
        p.A.base = dTokens.data_ptr(); p.A.sin = 0; p.A.ld = dout;
        p.B.base = X.data_ptr(); p.B.sin = 0; p.B.ld = Kp;
        p.Cbase = dWp.data_ptr(); p.Csin = 0; p.Cld = Kp;
        run_gemm(p, s, opts, true);
    }
    auto dW = torch::empty({dout, K0}, opts);
    launch_slice_cols(dWp.data_ptr(), dW.data_ptr(), dout, (int)Kp,
                      (int)K0, s);
    check_launch();

    auto dB = torch::empty({dout}, opts);
    {
        auto cws = torch::empty({(long)colsum_rb(M) * dout},
                                opts.dtype(at::kFloat));
        launch_colsum(dTokens.data_ptr(), cws.data_ptr<float>(),
                      dB.data_ptr(), 1, M, dout, s);
        check_launch();
    }

    torch::Tensor dImg = torch::empty({0}, opts);
    if (need_dimg) {
        auto Wp = torch::empty({dout, Kp}, opts);
        launch_pad_cols(w.data_ptr(), Wp.data_ptr(), dout, (int)K0,
                        (int)Kp, s);
        check_launch();
        // dX[m, k] = sum_o dTokens[m, o] * Wp[o, k]   (NN)
        auto dX = torch::empty({B, N, Kp}, opts);
        GemmParams p = base_params(M, Kp, dout, LAYOUT_NN, 1, 1, 1.0f);
        p.A.base = dTokens.data_ptr(); p.A.sin = 0; p.A.ld = dout;
        p.B.base = Wp.data_ptr(); p.B.sin = 0; p.B.ld = Kp;
        p.Cbase = dX.data_ptr(); p.Csin = 0; p.Cld = Kp;
        run_gemm(p, s, opts, true);
        dImg = torch::empty({B, C, imgH, imgW}, opts);
        launch_unpatchify(dX.data_ptr(), dImg.data_ptr(), (int)B, (int)C,
                          (int)imgH, (int)imgW, (int)P, (int)Kp, s);
        check_launch();
    }
    return {dW, dB, dImg};
}

torch::Tensor dpos(torch::Tensor dLevels) {
    CHECK_IN(dLevels);
    const int64_t B = dLevels.size(0), N = dLevels.size(1),
                  L = dLevels.size(2), d = dLevels.size(3);
    auto out = torch::empty({N, d}, dLevels.options());
    const int BCH = (int)std::min<int64_t>(B, 8);
    auto ws = torch::empty({BCH * N * d},
                           dLevels.options().dtype(at::kFloat));
    launch_dpos(dLevels.data_ptr(), out.data_ptr(),
                d % 8 == 0 ? ws.data_ptr<float>() : nullptr, BCH,
                (int)B, (int)N, (int)L, (int)d, cur_stream());
    check_launch();
    return out;
}

// ------------------------------------------------------------------ //
// fused AdamW: one kernel pass over all parameter tensors (bf16 grads ->
// fp32 m/v/master -> bf16 param write-back) with the deterministic
// global-norm grad clip fused. step_dev is a persistent device counter
// (incremented on-device so the op is hipGraph-replay-safe).

void fused_adamw(std::vector<torch::Tensor> grads,
                 std::vector<torch::Tensor> masters,
                 std::vector<torch::Tensor> m1s,
                 std::vector<torch::Tensor> m2s,
                 std::vector<torch::Tensor> params,
                 double lr, double b1, double b2, double eps, double wd,
                 double max_norm, torch::Tensor partials, torch::Tensor norm,
                 torch::Tensor step_dev) {
    const size_t nt = grads.size();
    TORCH_CHECK(nt > 0 && nt <= OPT_MAX_T, "fused_adamw: 1..", OPT_MAX_T,
                " tensors supported, got ", nt);
    TORCH_CHECK(masters.size() == nt && m1s.size() == nt
                && m2s.size() == nt && params.size() == nt);
    TORCH_CHECK(partials.numel() >= OPT_NPART
                && partials.scalar_type() == at::kFloat);
    TORCH_CHECK(norm.numel() >= 1 && norm.scalar_type() == at::kFloat);
    TORCH_CHECK(step_dev.numel() >= 1
                && step_dev.scalar_type() == at::kFloat);
    OptTable t;
    std::memset(&t, 0, sizeof(t));
    t.nt = (int)nt;
    long cum = 0;
    for (size_t i = 0; i < nt; i++) {
        CHECK_IN(grads[i]); CHECK_IN(params[i]);
        TORCH_CHECK(masters[i].is_contiguous()
                    && masters[i].scalar_type() == at::kFloat);
        TORCH_CHECK(m1s[i].is_contiguous() && m2s[i].is_contiguous());
        const long n = grads[i].numel();
        TORCH_CHECK(masters[i].numel() == n && m1s[i].numel() == n
                    && m2s[i].numel() == n && params[i].numel() == n);
        t.g[i] = (const unsigned short*)grads[i].data_ptr();
        t.mw[i] = masters[i].data_ptr<float>();
        t.m1[i] = m1s[i].data_ptr<float>();
        t.m2[i] = m2s[i].data_ptr<float>();
        t.pw[i] = (unsigned short*)params[i].data_ptr();
        t.cum[i] = cum;
        cum += n;
    }
    t.cum[nt] = cum;
    hipStream_t s = cur_stream();
    launch_grad_norm(t, partials.data_ptr<float>(), norm.data_ptr<float>(),
                     step_dev.data_ptr<float>(), s);
    check_launch();
    launch_adamw(t, (float)lr, (float)b1, (float)b2, (float)eps, (float)wd,
                 (float)max_norm, norm.data_ptr<float>(),
                 step_dev.data_ptr<float>(), s);
    check_launch();
}

std::string build_info() {
    return "glom_pytorch_amd HIP extension (gfx950, bf16 MFMA 16x16x32)";
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("grouped_ff_fwd", &grouped_ff_fwd, "grouped FF forward");
    m.def("grouped_ff_bwd", &grouped_ff_bwd, "grouped FF backward",
          py::arg("dY"), py::arg("tokens"), py::arg("levels"),
          py::arg("pos"), py::arg("w1"), py::arg("w2"),
          py::arg("Hpre"), py::arg("Hact"), py::arg("mode"),
          py::arg("w1t") = c10::nullopt,
          py::arg("w2t") = c10::nullopt,
          py::arg("td_in") = c10::nullopt);
    m.def("consensus_fwd", &consensus_fwd, "consensus attention forward");
    m.def("consensus_bwd", &consensus_bwd, "consensus attention backward");
    m.def("level_mix_fwd", &level_mix_fwd, "level mix forward",
          py::arg("levels"), py::arg("bu"), py::arg("td"), py::arg("cons"),
          py::arg("slab") = c10::nullopt, py::arg("slab_idx") = 0);
    m.def("level_mix_bwd", &level_mix_bwd, "level mix backward");
    m.def("glom_step_fwd", &glom_step_fwd, "full GLOM iteration forward",
          py::arg("tokens"), py::arg("levels"), py::arg("pos"),
          py::arg("bw1"), py::arg("bb1"), py::arg("bw2"), py::arg("bb2"),
          py::arg("tw1"), py::arg("tb1"), py::arg("tw2"), py::arg("tb2"),
          py::arg("attend_self"), py::arg("mask"),
          py::arg("slab") = c10::nullopt, py::arg("slab_idx") = 0);
    m.def("glom_step_bwd", &glom_step_bwd, "full GLOM iteration backward",
          py::arg("dnew"), py::arg("tokens"), py::arg("levels"),
          py::arg("pos"), py::arg("bw1"), py::arg("bw2"), py::arg("tw1"),
          py::arg("tw2"), py::arg("buHpre"), py::arg("buHact"),
          py::arg("tdHpre"), py::arg("tdHact"), py::arg("probs"),
          py::arg("rnorm"), py::arg("attend_self"), py::arg("mask"),
          py::arg("bw1t") = c10::nullopt, py::arg("bw2t") = c10::nullopt,
          py::arg("tw1t") = c10::nullopt, py::arg("tw2t") = c10::nullopt,
          py::arg("td_in") = c10::nullopt);
    m.def("patch_embed_fwd", &patch_embed_fwd,
          "patchify + embed GEMM (K1)");
    m.def("patch_embed_bwd", &patch_embed_bwd, "patch embed backward");
    m.def("dpos", &dpos, "positional embedding gradient reduction");
    m.def("fused_adamw", &fused_adamw,
          "fused AdamW + global-norm grad clip (torch semantics)");
    m.def("add4_into", &add4_into, "fused 4-way elementwise sum");
    m.def("ff_bwd_dh", &ff_bwd_dh);
    m.def("ff_bwd_dx", &ff_bwd_dx);
    m.def("ff_bwd_dw", &ff_bwd_dw, py::arg("dY"), py::arg("dHpre"),
          py::arg("tokens"), py::arg("levels"), py::arg("pos"),
          py::arg("Hact"), py::arg("mode"),
          py::arg("td_in") = c10::nullopt);
    m.def("set_nt8p", &set_nt8p, "toggle the 8-phase NT kernel (A/B)");
    m.def("set_gelu_pair", &set_gelu_pair,
          "toggle fused GELU-pair up-projection epilogue (A/B)");
    m.def("build_info", &build_info);
    m.def("bench_gemm", &bench_gemm, "raw GEMM microbench (tuning only)");
}
