// Torch bindings for the brainiak_amd HIP/CDNA4 kernels (gfx950).
//
// The device code lives in fcma_kernels.hip / procrustes.hip /
// tfa_kernels.hip (hand-written HIP, MFMA/LDS — no library GEMMs on the
// FCMA path); this file validates tensors, allocates outputs, and calls
// the extern "C" launchers on the current stream.

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

typedef long long ll;

extern "C" {
void launch_fcma_normalize(float*, ll, ll, ll, int, void*);
void launch_fcma_corr_norm(const void*, const void*, void*, float*, ll, ll,
                           ll, ll, ll, ll, int, int, ll, void*,
                           const void*);
int fcma_corr_variant(void);
int fcma_corr_norm_smem(ll, int);
void launch_fcma_gram_bf16(const void*, float*, ll, ll, ll, ll, void*);
void launch_fcma_gram_fp8(const void*, float*, ll, ll, ll, ll, void*);
void launch_fcma_gram_f32(const float*, float*, ll, ll, ll, void*);
int fcma_corr_norm_z8_supported(ll, int, ll);
void launch_fp8_cvt_probe(const float*, void*, ll, void*);
void launch_fcma_corr_raw(const void*, const void*, void*, ll, ll, ll,
                          ll, ll, int, void*);
void launch_fcma_gram_bf16_norm(const void*, float*, ll, ll, ll, ll,
                                int, void*);
void launch_fcma_corr_gram_duo(const void*, const void*, void*, ll, ll,
                               ll, ll, ll, int, const void*, float*,
                               ll, ll, ll, ll, const float*, float*,
                               ll, ll, int, void*);
void launch_fp8_cvt_probe_sw(const float*, void*, ll, void*);
void launch_fcma_corr_norm_z8(const void*, const void*, void*, ll, ll,
                              ll, ll, ll, int, void*);
void launch_jacobi_eigh(const float*, float*, float*, ll, int, void*);
void launch_tfa_factor(const float*, const float*, const float*, float*,
                       ll, int, void*);
void launch_tfa_recon(const float*, const float*, const float*, float*,
                      ll, ll, int, float, void*);
void launch_svm_cv(const float*, const float*, const int*, const int*,
                   const int*, const int*, int*, ll, int, int, float,
                   float, int, ll, void*);
ll fcma_supported_L(ll);
int fcma_corr_norm_smem(ll, int);
int fcma_fused_gram_supported(ll, int, ll);
void launch_isfc_accum(float*, const void*, ll, ll, int, void*);
void launch_isfc_fused(float*, const void*, const void*, ll, ll, ll,
                       void*);
void launch_stencil3d(const float*, const float*, float*, ll, int, int,
                      int, int, void*);
void launch_fcma_fused_corr_gram(const void*, const void*, float*, ll,
                                 ll, ll, ll, ll, int, void*);
}

static void* cur_stream() {
    return (void*)at::cuda::getCurrentCUDAStream().stream();
}

static void check_3d(const torch::Tensor& t, c10::ScalarType dt,
                     const char* name) {
    TORCH_CHECK(t.is_cuda(), name, " must be a GPU tensor");
    TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
    TORCH_CHECK(t.dim() == 3, name, " must be 3-D");
    TORCH_CHECK(t.scalar_type() == dt, name, " has wrong dtype");
}

// ---------------------------------------------------------------------------

torch::Tensor fcma_normalize_(torch::Tensor corr, int64_t P) {
    check_3d(corr, torch::kFloat32, "corr");
    ll C = corr.size(0), E = corr.size(1), V = corr.size(2);
    TORCH_CHECK(P > 0 && E % P == 0, "epochs_per_subj must divide E");
    launch_fcma_normalize(corr.data_ptr<float>(), C, E, V, (int)P,
                          cur_stream());
    return corr;
}

static int pick_p_for_raw(ll E) {
    for (int p : {8, 4, 2}) if (E % p == 0) return p;
    return 1;
}

torch::Tensor fcma_correlate(torch::Tensor A, torch::Tensor B,
                             int64_t start, int64_t count) {
    check_3d(A, torch::kBFloat16, "A");
    check_3d(B, torch::kBFloat16, "B");
    ll E = A.size(0), L = A.size(1), VA = A.size(2), VB = B.size(2);
    TORCH_CHECK(B.size(0) == E && B.size(1) == L, "A/B epoch shapes differ");
    TORCH_CHECK(start >= 0 && start + count <= VA, "voxel range");
    TORCH_CHECK(fcma_supported_L(L) == L,
                "epoch length must be padded to one of {8,16,24,32,40}");
    int P = pick_p_for_raw(E);
    auto out = torch::empty({count, E, VB},
                            A.options().dtype(torch::kFloat32));
    launch_fcma_corr_norm(A.data_ptr(), B.data_ptr(), nullptr,
                          out.data_ptr<float>(), E, L, VA, VB, start,
                          count, P, /*mode=*/2, E, cur_stream(),
                          nullptr);
    return out;
}

torch::Tensor fcma_corr_norm_z(torch::Tensor A, torch::Tensor B,
                               int64_t start, int64_t count, int64_t P,
                               int64_t padE,
                               c10::optional<torch::Tensor> out_opt,
                               bool raw) {
    check_3d(A, torch::kBFloat16, "A");
    check_3d(B, torch::kBFloat16, "B");
    ll E = A.size(0), L = A.size(1), VA = A.size(2), VB = B.size(2);
    TORCH_CHECK(B.size(0) == E && B.size(1) == L, "A/B epoch shapes differ");
    TORCH_CHECK(start >= 0 && start + count <= VA, "voxel range");
    TORCH_CHECK(P > 0 && E % P == 0, "epochs_per_subj must divide E");
    TORCH_CHECK(P <= 32, "epochs_per_subj > 32: use the staged path");
    TORCH_CHECK(fcma_corr_norm_smem(L, (int)P) <= 160 * 1024,
                "corr tile exceeds gfx950 LDS; use the staged path");
    TORCH_CHECK(fcma_supported_L(L) == L,
                "epoch length must be padded to one of {8,16,24,32,40}");
    ll Eout = std::max((ll)padE, E);
    torch::Tensor Z;
    bool fp8 = false;
    if (out_opt.has_value()) {
        // caller-provided persistent buffer (rows >= E; padding rows
        // must be pre-zeroed once by the caller).  A Float8_e4m3fn
        // buffer selects the fp8 Z path (half the HBM round trip).
        Z = out_opt.value();
        fp8 = Z.scalar_type() == torch::kFloat8_e4m3fn;
        TORCH_CHECK(Z.is_cuda() && Z.is_contiguous()
                    && (fp8 || Z.scalar_type() == torch::kBFloat16)
                    && Z.size(0) >= count && Z.size(1) >= Eout
                    && Z.size(2) == VB, "bad out buffer");
        Eout = Z.size(1);
        if (Z.size(0) != count)
            Z = Z.narrow(0, 0, count);
    } else {
        Z = (Eout == E) ? torch::empty({count, E, VB}, A.options())
                        : torch::zeros({count, Eout, VB}, A.options());
    }
    // dot3s variant reads the A operand voxel-major through the
    // scalar path: hand it the [count, E, L] transposed slice
    torch::Tensor At;
    const void* At_ptr = nullptr;
    if (fcma_corr_variant() >= 3) {
        At = A.narrow(2, start, count).permute({2, 0, 1}).contiguous();
        At_ptr = At.data_ptr();
    }
    // the kernel writes rows [0, E) with row stride Eout directly
    if (raw) {
        TORCH_CHECK(!fp8, "raw correlations are bf16-only (e4m3 cannot "
                          "resolve r near +-1 before Fisher-z)");
        TORCH_CHECK(At_ptr != nullptr && (P == 2 || P == 4),
                    "raw mode needs the dot3s variant and P in {2,4}");
        launch_fcma_corr_raw(At_ptr, B.data_ptr(), Z.data_ptr(),
                             E, L, VB, count, Eout, (int)P,
                             cur_stream());
        return Z;
    }
    if (fp8) {
        TORCH_CHECK(At_ptr != nullptr,
                    "fp8 Z output needs the dot3s corr kernel "
                    "(BRAINIAK_CORR_KERNEL=dot3s, the default)");
        TORCH_CHECK(fcma_corr_norm_z8_supported(E, (int)P, L),
                    "fp8 Z output needs P in {2,4} and even L");
        launch_fcma_corr_norm_z8(At_ptr, B.data_ptr(), Z.data_ptr(),
                                 E, L, VB, count, Eout, (int)P,
                                 cur_stream());
        return Z;
    }
    launch_fcma_corr_norm(A.data_ptr(), B.data_ptr(), Z.data_ptr(),
                          nullptr, E, L, VA, VB, start, count, (int)P,
                          /*mode=*/0, Eout, cur_stream(), At_ptr);
    return Z;
}

torch::Tensor fcma_gram_bf16(torch::Tensor Z, int64_t norm_P) {
    check_3d(Z, torch::kBFloat16, "Z");
    ll C = Z.size(0), E = Z.size(1), V = Z.size(2);
    TORCH_CHECK(E % 64 == 0, "E must be a multiple of 64 (host pads)");
    ll eb = E / 64;
    // V-split for latency hiding: PMC shows the kernel 84 % parked on
    // its serial k-tile loop at 4 blocks/CU — target ~8 blocks/CU
    ll base = C * eb * eb;
    ll ktAll = (V + 63) / 64;
    // measured sweep (profiles/README.md r2): ~8k blocks in flight
    // beats the old ~2k target by 0.3 ms/step; 16k regresses
    ll nsplit = std::min(ktAll, std::max((ll)1, (8191 + base) / base));
    if (const char* e = getenv("BRAINIAK_GRAM_NSPLIT"))
        nsplit = std::min(ktAll, std::max((ll)1, (ll)atoll(e)));
    TORCH_CHECK(norm_P == 0 || norm_P == 2 || norm_P == 4,
                "norm_P must be 0 (pre-normalized) or 2/4");
    if (norm_P)
        TORCH_CHECK(E % 64 == 0 && 64 % norm_P == 0,
                    "fused normalize needs band-aligned subjects");
    auto launch = [&](float* g, ll ns) {
        if (norm_P)
            launch_fcma_gram_bf16_norm(Z.data_ptr(), g, C, E, V, ns,
                                       (int)norm_P, cur_stream());
        else
            launch_fcma_gram_bf16(Z.data_ptr(), g, C, E, V, ns,
                                  cur_stream());
    };
    if (nsplit <= 1) {
        auto G = torch::empty({C, E, E},
                              Z.options().dtype(torch::kFloat32));
        launch(G.data_ptr<float>(), 1);
        return G;
    }
    auto Gp = torch::empty({nsplit, C, E, E},
                           Z.options().dtype(torch::kFloat32));
    launch(Gp.data_ptr<float>(), nsplit);
    return Gp.sum(0);
}

void fcma_corr_gram_duo(torch::Tensor A, torch::Tensor B,
                        int64_t start, int64_t count, int64_t P,
                        torch::Tensor Zout,
                        c10::optional<torch::Tensor> Zprev,
                        c10::optional<torch::Tensor> Gpart,
                        c10::optional<torch::Tensor> Gsum_part,
                        c10::optional<torch::Tensor> Gsum_out,
                        bool shrink) {
    // one grid = raw-corr blocks for [start, start+count) + Gram
    // (+in-register normalize) blocks for the PREVIOUS chunk's Z
    check_3d(A, torch::kBFloat16, "A");
    check_3d(B, torch::kBFloat16, "B");
    ll E = A.size(0), L = A.size(1), VA = A.size(2), VB = B.size(2);
    TORCH_CHECK(start >= 0 && start + count <= VA, "voxel range");
    TORCH_CHECK(P == 2 || P == 4, "duo path needs P in {2,4}");
    TORCH_CHECK(fcma_supported_L(L) == L, "L must be a supported "
                "template length");
    TORCH_CHECK(Zout.is_cuda() && Zout.is_contiguous()
                && Zout.scalar_type() == torch::kBFloat16
                && Zout.size(0) >= count && Zout.size(2) == VB,
                "bad Zout");
    ll Eout = Zout.size(1);
    auto At = A.narrow(2, start, count).permute({2, 0, 1}).contiguous();

    const void* zprev_ptr = nullptr;
    float* g_ptr = nullptr;
    ll Cg = 0, Eg = 0, Vg = 0, nsplit = 1;
    if (Gpart.has_value()) {
        TORCH_CHECK(Zprev.has_value(), "Gpart needs Zprev");
        auto& Zp = Zprev.value();
        auto& G = Gpart.value();
        TORCH_CHECK(Zp.is_cuda() && Zp.is_contiguous()
                    && Zp.scalar_type() == torch::kBFloat16, "bad Zprev");
        TORCH_CHECK(G.is_cuda() && G.is_contiguous() && G.dim() == 4
                    && G.scalar_type() == torch::kFloat32, "bad Gpart");
        Cg = G.size(1); Eg = Zp.size(1); Vg = Zp.size(2);
        nsplit = G.size(0);
        TORCH_CHECK(Eg % 64 == 0 && G.size(2) == Eg && G.size(3) == Eg,
                    "Gpart/Zprev shape mismatch");
        TORCH_CHECK(Zp.size(0) >= Cg, "Zprev rows < Cg");
        zprev_ptr = Zp.data_ptr();
        g_ptr = G.data_ptr<float>();
    }
    const float* gps_ptr = nullptr;
    float* gso_ptr = nullptr;
    ll Cs = 0, nsplit_sum = 1;
    if (Gsum_out.has_value()) {
        TORCH_CHECK(Gsum_part.has_value(), "Gsum_out needs Gsum_part");
        auto& Gp = Gsum_part.value();
        auto& Go = Gsum_out.value();
        TORCH_CHECK(Gp.is_cuda() && Gp.is_contiguous() && Gp.dim() == 4
                    && Gp.scalar_type() == torch::kFloat32,
                    "bad Gsum_part");
        TORCH_CHECK(Go.is_cuda() && Go.is_contiguous() && Go.dim() == 3
                    && Go.scalar_type() == torch::kFloat32,
                    "bad Gsum_out");
        Cs = Go.size(0);
        nsplit_sum = Gp.size(0);
        TORCH_CHECK(Gp.size(2) == Go.size(1)
                    && Gp.size(3) == Go.size(2)
                    && (Eg == 0 || Go.size(1) == Eg),
                    "Gsum shapes mismatch");
        // gsum_body strides by the partials' voxel-row count
        TORCH_CHECK(Gp.size(1) == Cs,
                    "Gsum_part rows must equal Gsum_out rows");
        gps_ptr = Gp.data_ptr<float>();
        gso_ptr = Go.data_ptr<float>();
    }
    launch_fcma_corr_gram_duo(At.data_ptr(), B.data_ptr(),
                              Zout.data_ptr(), E, L, VB, count, Eout,
                              (int)P, zprev_ptr, g_ptr, Cg, Eg, Vg,
                              nsplit, gps_ptr, gso_ptr, Cs, nsplit_sum,
                              shrink ? 1 : 0, cur_stream());
}

torch::Tensor debug_fp8_cvt(torch::Tensor x, bool sw) {
    TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kFloat32);
    auto xc = x.contiguous();
    auto o = torch::empty_like(xc, xc.options()
                                       .dtype(torch::kFloat8_e4m3fn));
    if (sw)
        launch_fp8_cvt_probe_sw(xc.data_ptr<float>(), o.data_ptr(),
                                xc.numel(), cur_stream());
    else
        launch_fp8_cvt_probe(xc.data_ptr<float>(), o.data_ptr(),
                             xc.numel(), cur_stream());
    return o;
}

torch::Tensor fcma_gram_fp8(torch::Tensor Z) {
    TORCH_CHECK(Z.is_cuda() && Z.dim() == 3 && Z.is_contiguous()
                && Z.scalar_type() == torch::kFloat8_e4m3fn,
                "Z must be contiguous [C,E,V] float8_e4m3fn on GPU");
    ll C = Z.size(0), E = Z.size(1), V = Z.size(2);
    TORCH_CHECK(E % 64 == 0, "E must be a multiple of 64 (host pads)");
    TORCH_CHECK(V % 16 == 0, "V must be a multiple of 16 (host pads)");
    ll eb = E / 64;
    ll base = C * eb * eb;
    ll ktAll = (V + 255) / 256;
    ll nsplit = std::min(ktAll, std::max((ll)1, (2047 + base) / base));
    if (nsplit <= 1) {
        auto G = torch::empty({C, E, E},
                              Z.options().dtype(torch::kFloat32));
        launch_fcma_gram_fp8(Z.data_ptr(), G.data_ptr<float>(), C, E, V,
                             1, cur_stream());
        return G;
    }
    auto Gp = torch::empty({nsplit, C, E, E},
                           Z.options().dtype(torch::kFloat32));
    launch_fcma_gram_fp8(Z.data_ptr(), Gp.data_ptr<float>(), C, E, V,
                         nsplit, cur_stream());
    return Gp.sum(0);
}

torch::Tensor fcma_gram(torch::Tensor corr) {
    check_3d(corr, torch::kFloat32, "corr");
    ll C = corr.size(0), E = corr.size(1), V = corr.size(2);
    ll Epad = ((E + 63) / 64) * 64;
    torch::Tensor in = corr;
    if (Epad != E) {
        in = torch::zeros({C, Epad, V}, corr.options());
        in.narrow(1, 0, E).copy_(corr);
    }
    auto G = torch::empty({C, Epad, Epad},
                          corr.options().dtype(torch::kFloat32));
    launch_fcma_gram_f32(in.data_ptr<float>(), G.data_ptr<float>(), C,
                         Epad, V, cur_stream());
    if (Epad != E)
        return G.narrow(1, 0, E).narrow(2, 0, E).contiguous();
    return G;
}

torch::Tensor fcma_fused_gram(torch::Tensor A, torch::Tensor B,
                              int64_t start, int64_t count, int64_t P) {
    ll E = A.size(0), L = A.size(1), VA = A.size(2), VB = B.size(2);
    if (fcma_fused_gram_supported(E, (int)P, L)) {
        // single-kernel path: Z never leaves LDS (see
        // fcma_kernels.hip::k_fused_corr_gram)
        check_3d(A, torch::kBFloat16, "A");
        check_3d(B, torch::kBFloat16, "B");
        TORCH_CHECK(B.size(0) == E && B.size(1) == L,
                    "A/B epoch shapes differ");
        TORCH_CHECK(start >= 0 && start + count <= VA, "voxel range");
        ll cTiles = (count + 7) / 8;
        ll vTiles = std::max((ll)1, (VB + 63) / 64);
        // enough workgroups to fill 256 CUs (1 resident WG per CU),
        // but never more v-splits than v-windows
        int nsplit = (int)std::min(
            vTiles, std::max((ll)1, (511 + cTiles) / cTiles));
        auto G = torch::empty({nsplit, count, E, E},
                              A.options().dtype(torch::kFloat32));
        launch_fcma_fused_corr_gram(A.data_ptr(), B.data_ptr(),
                                    G.data_ptr<float>(), L, VA, VB,
                                    start, count, nsplit, cur_stream());
        return nsplit == 1 ? G.squeeze(0) : G.sum(0);
    }
    ll Epad = ((E + 63) / 64) * 64;
    auto Z = fcma_corr_norm_z(A, B, start, count, P, Epad, c10::nullopt,
                              /*raw=*/false);
    auto G = fcma_gram_bf16(Z, /*norm_P=*/0);
    if (Epad != E)
        return G.narrow(1, 0, E).narrow(2, 0, E).contiguous();
    return G;
}

bool fcma_fused_gram_native(int64_t E, int64_t P, int64_t L) {
    return fcma_fused_gram_supported(E, (int)P, L) != 0;
}

// ---------------------------------------------------------------------------

std::vector<torch::Tensor> jacobi_eigh(torch::Tensor G) {
    check_3d(G, torch::kFloat32, "G");
    ll B = G.size(0);
    int K = (int)G.size(1);
    TORCH_CHECK(G.size(2) == K && K <= 64, "G must be [B,K,K], K<=64");
    auto evecs = torch::empty_like(G);
    auto evals = torch::empty({B, K}, G.options());
    launch_jacobi_eigh(G.data_ptr<float>(), evecs.data_ptr<float>(),
                       evals.data_ptr<float>(), B, K, cur_stream());
    return {evals, evecs};
}

torch::Tensor polar_invsqrt(torch::Tensor G) {
    // G^{-1/2} for a stack of SPD K x K Gram matrices via the batched
    // one-workgroup Jacobi eigensolver.  Lets callers with RAGGED
    // [V_i, K] factors (SRM subjects of different voxel counts) batch
    // the eigensolve while keeping per-subject GEMMs.
    TORCH_CHECK(G.is_cuda() && G.dim() == 3 && G.size(1) == G.size(2),
                "G must be [B,K,K] on GPU");
    TORCH_CHECK(G.size(1) <= 64, "K must be <= 64");
    auto Gf = G.to(torch::kFloat32).contiguous();
    auto ev = jacobi_eigh(Gf);
    auto lam = ev[0].clamp_min(1e-30);
    auto Vc = ev[1];
    return torch::bmm(Vc * lam.rsqrt().unsqueeze(1),
                      Vc.transpose(1, 2));
}

torch::Tensor batched_polar(torch::Tensor A, double perturb) {
    TORCH_CHECK(A.is_cuda() && A.dim() == 3, "A must be [B,V,K] on GPU");
    auto Af = A.to(torch::kFloat32).contiguous();
    ll K = Af.size(2);
    TORCH_CHECK(K <= 64, "K must be <= 64");
    if (perturb != 0.0) {
        Af = Af.clone();
        ll d = std::min(Af.size(1), Af.size(2));
        auto idx = torch::arange(d, torch::TensorOptions()
                                        .dtype(torch::kLong)
                                        .device(Af.device()));
        auto diag = Af.index({torch::indexing::Slice(), idx, idx});
        Af.index_put_({torch::indexing::Slice(), idx, idx},
                      diag + perturb);
    }
    auto G = torch::bmm(Af.transpose(1, 2), Af).contiguous();  // [B,K,K]
    // W = A G^{-1/2}
    return torch::bmm(Af, polar_invsqrt(G));
}

// ---------------------------------------------------------------------------

torch::Tensor tfa_factor(torch::Tensor centers, torch::Tensor widths,
                         torch::Tensor coords) {
    TORCH_CHECK(centers.is_cuda() && centers.dim() == 2 &&
                centers.size(1) == 3, "centers must be [K,3] on GPU");
    TORCH_CHECK(coords.dim() == 2 && coords.size(1) == 3,
                "coords must be [V,3]");
    auto c = centers.to(torch::kFloat32).contiguous();
    auto w = widths.to(torch::kFloat32).contiguous();
    auto x = coords.to(torch::kFloat32).contiguous();
    int K = (int)c.size(0);
    ll V = x.size(0);
    TORCH_CHECK(K <= 128, "K must be <= 128");
    auto F = torch::empty({V, K}, c.options());
    launch_tfa_factor(c.data_ptr<float>(), w.data_ptr<float>(),
                      x.data_ptr<float>(), F.data_ptr<float>(), V, K,
                      cur_stream());
    return F;
}

torch::Tensor tfa_recon(torch::Tensor X, torch::Tensor W, torch::Tensor F,
                        double scale) {
    TORCH_CHECK(X.is_cuda() && X.dim() == 2, "X must be [V,T] on GPU");
    auto Xf = X.to(torch::kFloat32).contiguous();
    auto Wf = W.to(torch::kFloat32).contiguous();
    auto Ff = F.to(torch::kFloat32).contiguous();
    ll V = Xf.size(0), T = Xf.size(1);
    int K = (int)Wf.size(0);
    TORCH_CHECK(Ff.size(0) == V && Ff.size(1) == K && Wf.size(1) == T,
                "shape mismatch");
    auto R = torch::empty({V * T}, Xf.options());
    launch_tfa_recon(Xf.data_ptr<float>(), Wf.data_ptr<float>(),
                     Ff.data_ptr<float>(), R.data_ptr<float>(), V, T, K,
                     (float)scale, cur_stream());
    return R;
}

torch::Tensor svm_cv(torch::Tensor kernels, torch::Tensor y,
                     torch::Tensor train_idx, torch::Tensor test_idx,
                     torch::Tensor n_train, torch::Tensor n_test,
                     double Creg, double tol, int64_t max_iter,
                     int64_t max_n_hint) {
    check_3d(kernels, torch::kFloat32, "kernels");
    ll C = kernels.size(0);
    int E = (int)kernels.size(1);
    TORCH_CHECK(kernels.size(2) == E, "kernels must be [C,E,E]");
    int F = (int)train_idx.size(0);
    TORCH_CHECK(y.is_cuda() && y.scalar_type() == torch::kFloat32 &&
                y.numel() == E, "y must be fp32 [E] on GPU");
    TORCH_CHECK(train_idx.scalar_type() == torch::kInt32 &&
                test_idx.scalar_type() == torch::kInt32, "idx int32");
    TORCH_CHECK(train_idx.size(1) == E && test_idx.size(1) == E,
                "idx must be [F,E]");
    auto nt = n_train.to(torch::kInt32).contiguous();
    auto ns = n_test.to(torch::kInt32).contiguous();
    // max_n_hint lets the caller skip the .item() device sync (needed
    // when this launch is enqueued on a side stream mid-pipeline)
    ll max_n = max_n_hint > 0
        ? max_n_hint
        : std::max<ll>(nt.max().item<int>(), ns.max().item<int>());
    TORCH_CHECK(max_n <= 128,
                "svm_cv supports fold sizes up to 128 samples");
    auto correct = torch::zeros({C, F}, kernels.options()
                                            .dtype(torch::kInt32));
    launch_svm_cv(kernels.data_ptr<float>(), y.data_ptr<float>(),
                  train_idx.data_ptr<int>(), test_idx.data_ptr<int>(),
                  nt.data_ptr<int>(), ns.data_ptr<int>(),
                  correct.data_ptr<int>(), C, E, F, (float)Creg,
                  (float)tol, (int)max_iter, max_n, cur_stream());
    return correct;
}

torch::Tensor stencil3d(torch::Tensor x, torch::Tensor w) {
    TORCH_CHECK(x.is_cuda() && w.is_cuda() && x.is_contiguous()
                && w.is_contiguous()
                && x.scalar_type() == torch::kFloat32
                && w.scalar_type() == torch::kFloat32
                && x.dim() == 4 && w.dim() == 3,
                "x must be [B,X,Y,Z] fp32, w [K,K,K] fp32 on GPU");
    ll K = w.size(0);
    TORCH_CHECK(w.size(1) == K && w.size(2) == K && K % 2 == 1 && K <= 9,
                "w must be cubic with odd K <= 9");
    int r = (int)(K / 2);
    ll B = x.size(0);
    int X = (int)x.size(1), Y = (int)x.size(2), Z = (int)x.size(3);
    TORCH_CHECK(X > 2 * r && Y > 2 * r && Z > 2 * r,
                "volume smaller than the kernel");
    auto out = torch::empty({B, X - 2 * r, Y - 2 * r, Z - 2 * r},
                            x.options());
    launch_stencil3d(x.data_ptr<float>(), w.data_ptr<float>(),
                     out.data_ptr<float>(), B, X, Y, Z, r,
                     cur_stream());
    return out;
}

torch::Tensor isfc_fused_(torch::Tensor acc, torch::Tensor Zs,
                          torch::Tensor Zm) {
    TORCH_CHECK(acc.is_cuda() && acc.is_contiguous() && acc.dim() == 2
                && acc.scalar_type() == torch::kFloat32
                && acc.size(0) == acc.size(1), "acc must be fp32 [V,V]");
    for (auto* Z : {&Zs, &Zm}) {
        TORCH_CHECK(Z->is_cuda() && Z->is_contiguous()
                    && Z->dim() == 3
                    && Z->scalar_type() == torch::kBFloat16,
                    "Z stacks must be bf16 [B,V,T]");
    }
    TORCH_CHECK(Zs.sizes() == Zm.sizes(), "Zs/Zm shape mismatch");
    TORCH_CHECK(Zs.size(1) == acc.size(0), "V mismatch");
    launch_isfc_fused(acc.data_ptr<float>(), Zs.data_ptr(),
                      Zm.data_ptr(), Zs.size(1), Zs.size(2),
                      Zs.size(0), cur_stream());
    return acc;
}

torch::Tensor isfc_accum_(torch::Tensor acc, torch::Tensor M) {
    // M: [V, V] or a [B, V, V] subject stack (the batch amortizes the
    // acc read-modify-write across B matrices in one pass)
    bool m_bf16 = M.scalar_type() == torch::kBFloat16;
    ll B = (M.dim() == 3) ? M.size(0) : 1;
    ll V = acc.size(0);
    TORCH_CHECK(acc.is_cuda() && M.is_cuda() && acc.is_contiguous()
                && M.is_contiguous()
                && acc.scalar_type() == torch::kFloat32
                && (m_bf16 || M.scalar_type() == torch::kFloat32)
                && acc.dim() == 2 && acc.size(1) == V
                && (M.dim() == 2 || M.dim() == 3)
                && M.size(-1) == V && M.size(-2) == V,
                "acc fp32 [V,V] / M fp32-or-bf16 [V,V] or [B,V,V]");
    launch_isfc_accum(acc.data_ptr<float>(), M.data_ptr(),
                      V, B, m_bf16 ? 1 : 0, cur_stream());
    return acc;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("stencil3d", &stencil3d,
          "direct [K,K,K] valid conv over [B,X,Y,Z] (searchlight ball)");
    m.def("isfc_accum_", &isfc_accum_,
          "acc += atanh(clamp((M + M^T)/2)) fused in one pass");
    m.def("isfc_fused_", &isfc_fused_,
          "acc += atanh(sym(Zs_b Zm_b^T)) per subject, M never "
          "materialized (bf16 MFMA tile pairs)");
    m.def("svm_cv", &svm_cv,
          "batched precomputed-kernel SVC cross-validation");
    m.def("fcma_normalize_", &fcma_normalize_,
          "in-place Fisher-z + within-subject z-score [C,E,V]");
    m.def("fcma_correlate", &fcma_correlate,
          "raw chunk correlation [count,E,VB] fp32");
    m.def("fcma_corr_norm_z", &fcma_corr_norm_z,
          "fused corr(+norm) -> Z (bf16/fp8; raw=True defers norm)",
          pybind11::arg("A"), pybind11::arg("B"), pybind11::arg("start"),
          pybind11::arg("count"), pybind11::arg("P"),
          pybind11::arg("padE"),
          pybind11::arg("out") = pybind11::none(),
          pybind11::arg("raw") = false);
    m.def("fcma_gram", &fcma_gram, "per-voxel Gram from fp32 [C,E,V]");
    m.def("debug_fp8_cvt", &debug_fp8_cvt,
          "device to_fp8 conversion probe", pybind11::arg("x"),
          pybind11::arg("sw") = false);
    m.def("fcma_gram_fp8", &fcma_gram_fp8,
          "per-voxel Gram from fp8(e4m3) Z [C,E,V]");
    m.def("fcma_corr_gram_duo", &fcma_corr_gram_duo,
          "one-grid raw-corr(chunk i) + gram(chunk i-1) + partial-sum/"
          "shrink(chunk i-2) co-residency",
          pybind11::arg("A"), pybind11::arg("B"), pybind11::arg("start"),
          pybind11::arg("count"), pybind11::arg("P"),
          pybind11::arg("Zout"),
          pybind11::arg("Zprev") = pybind11::none(),
          pybind11::arg("Gpart") = pybind11::none(),
          pybind11::arg("Gsum_part") = pybind11::none(),
          pybind11::arg("Gsum_out") = pybind11::none(),
          pybind11::arg("shrink") = true);
    m.def("fcma_gram_bf16", &fcma_gram_bf16,
          "per-voxel Gram from bf16 Z [C,E,V]; norm_P>0 applies "
          "Fisher-z + z-score to raw correlations in-tile",
          pybind11::arg("Z"), pybind11::arg("norm_P") = 0);
    m.def("fcma_fused_gram_native", &fcma_fused_gram_native,
          "true when the single-kernel corr+gram path covers (E, P, L)");
    m.def("fcma_fused_gram", &fcma_fused_gram,
          "corr+norm+Gram for a voxel chunk");
    m.def("jacobi_eigh", &jacobi_eigh, "batched KxK symmetric eigensolve");
    m.def("batched_polar", &batched_polar,
          "batched orthogonal Procrustes polar factor");
    m.def("polar_invsqrt", &polar_invsqrt,
          "G^{-1/2} for a stack of SPD KxK Gram matrices");
    m.def("tfa_factor", &tfa_factor, "TFA RBF factor matrix");
    m.def("tfa_recon", &tfa_recon, "TFA residual");
}
