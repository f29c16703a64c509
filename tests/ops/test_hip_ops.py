"""GPU numerics tests for the hand-written HIP/CDNA4 kernels.

Every kernel is validated against a plain PyTorch fp32 reference of the
same op (computed on CPU or via eager torch on device).
"""

import math

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ops():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from brainiak_amd import ops as _ops
    if _ops.load_extension() is None:
        pytest.fail("HIP extension not built — GPU box must load it")
    return _ops


def _zscored_epochs(g, E, L, V, device):
    # pad L to the kernel-supported set (zero rows are inert)
    Lpad = next(opt for opt in (8, 16, 24, 32, 40) if L <= opt)
    out = torch.zeros((E, Lpad, V), dtype=torch.float32)
    for e in range(E):
        m = torch.randn((L, V), generator=g)
        m = (m - m.mean(0)) / m.std(0, unbiased=False).clamp_min(1e-12)
        out[e, :L] = m / math.sqrt(L)
    return out.to(device)


def _ref_normalize(corr, P):
    num = 1.0 + corr
    den = 1.0 - corr
    num = torch.where(num <= 0, torch.full_like(num, 1e-4), num)
    den = torch.where(den <= 0, torch.full_like(den, 1e-4), den)
    z = 0.5 * torch.log(num / den)
    C, E, V = corr.shape
    z = z.view(C, E // P, P, V)
    mean = z.mean(dim=2, keepdim=True)
    var = (z * z).mean(dim=2, keepdim=True) - mean * mean
    inv = torch.where(var > 0, var.rsqrt(), torch.zeros_like(var))
    return ((z - mean) * inv).view(C, E, V)


def test_normalize_matches_reference(ops):
    g = torch.Generator().manual_seed(0)
    corr = (torch.rand((3, 16, 277), generator=g) * 2 - 1).float()
    corr[0, 0, 0] = 1.0
    corr[0, 1, 1] = -1.0
    expected = _ref_normalize(corr.clone(), 4)
    dev = corr.cuda()
    ops.fcma_normalize_(dev, 4)
    assert torch.allclose(dev.cpu(), expected, atol=1e-4)


def test_normalize_large_p_generic_path(ops):
    g = torch.Generator().manual_seed(1)
    corr = (torch.rand((2, 2 * 48, 64), generator=g) * 2 - 1).float()
    expected = _ref_normalize(corr.clone(), 48)
    dev = corr.cuda()
    ops.fcma_normalize_(dev, 48)
    assert torch.allclose(dev.cpu(), expected, atol=1e-4)


def test_correlate_matches_torch(ops):
    g = torch.Generator().manual_seed(2)
    A = _zscored_epochs(g, 8, 12, 300, "cuda")
    B = _zscored_epochs(g, 8, 12, 150, "cuda")
    out = ops.fcma_correlate(A.to(torch.bfloat16).contiguous(),
                             B.to(torch.bfloat16).contiguous(), 17, 41)
    ref = torch.einsum('elc,elv->cev',
                       A[:, :, 17:58].to(torch.bfloat16).float(),
                       B.to(torch.bfloat16).float())
    assert out.shape == (41, 8, 150)
    assert torch.allclose(out.cpu(), ref.cpu(), atol=2e-2, rtol=2e-2)


def test_gram_f32_identity_asymmetric(ops):
    """Transpose-detecting check of the f32 MFMA fragment maps (guide G9):
    asymmetric Z, compare G = Z Z^T elementwise."""
    g = torch.Generator().manual_seed(3)
    Z = torch.randn((2, 64, 96), generator=g).float()
    Z[0, 0, :] = torch.arange(96).float() / 96.0  # strongly asymmetric
    G = ops.fcma_gram(Z.cuda())
    ref = torch.bmm(Z, Z.transpose(1, 2))
    assert torch.allclose(G.cpu(), ref, atol=1e-3, rtol=1e-4)


def test_gram_f32_nonmultiple_shapes(ops):
    g = torch.Generator().manual_seed(4)
    Z = torch.randn((3, 24, 100), generator=g).float()  # E=24 → pad to 64
    G = ops.fcma_gram(Z.cuda())
    ref = torch.bmm(Z, Z.transpose(1, 2))
    assert G.shape == (3, 24, 24)
    assert torch.allclose(G.cpu(), ref, atol=1e-3, rtol=1e-4)


def test_gram_bf16_matches_reference(ops):
    g = torch.Generator().manual_seed(5)
    Z = torch.randn((2, 128, 77), generator=g).to(torch.bfloat16)
    G = ops.fcma_gram_bf16(Z.cuda().contiguous())
    Zf = Z.float()
    ref = torch.bmm(Zf, Zf.transpose(1, 2))
    assert G.shape == (2, 128, 128)
    assert torch.allclose(G.cpu(), ref, atol=0.5, rtol=2e-2)


def test_fused_gram_matches_staged_cpu(ops):
    g = torch.Generator().manual_seed(6)
    E, L, V, P = 16, 12, 256, 4
    A = _zscored_epochs(g, E, L, V, "cpu")
    B = _zscored_epochs(g, E, L, 192, "cpu")
    Ab = A.to(torch.bfloat16)
    Bb = B.to(torch.bfloat16)
    G = ops.fcma_fused_gram(Ab.cuda().contiguous(), Bb.cuda().contiguous(),
                            5, 32, P)
    # CPU reference from the same bf16-rounded inputs
    corr = torch.einsum('elc,elv->cev', Ab.float()[:, :, 5:37], Bb.float())
    nc = _ref_normalize(corr, P)
    ref = torch.bmm(nc, nc.transpose(1, 2))
    assert G.shape == (32, E, E)
    assert torch.allclose(G.cpu(), ref, atol=1.0, rtol=3e-2)


@pytest.mark.parametrize("E", [32, 128])
def test_svm_cv_kernel_matches_sklearn(ops, E):
    """E=128 exercises the 2-dual-variables-per-lane path
    (fold sizes > 64)."""
    from sklearn import model_selection, svm as sksvm

    from brainiak_amd.fcma.svm import _accuracy_gpu_hip
    g = torch.Generator().manual_seed(11)
    n = 12
    y = np.array([0, 1] * (E // 2))
    kernels = []
    for i in range(n):
        sep = 2.0 * i / n
        X = torch.randn((E, 10), generator=g).numpy() + sep * y[:, None]
        kernels.append((X @ X.T).astype(np.float32))
    kt = torch.tensor(np.stack(kernels)).cuda()
    accs = _accuracy_gpu_hip(kt, y, num_folds=4, C=1.0, tol=1e-3)
    skf = model_selection.StratifiedKFold(n_splits=4, shuffle=False)
    for i in range(n):
        ref = model_selection.cross_val_score(
            sksvm.SVC(kernel='precomputed', C=1.0), kernels[i], y=y,
            cv=skf).mean()
        assert abs(accs[i] - ref) <= 0.15, (i, accs[i], ref)
    # average agreement should be tight
    refs = [model_selection.cross_val_score(
        sksvm.SVC(kernel='precomputed', C=1.0), kernels[i], y=y,
        cv=skf).mean() for i in range(n)]
    assert abs(np.mean(accs) - np.mean(refs)) < 0.05


def test_batched_polar_matches_svd(ops):
    g = torch.Generator().manual_seed(7)
    A = torch.randn((5, 200, 50), generator=g).float().cuda()
    W = ops.batched_polar(A, 0.0)
    for b in range(5):
        U, _, Vt = torch.linalg.svd(A[b].cpu().double(),
                                    full_matrices=False)
        ref = (U @ Vt).float()
        assert torch.allclose(W[b].cpu(), ref, atol=5e-3)
    # orthogonality
    WtW = torch.bmm(W.transpose(1, 2), W).cpu()
    eye = torch.eye(50).expand(5, -1, -1)
    assert torch.allclose(WtW, eye, atol=5e-3)


def test_jacobi_eigh(ops):
    g = torch.Generator().manual_seed(8)
    M = torch.randn((4, 32, 32), generator=g).float()
    G = torch.bmm(M, M.transpose(1, 2)).cuda().contiguous()
    evals, evecs = ops.jacobi_eigh(G)
    for b in range(4):
        lam_ref = torch.linalg.eigvalsh(G[b].cpu().double())
        lam = evals[b].cpu().double().sort().values
        assert torch.allclose(lam, lam_ref, atol=1e-2, rtol=1e-4)
        # V diag(lam) V^T == G
        rec = (evecs[b].cpu() * evals[b].cpu()) @ evecs[b].cpu().T
        assert torch.allclose(rec, G[b].cpu(), atol=1e-2, rtol=1e-3)


def test_tfa_factor_and_recon(ops):
    g = torch.Generator().manual_seed(9)
    K, V, T = 20, 1000, 50
    centers = torch.randn((K, 3), generator=g).float().cuda()
    widths = (torch.rand((K,), generator=g) * 4 + 1).float().cuda()
    coords = torch.randn((V, 3), generator=g).float().cuda()
    F = ops.tfa_factor(centers, widths, coords)
    d = coords[:, None, :] - centers[None, :, :]
    ref_F = torch.exp(-(d * d).sum(-1) / widths[None, :])
    assert torch.allclose(F, ref_F, atol=1e-4, rtol=1e-4)

    X = torch.randn((V, T), generator=g).float().cuda()
    W = torch.randn((K, T), generator=g).float().cuda()
    R = ops.tfa_recon(X, W, F, 0.5)
    ref_R = (0.5 * (X - F @ W)).reshape(-1)
    assert torch.allclose(R, ref_R, atol=1e-3, rtol=1e-3)


def test_fused_corr_gram_native_e64(ops):
    """The single-kernel corr+gram path (E=64, P=4) against the CPU
    oracle and the two-kernel product."""
    from brainiak_amd.ops import load_extension
    ext = load_extension()
    assert ext.fcma_fused_gram_native(64, 4, 16)
    g = torch.Generator().manual_seed(7)
    E, L, V, P = 64, 16, 300, 4      # V non-multiple of 64 (tail window)
    # distinct A/B: the A==B self-correlation diagonal sits ON the
    # 1-r<=0 clamp boundary, where fp summation order flips the clamp
    # (reference behaviour, intentionally noise-driven) — keep the
    # oracle away from it
    A = _zscored_epochs(g, E, L, 64, "cpu")
    B = _zscored_epochs(g, E, L, V, "cpu")
    Ab = A.to(torch.bfloat16)
    Bb = B.to(torch.bfloat16)
    dA = Ab.cuda().contiguous()
    dB = Bb.cuda().contiguous()
    G = ops.fcma_fused_gram(dA, dB, 3, 40, P)      # C=40: c-tile tail
    corr = torch.einsum('elc,elv->cev', Ab.float()[:, :, 3:43], Bb.float())
    nc = _ref_normalize(corr, P)
    ref = torch.bmm(nc, nc.transpose(1, 2))
    assert G.shape == (40, E, E)
    assert torch.allclose(G.cpu(), ref, atol=1.0, rtol=3e-2)
    assert torch.allclose(G, G.transpose(1, 2))    # exact symmetry
    # two-kernel product from the same inputs (same bf16 z, same 64-wide
    # k-tiling) — tight agreement expected
    Z = ext.fcma_corr_norm_z(dA, dB, 3, 40, P, 64, None)
    G2 = ops.fcma_gram_bf16(Z)
    assert torch.allclose(G.cpu(), G2.cpu(), atol=5e-2, rtol=1e-3)


@pytest.mark.parametrize("P,L", [(2, 8), (2, 40), (4, 24), (4, 40),
                                 (8, 16), (8, 32), (16, 16), (16, 40),
                                 (5, 16), (3, 24)])
def test_corr_norm_z_all_templates(ops, P, L):
    """Every (epochs_per_subj, epoch_len) kernel template variant —
    incl. the generic runtime-P path (P=5, 3) and every padded L —
    against the fp32 oracle."""
    from brainiak_amd.ops import load_extension
    ext = load_extension()
    g = torch.Generator().manual_seed(100 * P + L)
    nsubj = 4 if P <= 8 else 3
    E, V = P * nsubj, 210
    A = _zscored_epochs(g, E, L - 2, 96, "cpu")
    B = _zscored_epochs(g, E, L - 2, V, "cpu")
    assert A.shape[1] == L                 # padded to the named L
    Ab, Bb = A.to(torch.bfloat16), B.to(torch.bfloat16)
    Epad = ((E + 63) // 64) * 64
    Z = ext.fcma_corr_norm_z(Ab.cuda().contiguous(),
                             Bb.cuda().contiguous(), 7, 33, P, Epad,
                             None)
    corr = torch.einsum('elc,elv->cev', Ab.float()[:, :, 7:40],
                        Bb.float())
    ref = _ref_normalize(corr, P)
    got = Z[:, :E, :].float().cpu()
    # mask columns whose within-subject FISHER-Z variance is near the
    # fp noise floor: there the z-score is sign(noise) / the
    # var<=0 -> 0 clamp (sharpest at P=2, where z = sign(z0 - z1))
    zf = 0.5 * torch.log((1 + corr).clamp_min(1e-4)
                         / (1 - corr).clamp_min(1e-4))
    zf = zf.view(corr.shape[0], E // P, P, -1)
    var = zf.var(dim=2, unbiased=False, keepdim=True)
    ok = (var > 1e-6).expand_as(zf).reshape_as(ref)
    assert ok.float().mean() > 0.95
    assert torch.allclose(got[ok], ref[ok], atol=3e-2, rtol=3e-2), \
        (P, L, (got[ok] - ref[ok]).abs().max().item())
    # padding rows stay zero
    if Epad != E:
        assert torch.all(Z[:, E:, :].float().cpu() == 0)


@pytest.mark.parametrize("V", [64, 150, 333])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_isfc_accum_matches_torch(ops, V, dtype):
    g = torch.Generator().manual_seed(V)
    M = (torch.randn((V, V), generator=g) * 0.4).to(dtype)
    M = M.cuda().contiguous()
    acc0 = torch.randn((V, V), generator=g).cuda().contiguous()
    got = ops.isfc_accum_(acc0.clone(), M)
    Mf = M.float()
    sym = (Mf + Mf.T) / 2
    ref = acc0 + torch.atanh(sym.clamp(-1 + 1e-7, 1 - 1e-7))
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-5), \
        (got - ref).abs().max().item()


@pytest.mark.parametrize("shape,r", [((2, 18, 15, 17), 3),
                                     ((1, 9, 9, 9), 1),
                                     ((3, 20, 12, 11), 2)])
def test_stencil3d_matches_conv3d(ops, shape, r):
    g = torch.Generator().manual_seed(r)
    x = torch.randn(shape, generator=g).cuda().contiguous()
    K = 2 * r + 1
    w = torch.rand((K, K, K), generator=g).cuda().contiguous()
    w = w * (torch.rand((K, K, K), generator=g).cuda() > 0.3)  # zeros
    got = ops.stencil3d(x, w)
    ref = torch.nn.functional.conv3d(x[:, None], w[None, None])[:, 0]
    assert got.shape == ref.shape
    assert torch.allclose(got, ref, atol=1e-4, rtol=1e-4), \
        (got - ref).abs().max().item()


def test_gram_fp8_identity_asymmetric(ops):
    """Transpose-detecting fragment-map check for the fp8 MFMA Gram:
    asymmetric Z, values exactly representable in e4m3."""
    g = torch.Generator().manual_seed(13)
    # e4m3-exact values: multiples of 0.25 in [-2, 2]
    Z = (torch.randint(-8, 9, (2, 64, 128), generator=g).float() / 4.0)
    Z[0, 0, :] = (torch.arange(128) % 8).float() / 4.0  # asymmetric row
    Z8 = Z.to(torch.float8_e4m3fn).cuda().contiguous()
    G = ops.fcma_gram_fp8(Z8)
    ref = torch.bmm(Z, Z.transpose(1, 2))
    # inputs exact in e4m3 and fp32 accumulate → exact match expected
    assert torch.allclose(G.cpu(), ref, atol=1e-3, rtol=1e-5)


def test_gram_fp8_matches_fp32_reference(ops):
    """Random z-score-scale data: fp8 Gram within quantization error of
    the fp32 oracle (relative to the diagonal scale)."""
    g = torch.Generator().manual_seed(14)
    C, E, V = 3, 64, 1024
    Z = torch.randn((C, E, V), generator=g).float().clamp_(-1.74, 1.74)
    Z8 = Z.to(torch.float8_e4m3fn)
    G = ops.fcma_gram_fp8(Z8.cuda().contiguous())
    Zq = Z8.float()                      # same quantized values
    ref = torch.bmm(Zq, Zq.transpose(1, 2))
    assert torch.allclose(G.cpu(), ref, atol=0.5, rtol=1e-3)
    # and against the unquantized oracle: ~3% of the row-norm scale
    full = torch.bmm(Z, Z.transpose(1, 2))
    scale = full.diagonal(dim1=1, dim2=2).mean()
    assert (G.cpu() - full).abs().max() < 0.12 * scale


def test_gram_fp8_vsplit_path(ops):
    """V large enough to trigger the nsplit partial-sum path."""
    g = torch.Generator().manual_seed(15)
    Z = torch.randn((1, 64, 16384), generator=g).float().clamp_(-2, 2)
    Z8 = Z.to(torch.float8_e4m3fn)
    G = ops.fcma_gram_fp8(Z8.cuda().contiguous())
    Zq = Z8.float()
    ref = torch.bmm(Zq, Zq.transpose(1, 2))
    assert torch.allclose(G.cpu(), ref, atol=2.0, rtol=1e-3)


def test_corr_norm_z_fp8_output(ops):
    """fp8 Z output of the fused corr+normalize matches the bf16-path
    values to e4m3 quantization."""
    g = torch.Generator().manual_seed(16)
    E, L, V, P = 16, 12, 256, 4
    # cross-correlation (A != B): a self-correlation's r=1 diagonal
    # clamps in fisher_z and its degenerate z-score columns are
    # rounding-order sensitive ACROSS template instantiations
    A = _zscored_epochs(g, E, L, V, "cpu").to(torch.bfloat16)
    B = _zscored_epochs(g, E, L, V, "cpu").to(torch.bfloat16)
    Acu = A.cuda().contiguous()
    Bcu = B.cuda().contiguous()
    ext = ops.load_extension()
    Epad = 64
    z_bf = torch.zeros((32, Epad, V), dtype=torch.bfloat16,
                       device="cuda")
    z_f8 = torch.zeros((32, Epad, V), dtype=torch.float8_e4m3fn,
                       device="cuda")
    ext.fcma_corr_norm_z(Acu, Bcu, 5, 32, P, Epad, out=z_bf)
    ext.fcma_corr_norm_z(Acu, Bcu, 5, 32, P, Epad, out=z_f8)
    a = z_bf.float().cpu()
    b = z_f8.float().cpu()
    # z-scores bounded by sqrt(P-1); e4m3 relative step is 2^-3
    assert (a - b).abs().max() < 0.15
    assert torch.allclose(a, b, atol=0.15, rtol=0.13)


def test_fp8_pipeline_end_to_end(ops):
    """Full chunk pipeline with z_fp8=True: Gram close to the fp32
    chain, CV selection rankings close to the bf16 default."""
    import numpy as np

    from brainiak_amd.fcma.core import CorrelationPipeline
    from brainiak_amd.fcma.svm import cross_validate_voxels
    rng = np.random.RandomState(21)
    E, L, V = 32, 12, 500    # V deliberately not a multiple of 16
    labels = np.array([e % 2 for e in range(E)])
    raw = []
    for e in range(E):
        m = rng.randn(L, V).astype(np.float32)
        if e % 2:
            m[:, :V // 2] += 0.8 * rng.randn(L, 1)
        m = (m - m.mean(0)) / m.std(0)
        raw.append((m / np.sqrt(L)).astype(np.float32))
    accs = {}
    for tag, fp8 in (("bf16", False), ("fp8", True)):
        pipe = CorrelationPipeline(raw, None, 4, device="cuda",
                                   z_fp8=fp8)
        kernels = pipe.pipelined_kernel_matrices([(0, 256), (256, V - 256)])
        assert kernels.shape[0] == V
        accs[tag] = np.asarray(cross_validate_voxels(kernels, labels, 4))
    from scipy.stats import spearmanr
    r = spearmanr(accs["bf16"], accs["fp8"]).statistic
    assert r > 0.85, r
    k = V // 4
    top_b = set(np.argsort(accs["bf16"])[-k:])
    top_f = set(np.argsort(accs["fp8"])[-k:])
    assert len(top_b & top_f) >= 0.75 * k


def test_gram_deferred_normalize_matches_fused(ops):
    """Raw-r corr + in-tile normalize gram == fused corr+norm + plain
    gram (same bf16 Z quantization points differ: raw path quantizes r,
    fused path quantizes z — compare against the r-quantized oracle)."""
    import numpy as np
    g = torch.Generator().manual_seed(23)
    E, L, V, P = 16, 12, 512, 4
    A = _zscored_epochs(g, E, L, V, "cpu").to(torch.bfloat16)
    B = _zscored_epochs(g, E, L, V, "cpu").to(torch.bfloat16)
    Acu, Bcu = A.cuda().contiguous(), B.cuda().contiguous()
    ext = ops.load_extension()
    Epad = 64
    z_raw = torch.zeros((32, Epad, V), dtype=torch.bfloat16,
                        device="cuda")
    ext.fcma_corr_norm_z(Acu, Bcu, 5, 32, P, Epad, out=z_raw, raw=True)
    G = ops.fcma_gram_bf16(z_raw, norm_P=P)[:, :E, :E]

    # oracle: same r-quantization (bf16), fp32 normalize + gram
    corr = torch.einsum('elc,elv->cev', A.float()[:, :, 5:37], B.float())
    rq = corr.to(torch.bfloat16).float()          # raw-r bf16 rounding
    nc = _ref_normalize(rq, P)
    ref = torch.bmm(nc, nc.transpose(1, 2))
    assert torch.allclose(G.cpu(), ref, atol=1.5, rtol=5e-2)


def test_pipeline_raw_split_matches_no_raw(ops):
    """End-to-end: default (raw-split) pipeline grams vs the
    BRAINIAK_NO_RAWCORR fused path — close at bf16 tolerance, and CV
    rankings effectively identical."""
    import os

    import numpy as np

    from brainiak_amd.fcma.core import CorrelationPipeline
    from brainiak_amd.fcma.svm import cross_validate_voxels
    rng = np.random.RandomState(31)
    E, L, V = 32, 12, 512
    labels = np.array([e % 2 for e in range(E)])
    raw = []
    for e in range(E):
        m = rng.randn(L, V).astype(np.float32)
        if e % 2:
            m[:, :V // 2] += 0.8 * rng.randn(L, 1)
        m = (m - m.mean(0)) / m.std(0)
        raw.append((m / np.sqrt(L)).astype(np.float32))
    pipe = CorrelationPipeline(raw, None, 4, device="cuda")
    assert pipe._raw_split
    g_split = pipe.pipelined_kernel_matrices([(0, 256), (256, 256)])
    os.environ["BRAINIAK_NO_RAWCORR"] = "1"
    try:
        pipe2 = CorrelationPipeline(raw, None, 4, device="cuda")
        assert not pipe2._raw_split
        g_fused = pipe2.pipelined_kernel_matrices([(0, 256), (256, 256)])
    finally:
        del os.environ["BRAINIAK_NO_RAWCORR"]
    a = np.asarray(cross_validate_voxels(g_split, labels, 4))
    b = np.asarray(cross_validate_voxels(g_fused, labels, 4))
    from scipy.stats import spearmanr
    assert spearmanr(a, b).statistic > 0.95
    assert torch.allclose(g_split.float(), g_fused.float(),
                          atol=2.0, rtol=5e-2)


def test_duo_pipeline_matches_streamed(ops):
    """Duo-kernel pipeline == two-stream raw-split pipeline (identical
    kernels, different scheduling)."""
    import os

    import numpy as np

    from brainiak_amd.fcma.core import CorrelationPipeline
    rng = np.random.RandomState(41)
    E, L, V = 32, 12, 520
    raw = []
    for _ in range(E):
        m = rng.randn(L, V).astype(np.float32)
        m = (m - m.mean(0)) / m.std(0)
        raw.append((m / np.sqrt(L)).astype(np.float32))
    chunks = [(0, 200), (200, 200), (400, V - 400)]
    pipe = CorrelationPipeline(raw, None, 4, device="cuda")
    g_duo = pipe.pipelined_kernel_matrices(chunks)
    os.environ["BRAINIAK_NO_DUO"] = "1"
    try:
        pipe2 = CorrelationPipeline(raw, None, 4, device="cuda")
        g_str = pipe2.pipelined_kernel_matrices(chunks)
    finally:
        del os.environ["BRAINIAK_NO_DUO"]
    assert g_duo.shape == g_str.shape == (V, E, E)
    assert torch.allclose(g_duo.float(), g_str.float(), atol=1e-3,
                          rtol=1e-4)


def test_overlapped_cv_matches_post_pass(ops):
    """VoxelSelector's side-stream per-chunk CV == the single batched
    post-pass CV (same kernels, same SMO; only the scheduling and the
    per-chunk shrink ordering differ)."""
    import os

    import numpy as np

    from brainiak_amd.fcma.voxelselector import VoxelSelector
    rng = np.random.RandomState(17)
    E, L, V, P = 32, 12, 640, 4
    raw = []
    for _ in range(E):
        m = rng.randn(L, V).astype(np.float32)
        m = (m - m.mean(0)) / m.std(0)
        raw.append((m / np.sqrt(L)).astype(np.float32))
    labels = np.tile([0, 1], E // 2)
    sel = VoxelSelector(labels, P, 4, raw, voxel_unit=256,
                        device="cuda")
    from sklearn import svm
    clf = svm.SVC(kernel='precomputed', shrinking=False, C=1.0)
    res_overlap = sel.run(clf)
    os.environ["BRAINIAK_NO_CV_OVERLAP"] = "1"
    try:
        res_post = VoxelSelector(labels, P, 4, raw, voxel_unit=256,
                                 device="cuda").run(clf)
    finally:
        del os.environ["BRAINIAK_NO_CV_OVERLAP"]
    assert len(res_overlap) == len(res_post) == V
    a = dict(res_overlap)
    b = dict(res_post)
    assert all(abs(a[v] - b[v]) < 1e-6 for v in a)


def test_duo_gsum_fuzz_shapes(ops):
    """Randomized duo-pipeline sweep: odd voxel counts, P in {2,4},
    several epoch lengths, ragged chunk patterns, E both at and below
    the 64-row tile (exercises the DZ padding path and the in-grid
    partial-sum population) — each case vs the streamed two-kernel
    pipeline."""
    import os

    import numpy as np

    from brainiak_amd.fcma.core import CorrelationPipeline
    rng = np.random.RandomState(7)
    cases = [
        (32, 12, 700, 4, [(0, 256), (256, 256), (512, 188)]),
        (64, 8, 530, 2, [(0, 200), (200, 200), (400, 130)]),
        (48, 20, 464, 4, [(0, 128), (128, 128), (256, 128),
                          (384, 80)]),
        (64, 40, 333, 4, [(0, 333)]),
    ]
    for E, L, V, P, chunks in cases:
        raw = []
        for _ in range(E):
            m = rng.randn(L, V).astype(np.float32)
            m = (m - m.mean(0)) / np.maximum(m.std(0), 1e-6)
            raw.append((m / np.sqrt(L)).astype(np.float32))
        pipe = CorrelationPipeline(raw, None, P, device="cuda")
        g_duo = pipe.pipelined_kernel_matrices(chunks)
        os.environ["BRAINIAK_NO_DUO"] = "1"
        try:
            pipe2 = CorrelationPipeline(raw, None, P, device="cuda")
            g_str = pipe2.pipelined_kernel_matrices(chunks)
        finally:
            del os.environ["BRAINIAK_NO_DUO"]
        assert g_duo.shape == g_str.shape == (V, E, E), (E, L, V, P)
        assert torch.allclose(g_duo.float(), g_str.float(),
                              atol=2e-3, rtol=1e-3), (E, L, V, P)


def test_isfc_fused_matches_gemm_accum(ops):
    """Fused tile-pair ISFC kernel == bf16 GEMM + k_isfc_accum on the
    same inputs (both bf16 operands, fp32 accumulation)."""
    import numpy as np

    from brainiak_amd import ops as _ops
    torch.manual_seed(0)
    dev = "cuda"
    B, V, T = 3, 333, 50          # odd V exercises the tail guards
    Zs = torch.randn(B, V, T, device=dev)
    Zs = Zs / Zs.norm(dim=2, keepdim=True)
    Zm = torch.randn(B, V, T, device=dev)
    Zm = Zm / Zm.norm(dim=2, keepdim=True)
    Zs16, Zm16 = Zs.to(torch.bfloat16), Zm.to(torch.bfloat16)

    acc_f = torch.zeros(V, V, device=dev)
    _ops.isfc_fused_(acc_f, Zs16.contiguous(), Zm16.contiguous())

    acc_r = torch.zeros(V, V, device=dev)
    for b in range(B):
        m = (Zs16[b] @ Zm16[b].T).float()
        m = (m + m.T) / 2
        acc_r += torch.atanh(m.clamp(-1 + 1e-7, 1 - 1e-7))
    assert torch.allclose(acc_f, acc_r, atol=5e-3, rtol=1e-3), \
        (acc_f - acc_r).abs().max()


def test_isfc_distributed_fused_matches_gemm_path(ops):
    """isfc_distributed bf16: fused kernel vs the GEMM+accum pipeline."""
    import os

    import numpy as np

    from brainiak_amd.isc import isfc_distributed
    from brainiak_amd.parallel import DistContext
    rng = np.random.RandomState(3)
    data = [rng.randn(40, 300).astype(np.float32) for _ in range(5)]
    ctx = DistContext(device="cuda")
    os.environ["BRAINIAK_ISFC_FUSED"] = "1"
    try:
        fused = isfc_distributed(data, ctx, summary_statistic='mean',
                                 precision='bf16', return_tensor=True)
    finally:
        del os.environ["BRAINIAK_ISFC_FUSED"]
    ref = isfc_distributed(data, ctx, summary_statistic='mean',
                           precision='bf16', return_tensor=True)
    assert torch.allclose(fused, ref, atol=2e-3), \
        (fused - ref).abs().max()
