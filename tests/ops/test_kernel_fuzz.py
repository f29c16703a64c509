"""Deterministic random-shape sweeps of every production kernel
against its torch oracle (seeded; the shapes cover tails, non-multiples
and both dtypes)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ops():
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    from brainiak_amd import ops as _ops
    if _ops.load_extension() is None:
        pytest.fail("HIP extension not built")
    return _ops


def test_stencil3d_shape_sweep(ops):
    rng = np.random.RandomState(2)
    for t in range(12):
        r = int(rng.randint(1, 5))
        K = 2 * r + 1
        B = int(rng.randint(1, 5))
        X, Y, Z = [int(rng.randint(2 * r + 2, 30)) for _ in range(3)]
        g = torch.Generator().manual_seed(t)
        x = torch.randn((B, X, Y, Z), generator=g).cuda().contiguous()
        w = torch.randn((K, K, K), generator=g).cuda().contiguous()
        got = ops.stencil3d(x, w)
        ref = torch.nn.functional.conv3d(x[:, None], w[None, None])[:, 0]
        assert torch.allclose(got, ref, atol=2e-3, rtol=1e-3), (t, r)


def test_isfc_accum_shape_sweep(ops):
    rng = np.random.RandomState(3)
    for t in range(10):
        V = int(rng.randint(3, 500))
        dt = torch.bfloat16 if t % 2 else torch.float32
        g = torch.Generator().manual_seed(100 + t)
        M = (torch.randn((V, V), generator=g) * 0.5).to(dt)
        M = M.cuda().contiguous()
        acc = torch.randn((V, V), generator=g).cuda().contiguous()
        got = ops.isfc_accum_(acc.clone(), M)
        Mf = M.float()
        sym = (Mf + Mf.T) / 2
        ref = acc + torch.atanh(sym.clamp(-1 + 1e-7, 1 - 1e-7))
        assert torch.allclose(got, ref, atol=2e-4, rtol=1e-4), (t, V)


def test_gram_bf16_shape_sweep(ops):
    rng = np.random.RandomState(4)
    for t in range(8):
        E = 64 * int(rng.randint(1, 3))
        V = int(rng.randint(3, 900))
        C = int(rng.randint(1, 7))
        g = torch.Generator().manual_seed(200 + t)
        Zt = torch.randn((C, E, V), generator=g).to(torch.bfloat16)
        Zt = Zt.cuda().contiguous()
        G = ops.fcma_gram_bf16(Zt)
        Zf = Zt.float()
        ref = torch.bmm(Zf, Zf.transpose(1, 2))
        assert torch.allclose(G.cpu(), ref.cpu(), atol=0.6, rtol=2e-2), \
            (t, E, V, C)


def test_batched_polar_shape_sweep(ops):
    rng = np.random.RandomState(5)
    for t in range(8):
        K = int(rng.randint(2, 33))
        B = int(rng.randint(1, 5))
        V = int(rng.randint(K, 200))
        g = torch.Generator().manual_seed(300 + t)
        A = torch.randn((B, V, K), generator=g).cuda()
        W = ops.batched_polar(A, 0.0)
        WtW = torch.bmm(W.transpose(1, 2), W).cpu()
        assert torch.allclose(WtW, torch.eye(K).expand(B, K, K),
                              atol=2e-2), (t, K)
