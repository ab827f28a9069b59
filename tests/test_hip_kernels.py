"""GPU numerics: every HIP kernel vs the plain PyTorch fp32 reference
(torch_ref.py) on-device.  All marked @gpu (run on MI355X via gpurun)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ext():
    from blades_amd import ops

    assert ops.hip_available(), "HIP extension must be built"
    return ops


def randU(K, d, seed=0, pad=False):
    g = torch.Generator(device="cuda")
    g.manual_seed(seed)
    if pad:
        d_pad = (d + 3) // 4 * 4 + 4
        buf = torch.randn(K, d_pad, generator=g, device="cuda")
        return buf[:, :d]
    return torch.randn(K, d, generator=g, device="cuda")


SHAPES = [(9, 1000), (100, 59850), (33, 12345), (100, 262144)]


@pytest.mark.parametrize("K,d", SHAPES)
def test_col_mean(ext, K, d):
    from blades_amd.ops import torch_ref
    U = randU(K, d)
    out = ext.col_mean(U)
    ref = torch_ref.col_mean(U)
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("pad", [False, True])
def test_col_mean_strided(ext, pad):
    from blades_amd.ops import torch_ref
    U = randU(17, 1001, pad=pad)
    assert torch.allclose(ext.col_mean(U), torch_ref.col_mean(U),
                          atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("K,d", SHAPES)
def test_weighted_col_sum(ext, K, d):
    from blades_amd.ops import torch_ref
    U = randU(K, d)
    w = torch.rand(K, device="cuda")
    assert torch.allclose(ext.weighted_col_sum(U, w),
                          torch_ref.weighted_col_sum(U, w),
                          atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("K,d", SHAPES)
def test_masked_col_mean_std(ext, K, d):
    from blades_amd.ops import torch_ref
    U = randU(K, d)
    mask = torch.rand(K, device="cuda") > 0.3
    mask[0] = mask[1] = True
    mu, sd = ext.masked_col_mean_std(U, mask, True)
    rmu, rsd = torch_ref.masked_col_mean_std(U, mask, True)
    assert torch.allclose(mu, rmu, atol=1e-5, rtol=1e-5)
    assert torch.allclose(sd, rsd, atol=1e-4, rtol=1e-3)
    m2 = ext.masked_col_mean(U, mask)
    assert torch.allclose(m2, rmu, atol=1e-5, rtol=1e-5)


@pytest.mark.parametrize("K,b", [(9, 2), (100, 20), (100, 49), (64, 31),
                                 (7, 0), (20, 8), (70, 32), (33, 16),
                                 (1000, 20), (12, 5), (1000, 499),
                                 (1280, 500), (1300, 500)])
def test_trimmed_mean(ext, K, b):
    from blades_amd.ops import torch_ref
    U = randU(K, 50000, seed=b)
    out = ext.trimmed_mean(U, b)
    ref = torch_ref.trimmed_mean(U, b) if b > 0 else U.mean(0)
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-5)


def test_trimmed_mean_with_duplicates(ext):
    from blades_amd.ops import torch_ref
    U = randU(20, 1024).round()  # many ties
    assert torch.allclose(ext.trimmed_mean(U, 5),
                          torch_ref.trimmed_mean(U, 5), atol=1e-5)


@pytest.mark.parametrize("K", [5, 8, 100, 101])
def test_col_median(ext, K):
    from blades_amd.ops import torch_ref
    U = randU(K, 30000, seed=K)
    assert torch.allclose(ext.col_median(U), torch_ref.col_median(U),
                          atol=1e-6)


@pytest.mark.parametrize("K,d", SHAPES)
def test_row_ops(ext, K, d):
    from blades_amd.ops import torch_ref
    U = randU(K, d)
    z = torch.randn(d, device="cuda")
    assert torch.allclose(ext.row_sq_norms(U), torch_ref.row_sq_norms(U),
                          rtol=1e-4, atol=1e-3)
    assert torch.allclose(ext.row_diff_norms(U, z),
                          torch_ref.row_diff_norms(U, z), rtol=1e-4, atol=1e-3)
    assert torch.allclose(ext.row_dots(U, z), torch_ref.row_dots(U, z),
                          rtol=1e-3, atol=1e-2)


@pytest.mark.parametrize("K,d", [(7, 512), (32, 4096), (100, 59850),
                                 (100, 262144), (130, 10000)])
def test_gram_mfma(ext, K, d):
    """MFMA f32 Gram vs torch matmul — exact-f32 class numerics.
    Asymmetric data (not identity) per the transpose-detection rule."""
    U = randU(K, d, seed=d)
    G = ext.gram(U)
    ref = U @ U.t()
    assert torch.allclose(G, ref, rtol=1e-4, atol=1e-2)
    # symmetric by construction
    assert torch.allclose(G, G.t(), rtol=1e-5, atol=1e-3)


def test_pairwise_sq_dists_vs_ref(ext):
    from blades_amd.ops import torch_ref
    U = randU(50, 20000)
    D = ext.pairwise_sq_dists(U)
    ref = torch_ref.pairwise_sq_dists(U)
    assert torch.allclose(D, ref, rtol=1e-3, atol=1e-1)
    assert (D.diagonal() == 0).all()


@pytest.mark.parametrize("K,f", [(25, 5), (100, 20), (1000, 10),
                                 (1000, 200), (2000, 660)])
def test_krum_scores_kernel(ext, K, f):
    """K5 — the selection-kernel scores must match the host topk reference
    on the symmetric zero-diagonal distance matrix (small f -> register
    kernel, large f -> asymmetric radix path)."""
    from blades_amd import ops
    from blades_amd.ops import torch_ref
    U = randU(K, 256, seed=K + f)
    D = ops.pairwise_sq_dists(U)
    s_gpu = ops.krum_scores(D, f)
    s_ref = torch_ref.krum_scores(D.double(), f).float()
    assert torch.allclose(s_gpu, s_ref, rtol=1e-5, atol=1e-3), \
        (s_gpu - s_ref).abs().max().item()


@pytest.mark.parametrize("K,blo,bhi", [(50, 0, 21), (50, 3, 0), (100, 5, 40),
                                       (300, 0, 120)])
def test_col_trimmed_sum_asymmetric(ext, K, blo, bhi):
    from blades_amd import ops
    U = randU(K, 30000, seed=K + blo + bhi)
    out = ops.col_trimmed_sum(U, blo, bhi)
    s = U.double().sum(0)
    if blo:
        s -= torch.topk(U.double(), blo, dim=0, largest=False).values.sum(0)
    if bhi:
        s -= torch.topk(U.double(), bhi, dim=0, largest=True).values.sum(0)
    assert torch.allclose(out, s.float(), rtol=1e-5, atol=1e-4)


def test_centered_clip_iter(ext):
    from blades_amd.ops import torch_ref
    U = randU(16, 40000)
    v = torch.randn(40000, device="cuda")
    out = ext.centered_clip_iter(U, v, 2.0)
    ref = torch_ref.centered_clip_iter(U, v, 2.0)
    assert torch.allclose(out, ref, rtol=1e-4, atol=1e-4)


def test_aggregators_gpu_match_cpu():
    """End aggregators on GPU (HIP path) vs CPU (torch path)."""
    from blades_amd.aggregators import (Geomed, Krum, Mean, Median,
                                        Trimmedmean)

    U = randU(40, 100000, seed=3)
    Ucpu = U.cpu()
    for agg_fn in [Mean, Median, lambda: Trimmedmean(nb=10),
                   lambda: Krum(num_clients=40, num_byzantine=10), Geomed]:
        gpu_out = agg_fn()(U).cpu()
        cpu_out = agg_fn()(Ucpu)
        assert torch.allclose(gpu_out, cpu_out, rtol=1e-3, atol=1e-4), agg_fn


def test_fail_loudly_without_force(ext, monkeypatch):
    """GPU tensors must never silently fall back to eager torch."""
    import blades_amd.ops as ops

    monkeypatch.setattr(ops, "_EXT", None)
    monkeypatch.setattr(ops, "_EXT_ERR", "simulated missing ext")
    U = torch.randn(4, 64, device="cuda")
    with pytest.raises(RuntimeError, match="refusing to fall back"):
        ops.col_mean(U)
    monkeypatch.setattr(ops, "_EXT_ERR", None)


@pytest.mark.parametrize("K,b", [(1000, 499), (1000, 300), (2000, 999),
                                 (517, 258), (4096, 1000)])
def test_radix_trimmed_large_K(ext, K, b):
    """Dual radix-select path (large b beyond the LDS kernel's range).

    Reference computed in fp64: the torch fp32 sum-minus-topk form loses
    ~1e-5 to cancellation when only 1-2 of K values survive the trim,
    while the kernel accumulates in fp64."""
    from blades_amd.ops import torch_ref
    U = randU(K, 20000, seed=K + b)
    out = ext.trimmed_mean(U, b)
    ref = torch_ref.trimmed_mean(U.double(), b).float()
    assert torch.allclose(out, ref, atol=1e-6, rtol=1e-5)


def test_radix_trimmed_with_ties(ext):
    from blades_amd.ops import torch_ref
    U = randU(999, 4096, seed=1).round()  # heavy ties
    out = ext.trimmed_mean(U, 400)
    ref = torch_ref.trimmed_mean(U, 400)
    assert torch.allclose(out, ref, atol=1e-5)


def test_col_median_large_K(ext):
    from blades_amd.ops import torch_ref
    for K in (999, 1000):
        U = randU(K, 30000, seed=K)
        ref = torch_ref.col_median(U.double()).float()
        assert torch.allclose(ext.col_median(U), ref, atol=1e-6), K


def test_trimmed_mean_huge_outlier(ext):
    """Byzantine-magnitude outliers (1e8) must not corrupt the trimmed mean
    (fp32 cancellation regression — the reference's formula fails this)."""
    from blades_amd.ops import torch_ref
    U = randU(20, 8192, seed=42)
    U[0] = 1e8
    U[1] = -1e8
    out = ext.trimmed_mean(U, 5)
    ref = torch_ref.trimmed_mean(U.double(), 5).float()
    assert torch.allclose(out, ref, atol=1e-5, rtol=1e-5)
    lo = U[2:].min(0).values
    hi = U[2:].max(0).values
    assert (out >= lo - 1e-4).all() and (out <= hi + 1e-4).all()
