"""Property-based aggregator tests (hypothesis).

Robust-aggregation invariants from the literature: permutation invariance
(the aggregate cannot depend on client ordering), translation equivariance
for mean/median/trimmed/geomed (agg(U + c) = agg(U) + c), and scale
equivariance.  These catch index-bookkeeping bugs no fixed-seed test hits.
"""
import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

from blades_amd.aggregators import (Geomed, Krum, Mean, Median, Trimmedmean)


def _U(K, d, seed):
    g = torch.Generator().manual_seed(seed)
    return torch.randn(K, d, generator=g)


@settings(max_examples=20, deadline=None)
@given(K=st.integers(5, 24), d=st.integers(1, 40), seed=st.integers(0, 10**6))
def test_permutation_invariance(K, d, seed):
    U = _U(K, d, seed)
    perm = torch.randperm(K, generator=torch.Generator().manual_seed(seed + 1))
    for agg in (Mean(), Median(), Trimmedmean(nb=max(1, (K - 1) // 3)),
                Geomed(maxiter=50)):
        a = agg(U.clone())
        b = agg(U[perm].clone())
        assert torch.allclose(a, b, atol=1e-4), type(agg).__name__


@settings(max_examples=20, deadline=None)
@given(K=st.integers(5, 24), d=st.integers(1, 40), seed=st.integers(0, 10**6),
       shift=st.floats(-5, 5), scale=st.floats(0.1, 4))
def test_translation_and_scale_equivariance(K, d, seed, shift, scale):
    U = _U(K, d, seed)
    c = torch.full((d,), float(shift))
    for agg_fn in (Mean, Median, lambda: Trimmedmean(nb=max(1, (K - 1) // 3))):
        agg = agg_fn()
        base = agg(U.clone())
        shifted = agg((U + c).clone())
        assert torch.allclose(shifted, base + c, atol=1e-4), agg_fn
        scaled = agg((U * scale).clone())
        assert torch.allclose(scaled, base * scale, atol=1e-3), agg_fn


@settings(max_examples=15, deadline=None)
@given(K=st.integers(8, 20), d=st.integers(2, 30), seed=st.integers(0, 10**6))
def test_krum_selects_a_row(K, d, seed):
    """Krum's output must be one of the input rows (m=1)."""
    U = _U(K, d, seed)
    f = max(1, (K - 3) // 2 - 1)
    out = Krum(num_clients=K, num_byzantine=f)(U)
    dists = (U - out.unsqueeze(0)).norm(dim=1)
    assert dists.min() < 1e-5


@settings(max_examples=15, deadline=None)
@given(K=st.integers(5, 20), d=st.integers(1, 30), seed=st.integers(0, 10**6))
def test_aggregate_within_coordinate_hull(K, d, seed):
    """Coordinate-wise robust aggregates lie within the per-coordinate
    min/max envelope of the inputs."""
    U = _U(K, d, seed)
    lo, hi = U.min(0).values, U.max(0).values
    for agg in (Mean(), Median(), Trimmedmean(nb=max(1, (K - 1) // 3))):
        out = agg(U.clone())
        assert (out >= lo - 1e-5).all() and (out <= hi + 1e-5).all()


@settings(max_examples=10, deadline=None)
@given(K=st.integers(6, 16), d=st.integers(2, 20), seed=st.integers(0, 10**6))
def test_single_outlier_bounded_influence(K, d, seed):
    """Moving one client arbitrarily far must not move Median/TrimmedMean
    outside the envelope of the remaining honest clients by more than the
    honest spread (breakdown-point sanity)."""
    U = _U(K, d, seed)
    honest = U[1:]
    lo, hi = honest.min(0).values, honest.max(0).values
    U_attacked = U.clone()
    U_attacked[0] = 1e8
    for agg in (Median(), Trimmedmean(nb=max(1, K // 4))):
        out = agg(U_attacked.clone())
        assert (out >= lo - 1e-4).all() and (out <= hi + 1e-4).all(), \
            type(agg).__name__


def test_col_trimmed_sum_cpu_fallback():
    import torch

    from blades_amd import ops
    U = torch.randn(40, 100)
    out = ops.col_trimmed_sum(U, 3, 7)
    s = U.double().sum(0)
    s -= torch.topk(U.double(), 3, dim=0, largest=False).values.sum(0)
    s -= torch.topk(U.double(), 7, dim=0, largest=True).values.sum(0)
    assert torch.allclose(out, s.float(), atol=1e-5)


def test_krum_scores_cpu_matches_ref():
    import torch

    from blades_amd import ops
    from blades_amd.ops import torch_ref
    U = torch.randn(30, 64)
    D = torch_ref.pairwise_sq_dists(U)
    assert torch.allclose(ops.krum_scores(D, 5), torch_ref.krum_scores(D, 5),
                          atol=1e-4)
