"""Pure-PyTorch reference implementations of every aggregation/attack op.

These are (a) the CPU execution path, and (b) the fp32 numerics reference the
HIP kernels are tested against (SURVEY.md §2.4 K1-K18 inventory).  Each
function documents the reference site whose math it reproduces.

All ops take the stacked update matrix ``U ∈ R^{K×d}`` (fp32, K clients,
d model dims) unless noted.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

Tensor = torch.Tensor


# ---------------------------------------------------------------- column ops

def col_mean(U: Tensor) -> Tensor:
    """K1 — column mean (reference: aggregators/mean.py:72)."""
    return U.mean(dim=0)


def col_median(U: Tensor) -> Tensor:
    """K2 — coordinate-wise median, averaging the two middle elements for
    even K (reference: aggregators/median.py:23-25 computes
    ``(median(U) - median(-U)) / 2`` to the same effect)."""
    values_upper, _ = U.median(dim=0)
    values_lower, _ = (-U).median(dim=0)
    return (values_upper - values_lower) / 2


def trimmed_mean(U: Tensor, b: int) -> Tensor:
    """K3 — per coordinate: drop the b largest and b smallest, mean the rest
    (reference: aggregators/trimmedmean.py:38-41).

    Accumulated in fp64: the reference's fp32 ``sum − topk + neg_topk`` form
    cancels catastrophically when Byzantine rows contain huge values (a 1e8
    outlier swallows every honest value's bits before being subtracted —
    caught by tests/test_properties.py::test_single_outlier_bounded_influence;
    the HIP kernel sums in fp64 for the same reason).
    """
    K = U.shape[0]
    if K - 2 * b <= 0:
        raise ValueError(f"trimmed_mean needs K > 2b (K={K}, b={b})")
    if b == 0:
        return U.mean(dim=0)
    Ud = U.double()
    largest, _ = torch.topk(Ud, b, 0)
    neg_smallest, _ = torch.topk(-Ud, b, 0)
    out = (Ud.sum(0) - largest.sum(0) + neg_smallest.sum(0)) / (K - 2 * b)
    return out.to(U.dtype)


def col_var(U: Tensor, unbiased: bool = False) -> Tensor:
    """K17 — column variance diagnostics (reference: simulator.py:309-313)."""
    return U.var(dim=0, unbiased=unbiased)


def weighted_col_sum(U: Tensor, w: Tensor) -> Tensor:
    """Σ_k w_k · U_k — the GeoMed/FLTrust accumulation primitive."""
    return (w.to(U.dtype).unsqueeze(1) * U).sum(dim=0)


def masked_col_mean(U: Tensor, mask: Tensor) -> Tensor:
    """K11 — column mean over rows where ``mask`` is True (IPM honest mean,
    reference: attackers/ipmclient.py:10-16)."""
    m = mask.to(U.dtype)
    return (U * m.unsqueeze(1)).sum(dim=0) / m.sum().clamp_min(1)


def masked_col_mean_std(U: Tensor, mask: Tensor,
                        unbiased: bool = True) -> Tuple[Tensor, Tensor]:
    """K10 — column mean and std over masked (honest) rows
    (reference: attackers/alieclient.py:32-35 uses torch.mean/torch.std,
    i.e. Bessel-corrected std)."""
    rows = U[mask]
    mu = rows.mean(dim=0)
    std = rows.std(dim=0, unbiased=unbiased)
    return mu, std


# ------------------------------------------------------------------- row ops

def row_sq_norms(U: Tensor) -> Tensor:
    """‖u_i‖² per row."""
    return (U * U).sum(dim=1)


def row_norms(U: Tensor) -> Tensor:
    return row_sq_norms(U).sqrt()


def row_diff_norms(U: Tensor, z: Tensor) -> Tensor:
    """K6 — ‖z − u_i‖ for every row (Weiszfeld distance pass,
    reference: aggregators/geomed.py:71-75)."""
    return (U - z.unsqueeze(0)).norm(dim=1)


def row_dots(U: Tensor, v: Tensor) -> Tensor:
    """u_i · v per row (FLTrust cosine numerator)."""
    return U @ v


def pairwise_sq_dists(U: Tensor) -> Tensor:
    """K4 — K×K squared-distance matrix.

    The reference loops K(K-1)/2 norm calls (aggregators/krum.py:73-90);
    here ‖u_i−u_j‖² = ‖u_i‖² + ‖u_j‖² − 2·u_i·u_j via one Gram matmul (the
    HIP path runs this on MFMA).  Clamped at 0 against cancellation.
    """
    G = U @ U.t()
    sq = G.diagonal()
    D = sq.unsqueeze(0) + sq.unsqueeze(1) - 2 * G
    D.fill_diagonal_(0)
    return D.clamp_min_(0)


def cos_sim_gram(U: Tensor, eps: float = 1e-8) -> Tensor:
    """K8 — K×K cosine-similarity matrix (reference: aggregators/
    clustering.py:28-33 computes it with a scipy double loop)."""
    norms = row_norms(U).clamp_min(eps)
    Un = U / norms.unsqueeze(1)
    G = Un @ Un.t()
    return G.clamp_(-1.0, 1.0)


# ----------------------------------------------------------- fused/iterated

def krum_scores(sqdists: Tensor, f: int) -> Tensor:
    """K5 — per-row sum of the n−f−2 smallest squared distances to OTHER
    rows (reference: aggregators/krum.py:9-25)."""
    n = sqdists.shape[0]
    k = n - f - 2
    if k < 1:
        raise ValueError(f"krum needs n - f - 2 >= 1 (n={n}, f={f})")
    # mask the diagonal out by setting it to +inf, then take k smallest
    D = sqdists.clone()
    D.fill_diagonal_(float("inf"))
    smallest, _ = torch.topk(D, k, dim=1, largest=False)
    return smallest.sum(dim=1)


def clip_to_norm(v: Tensor, tau: float) -> Tensor:
    """Scale v to norm ≤ tau (reference: centeredclipping.py:30-33)."""
    v_norm = v.norm()
    scale = torch.clamp(tau / v_norm.clamp_min(1e-12), max=1.0)
    return v * scale


def centered_clip_iter(U: Tensor, v: Tensor, tau: float) -> Tensor:
    """K7 — one centered-clipping iteration:
    v' = v + mean_k(clip(u_k − v, tau)) (reference: centeredclipping.py:40-44).
    """
    diff = U - v.unsqueeze(0)
    norms = diff.norm(dim=1).clamp_min(1e-12)
    scale = torch.clamp(tau / norms, max=1.0)
    return v + (diff * scale.unsqueeze(1)).mean(dim=0)


def row_clip_to_norm_(U: Tensor, max_norms: Tensor) -> Tensor:
    """Clip each row of U to its per-row threshold in place
    (Clippedclustering's historical-median clip, reference:
    clippedclustering.py:37-46 via torch_utils.clip_tensor_norm_)."""
    norms = U.norm(dim=1).clamp_min(1e-12)
    scale = torch.clamp(max_norms / norms, max=1.0)
    U.mul_(scale.unsqueeze(1))
    return U


# ----------------------------------------------------------------- RNG fill

def philox_normal(shape, mean: float, std: float, seed: int,
                  device=None, dtype=torch.float32) -> Tensor:
    """K12 — deterministic normal fill from a dedicated generator stream.

    Counter-based keying happens in the caller via
    :func:`blades_amd.utils.client_philox_seed`; this draws from a private
    torch.Generator so the global RNG stream is untouched (the reference
    instead relied on driver-side RNG save/restore, simulator.py:153-165).
    """
    dev = torch.device(device) if device is not None else torch.device("cpu")
    g = torch.Generator(device=dev)
    g.manual_seed(seed)
    return torch.empty(shape, device=dev, dtype=dtype).normal_(mean, std, generator=g)


# ----------------------------------------------------------- server-side ops

def flat_sgd_step_(theta: Tensor, delta: Tensor, lr: float) -> Tensor:
    """K16 — fused flat-vector pseudo-gradient SGD: θ ← θ + lr·Δ.

    Matches the reference server semantics: ``p.grad = -update_slice`` then
    ``SGD(lr).step()`` (reference: server.py:54-75), i.e. θ ← θ − lr·(−Δ).
    """
    return theta.add_(delta, alpha=lr)


def sanitize_(U: Tensor) -> Tensor:
    """K18 — nan_to_num on update read (reference: client.py:198)."""
    return torch.nan_to_num_(U)
