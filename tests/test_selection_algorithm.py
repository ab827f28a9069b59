"""CPU model of the trimmed_regsel_kernel algorithm (csrc/blades_kernels.hip).

Bit-for-bit Python replica of the device algorithm — bitonic group sort,
in-place half merges (lo pairs with reversed g; hi pairs with g directly,
by the negation duality), threshold gate, bubble tail — validated against
a sort-based reference over randomized shapes, trims and tie patterns.
This is the cheap insurance that caught the hi-merge index bug in round 2
before it could only be seen on hardware.
"""
import numpy as np
import pytest


def bitonic_sort_asc(g):
    NB = len(g)
    k = 2
    while k <= NB:
        j = k >> 1
        while j > 0:
            for i in range(NB):
                l = i ^ j
                if l > i:
                    up = (i & k) == 0
                    a, b = g[i], g[l]
                    g[i] = min(a, b) if up else max(a, b)
                    g[l] = max(a, b) if up else min(a, b)
            j >>= 1
        k <<= 1
    return g


def bitonic_clean(m, asc):
    NB = len(m)
    j = NB >> 1
    while j > 0:
        for i in range(NB):
            l = i ^ j
            if l > i:
                a, b = m[i], m[l]
                m[i] = min(a, b) if asc else max(a, b)
                m[l] = max(a, b) if asc else min(a, b)
        j >>= 1
    return m


def regsel_trimmed_sum(col, NB, b_lo, b_hi):
    K = len(col)
    lo = [np.inf] * NB
    hi = [-np.inf] * NB
    s = 0.0
    k = 0
    while k + NB <= K:
        g = list(col[k:k + NB])
        s += sum(g)
        if any(v < lo[NB - 1] or v > hi[NB - 1] for v in g):
            bitonic_sort_asc(g)
            for i in range(NB):
                lo[i] = min(lo[i], g[NB - 1 - i])
            bitonic_clean(lo, True)
            for i in range(NB):
                hi[i] = max(hi[i], g[i])
            bitonic_clean(hi, False)
        k += NB
    while k < K:
        v = col[k]
        s += v
        c = v
        for i in range(NB):
            a = lo[i]
            lo[i] = min(a, c)
            c = max(a, c)
        c = v
        for i in range(NB):
            a = hi[i]
            hi[i] = max(a, c)
            c = min(a, c)
        k += 1
    return s - sum(lo[:b_lo]) - sum(hi[:b_hi])


@pytest.mark.parametrize("seed", range(6))
def test_regsel_algorithm_randomized(seed):
    rng = np.random.default_rng(seed)
    for _ in range(60):
        K = int(rng.integers(3, 260))
        NB = int(rng.choice([8, 16, 32]))
        b_lo = int(rng.integers(0, min(NB, (K - 1) // 2) + 1))
        b_hi = int(rng.integers(0, min(NB, K - b_lo - 1) + 1))
        col = rng.normal(size=K)
        if rng.random() < 0.3:
            col = np.round(col)  # heavy ties
        got = regsel_trimmed_sum(col, NB, b_lo, b_hi)
        srt = np.sort(col)
        ref = srt[b_lo:K - b_hi].sum()
        assert abs(got - ref) <= 1e-8 * max(1.0, abs(ref)), \
            (K, NB, b_lo, b_hi, got, ref)
