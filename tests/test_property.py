"""Property-based GAR tests (hypothesis): random shapes, NaN/inf patterns,
torch implementation vs the independent NumPy oracle."""

import numpy as np
from hypothesis import given, settings, strategies as st

from aggregathor_amd.ops import reference as R

import oracle
import torch


def _grads(draw, n, d, nan_frac, inf_frac, scale):
    rng = np.random.default_rng(draw)
    g = (rng.standard_normal((n, d)) * scale).astype(np.float32)
    if nan_frac > 0:
        mask = rng.random((n, d)) < nan_frac
        g[mask] = np.nan
    if inf_frac > 0:
        mask = rng.random((n, d)) < inf_frac
        g[mask] = np.inf * rng.choice([-1, 1], size=mask.sum())
    return g


@settings(max_examples=25, deadline=None)
@given(n=st.integers(4, 12), d=st.integers(1, 40),
       f=st.integers(0, 3), seed=st.integers(0, 10**6),
       nan=st.sampled_from([0.0, 0.05]), scale=st.sampled_from([1.0, 1e6]))
def test_krum_property(n, d, f, seed, nan, scale):
    if n - f - 2 < 1:
        return
    g = _grads(seed, n, d, nan, 0.0, scale)
    got = R.krum(torch.from_numpy(g.copy()), f).numpy()
    want = oracle.krum(g, f)
    both_nan = np.isnan(got) & np.isnan(want)
    np.testing.assert_allclose(got[~both_nan], want[~both_nan],
                               rtol=1e-4, atol=1e-5)


@settings(max_examples=25, deadline=None)
@given(n=st.integers(2, 16), d=st.integers(1, 30), seed=st.integers(0, 10**6),
       nan=st.sampled_from([0.0, 0.1, 0.6]),
       inf=st.sampled_from([0.0, 0.1]))
def test_median_property(n, d, seed, nan, inf):
    g = _grads(seed, n, d, nan, inf, 1.0)
    got = R.median(torch.from_numpy(g.copy())).numpy()
    want = oracle.median(g)
    both_nan = np.isnan(got) & np.isnan(want)
    np.testing.assert_array_equal(got[~both_nan], want[~both_nan])
    np.testing.assert_array_equal(np.isnan(got), np.isnan(want))


@settings(max_examples=20, deadline=None)
@given(n=st.integers(2, 12), d=st.integers(1, 30), seed=st.integers(0, 10**6),
       f=st.integers(0, 4), nan=st.sampled_from([0.0, 0.1]))
def test_averaged_median_property(n, d, seed, f, nan):
    beta = n - min(f, n - 1)
    g = _grads(seed, n, d, nan, 0.0, 1.0)
    got = R.averaged_median(torch.from_numpy(g.copy()), beta).numpy()
    want = oracle.averaged_median(g, beta)
    both_nan = np.isnan(got) & np.isnan(want)
    np.testing.assert_allclose(got[~both_nan], want[~both_nan],
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_array_equal(np.isnan(got), np.isnan(want))


@settings(max_examples=15, deadline=None)
@given(n=st.integers(7, 15), d=st.integers(1, 25), seed=st.integers(0, 10**6))
def test_bulyan_property(n, d, seed):
    f = (n - 3) // 4
    if f < 0 or n - 4 * f - 2 < 1:
        return
    g = _grads(seed, n, d, 0.0, 0.0, 1.0)
    got = R.bulyan(torch.from_numpy(g.copy()), f).numpy()
    want = oracle.bulyan(g, f)
    np.testing.assert_allclose(got, want, rtol=1e-3, atol=1e-4)


@settings(max_examples=20, deadline=None)
@given(n=st.integers(2, 10), d=st.integers(1, 30), seed=st.integers(0, 10**6),
       nan=st.sampled_from([0.1, 0.5, 0.95]))
def test_average_nan_property(n, d, seed, nan):
    g = _grads(seed, n, d, nan, 0.0, 1.0)
    got = R.average_nan(torch.from_numpy(g.copy())).numpy()
    want = oracle.average_nan(g)
    both_nan = np.isnan(got) & np.isnan(want)
    np.testing.assert_allclose(got[~both_nan], want[~both_nan],
                               rtol=1e-4, atol=1e-5)
    np.testing.assert_array_equal(np.isnan(got), np.isnan(want))
