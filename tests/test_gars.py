"""GAR numerics: PyTorch reference implementations vs independent NumPy oracle."""

import math

import numpy as np
import pytest
import torch

from aggregathor_amd import aggregators
from aggregathor_amd.ops import reference as R

import oracle


def _rand(n, d, seed=0):
    rng = np.random.default_rng(seed)
    return rng.standard_normal((n, d)).astype(np.float32)


def _t(a):
    return torch.from_numpy(np.ascontiguousarray(a))


# ---------------------------------------------------------------------------- #
# Pairwise distances


@pytest.mark.parametrize("n,d", [(4, 7), (8, 100), (11, 33)])
def test_pairwise_sqdist(n, d):
    g = _rand(n, d, seed=n * d)
    got = R.pairwise_sqdist(_t(g)).numpy()
    want = oracle.pairwise_sqdist(g)
    off = ~np.eye(n, dtype=bool)
    np.testing.assert_allclose(got[off], want[off], rtol=1e-5)
    assert np.all(np.isinf(np.diag(got)))


# ---------------------------------------------------------------------------- #
# Krum


@pytest.mark.parametrize("n,f", [(5, 0), (8, 2), (11, 3), (16, 4)])
def test_krum_matches_oracle(n, f):
    g = _rand(n, 50, seed=n + f)
    got = R.krum(_t(g), f).numpy()
    want = oracle.krum(g, f)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def test_krum_selects_honest_majority():
    # 7 honest gradients near a common point + 1 wild outlier: the outlier
    # must never be selected with f=1 (m = 8 - 1 - 2 = 5).
    g = _rand(8, 32, seed=3) * 0.01 + 1.0
    g[5] = 1e6
    agg = R.krum(_t(g), 1).numpy()
    assert np.all(np.abs(agg - 1.0) < 1.0)


def test_krum_nan_gradient_excluded():
    g = _rand(8, 16, seed=4)
    g[2, 5] = np.nan  # all of gradient 2's distances become NaN -> ordered last
    got = R.krum(_t(g), 2).numpy()
    want = oracle.krum(g, 2)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)
    assert np.all(np.isfinite(got))


def test_krum_m_equals_n_is_average():
    # Reference fast path (aggregators/krum.py:56-62): m == n averages all.
    g = _rand(6, 10, seed=5)
    got = R.krum(_t(g), 0, m=6).numpy()
    np.testing.assert_allclose(got, g.mean(axis=0), rtol=1e-5, atol=1e-6)


# ---------------------------------------------------------------------------- #
# Bulyan


@pytest.mark.parametrize("n,f", [(7, 1), (11, 2), (15, 3)])
def test_bulyan_matches_oracle(n, f):
    g = _rand(n, 40, seed=10 * n + f)
    got = R.bulyan(_t(g), f).numpy()
    want = oracle.bulyan(g, f)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_bulyan_resists_outlier():
    g = _rand(11, 24, seed=6) * 0.01 + 2.0
    g[0] = -1e8
    g[7] = 1e8
    agg = R.bulyan(_t(g), 2).numpy()
    assert np.all(np.abs(agg - 2.0) < 1.0)


def test_bulyan_requires_enough_workers():
    from aggregathor_amd import tools
    with pytest.raises(tools.UserException):
        aggregators.instantiate("bulyan", 8, 2)  # 8 < 4*2+3


# ---------------------------------------------------------------------------- #
# Coordinate-wise rules


@pytest.mark.parametrize("n,d", [(3, 17), (8, 64), (9, 5)])
def test_median_matches_oracle(n, d):
    g = _rand(n, d, seed=n * 7 + d)
    got = R.median(_t(g)).numpy()
    want = oracle.median(g)
    np.testing.assert_array_equal(got, want)  # exact: median picks an element


def test_median_nan_last():
    g = np.array([[1.0, np.nan], [np.nan, 5.0], [3.0, np.nan], [2.0, 7.0]],
                 dtype=np.float32)
    got = R.median(_t(g)).numpy()
    want = oracle.median(g)
    # col 0: sorted [1,2,3,nan] -> idx 2 = 3; col 1: [5,7,nan,nan] -> idx 2 = nan
    assert got[0] == want[0] == 3.0
    assert math.isnan(got[1]) and math.isnan(want[1])


@pytest.mark.parametrize("n,f", [(4, 1), (8, 2), (9, 0)])
def test_averaged_median_matches_oracle(n, f):
    g = _rand(n, 31, seed=n + 100 * f)
    beta = n - f
    got = R.averaged_median(_t(g), beta).numpy()
    want = oracle.averaged_median(g, beta)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def test_average_nan_matches_oracle():
    g = _rand(6, 20, seed=8)
    g[0, 3] = np.nan
    g[1, 3] = np.inf
    g[:, 7] = np.nan  # whole coordinate lost -> NaN output
    got = R.average_nan(_t(g)).numpy()
    want = oracle.average_nan(g)
    finite = np.isfinite(want)
    np.testing.assert_allclose(got[finite], want[finite], rtol=1e-5, atol=1e-6)
    assert math.isnan(got[7]) and math.isnan(want[7])


# ---------------------------------------------------------------------------- #
# Registry / plugin layer


def test_registry_names():
    names = set(aggregators.itemize())
    for want in ("average", "average-nan", "median", "averaged-median",
                 "krum", "krum-py", "krum-tf", "krum-co",
                 "bulyan", "bulyan-py", "bulyan-co"):
        assert want in names, f"missing GAR {want!r}"


@pytest.mark.parametrize("name,n,f", [
    ("average", 4, 0), ("average-nan", 4, 0), ("median", 5, 1),
    ("averaged-median", 8, 2), ("krum", 8, 2), ("bulyan", 11, 2),
])
def test_aggregate_via_registry(name, n, f):
    gar = aggregators.instantiate(name, n, f)
    g = _t(_rand(n, 12, seed=42))
    out = gar.aggregate(g)
    assert out.shape == (12,)
    assert out.dtype == torch.float32


def test_krum_invalid_config():
    from aggregathor_amd import tools
    with pytest.raises(tools.UserException):
        aggregators.instantiate("krum", 3, 1)  # n - f - 2 = 0
