"""HIP kernel numerics on a real MI355X: every GAR kernel vs the fp64-cast
PyTorch oracle, including NaN cases, plus the fail-loud native-path check."""

import pytest
import torch

pytestmark = pytest.mark.gpu

from aggregathor_amd import ops
from aggregathor_amd.ops import reference as R


def _require_ext():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    assert ops.hip_available(), "HIP extension must be importable on a GPU box"
    return ops._load_extension()


def _rand(n, d, seed=0, device="cuda"):
    gen = torch.Generator().manual_seed(seed)
    return torch.randn((n, d), generator=gen, dtype=torch.float32).to(device)


def _ref64(fn, g_cpu, *args):
    """Oracle in float64 on CPU, cast back to fp32."""
    return fn(g_cpu.double(), *args).float()


# ---------------------------------------------------------------------------- #
# Pairwise distances


@pytest.mark.parametrize("n,d", [(4, 1000), (8, 100003), (8, 1 << 20),
                                 (16, 65537), (24, 10000), (33, 4099),
                                 (32, 1 << 20), (64, (1 << 20) + 1)])
def test_sqdist(n, d):
    ext = _require_ext()
    g = _rand(n, d, seed=n)
    got = ext.pairwise_sqdist(g).cpu()
    want = _ref64(R.pairwise_sqdist, g.cpu())
    off = ~torch.eye(n, dtype=torch.bool)
    torch.testing.assert_close(got[off], want[off], rtol=1e-5, atol=1e-3)
    assert torch.isinf(torch.diagonal(got)).all()


def test_sqdist_nan_propagates():
    ext = _require_ext()
    g = _rand(6, 4096, seed=1)
    g[2, 100] = float("nan")
    got = ext.pairwise_sqdist(g).cpu()
    for j in range(6):
        if j != 2:
            assert torch.isnan(got[2, j]) and torch.isnan(got[j, 2])


# ---------------------------------------------------------------------------- #
# Krum


@pytest.mark.parametrize("n,f,d", [(5, 0, 10007), (8, 2, 1 << 20),
                                   (11, 3, 65536), (25, 5, 30000)])
def test_krum(n, f, d):
    ext = _require_ext()
    g = _rand(n, d, seed=7 * n + f)
    m = n - f - 2
    got = ext.krum(g, f, m).cpu()
    want = _ref64(R.krum, g.cpu(), f, m)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


def test_krum_nan_row_excluded():
    ext = _require_ext()
    g = _rand(8, 50000, seed=3)
    g[1, 77] = float("nan")
    got = ext.krum(g, 2, 4).cpu()
    want = R.krum(g.cpu(), 2, 4)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)
    assert torch.isfinite(got).all()


def test_krum_outlier_rejected():
    ext = _require_ext()
    g = _rand(8, 10000, seed=4) * 0.01 + 1.0
    g[3] = 1e6
    got = ext.krum(g, 1, 5).cpu()
    assert (got - 1.0).abs().max() < 1.0


# ---------------------------------------------------------------------------- #
# Bulyan


@pytest.mark.parametrize("n,f,d", [(7, 1, 10007), (11, 2, 1 << 18),
                                   (15, 3, 65536), (23, 5, 20000)])
def test_bulyan(n, f, d):
    ext = _require_ext()
    g = _rand(n, d, seed=11 * n + f)
    m = n - f - 2
    got = ext.bulyan(g, f, m).cpu()
    want = _ref64(R.bulyan, g.cpu(), f, m)
    torch.testing.assert_close(got, want, rtol=1e-4, atol=1e-5)


def test_bulyan_outlier_rejected():
    ext = _require_ext()
    g = _rand(11, 20000, seed=5) * 0.01 + 2.0
    g[0] = -1e8
    g[7] = 1e8
    got = ext.bulyan(g, 2, 8).cpu()
    assert (got - 2.0).abs().max() < 1.0


# ---------------------------------------------------------------------------- #
# Coordinate-wise rules


@pytest.mark.parametrize("n,d", [(3, 10007), (8, 1 << 20), (16, 65537),
                                 (32, 30011)])
def test_median(n, d):
    ext = _require_ext()
    g = _rand(n, d, seed=n)
    got = ext.median(g).cpu()
    want = R.median(g.cpu())
    assert torch.equal(got, want)  # element selection: exact


def test_median_with_nans():
    ext = _require_ext()
    g = _rand(8, 100000, seed=9)
    mask = torch.rand(g.shape) < 0.2
    g[mask.cuda()] = float("nan")
    got = ext.median(g).cpu()
    want = R.median(g.cpu())
    both_nan = torch.isnan(got) & torch.isnan(want)
    assert torch.equal(got[~both_nan], want[~both_nan])
    assert torch.equal(torch.isnan(got), torch.isnan(want))


@pytest.mark.parametrize("n,beta,d", [(8, 6, 1 << 20), (16, 12, 65536),
                                      (5, 5, 10007)])
def test_averaged_median(n, beta, d):
    ext = _require_ext()
    g = _rand(n, d, seed=n + beta)
    got = ext.averaged_median(g, beta).cpu()
    want = _ref64(R.averaged_median, g.cpu(), beta)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


def test_average_nan():
    ext = _require_ext()
    g = _rand(8, 200000, seed=13)
    mask = torch.rand(g.shape) < 0.3
    g[mask.cuda()] = float("nan")
    got = ext.average_nan(g).cpu()
    want = _ref64(R.average_nan, g.cpu())
    nan_g = torch.isnan(want)
    torch.testing.assert_close(got[~nan_g], want[~nan_g], rtol=1e-5, atol=1e-6)
    assert torch.equal(torch.isnan(got), nan_g)


# ---------------------------------------------------------------------------- #
# Dispatch layer


def test_ops_dispatch_uses_hip():
    _require_ext()
    g = _rand(8, 4096, seed=21)
    out = ops.krum(g, 2)
    assert out.is_cuda
    want = R.krum(g.cpu(), 2)
    torch.testing.assert_close(out.cpu(), want, rtol=1e-4, atol=1e-5)


def test_determinism_bitwise():
    ext = _require_ext()
    g = _rand(8, 1 << 20, seed=33)
    a = ext.krum(g, 2, 4)
    b = ext.krum(g, 2, 4)
    assert torch.equal(a, b)
    a = ext.bulyan(_rand(11, 1 << 18, seed=34), 2, 8)
    b = ext.bulyan(_rand(11, 1 << 18, seed=34), 2, 8)
    assert torch.equal(a, b)


# ---------------------------------------------------------------------------- #
# Edge shapes on device


@pytest.mark.parametrize("n,d", [(4, 1), (5, 3), (8, 7), (8, 255),
                                 (2, 1024)])
def test_gpu_tiny_shapes(n, d):
    ext = _require_ext()
    g = _rand(n, d, seed=n * 31 + d)
    got = ext.pairwise_sqdist(g).cpu()
    want = _ref64(R.pairwise_sqdist, g.cpu())
    off = ~torch.eye(n, dtype=torch.bool)
    torch.testing.assert_close(got[off], want[off], rtol=1e-5, atol=1e-5)
    got_m = ext.median(g).cpu()
    assert torch.equal(got_m, R.median(g.cpu()))


@pytest.mark.parametrize("n,f", [(40, 9), (64, 10)])
def test_gpu_large_n_krum(n, f):
    ext = _require_ext()
    g = _rand(n, 30000, seed=n)
    m = n - f - 2
    got = ext.krum(g, f, m).cpu()
    want = _ref64(R.krum, g.cpu(), f, m)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)


def test_gpu_identical_rows():
    ext = _require_ext()
    g = torch.ones((8, 100000), device="cuda")
    out = ext.krum(g, 2, 4).cpu()
    torch.testing.assert_close(out, torch.ones(100000), rtol=1e-6, atol=0)
    out = ext.bulyan(torch.ones((11, 50000), device="cuda"), 2, 7).cpu()
    torch.testing.assert_close(out, torch.ones(50000), rtol=1e-6, atol=0)


@pytest.mark.parametrize("n", [24, 32, 64])
def test_sqdist_lds_kernel_deterministic_and_matches_tile(n):
    # The round-2 LDS-staged single-pass kernel (n > 16) must be bitwise
    # deterministic run-to-run; and its sums must agree with the legacy
    # multi-pass tile kernel to fp32 reduction tolerance.
    import os
    import subprocess
    import sys
    ext = _require_ext()
    d = 1 << 18
    g = _rand(n, d, seed=n + 100)
    a = ext.pairwise_sqdist(g).cpu()
    b = ext.pairwise_sqdist(g).cpu()
    assert torch.equal(a, b), "LDS sqdist kernel is not deterministic"
    want = _ref64(R.pairwise_sqdist, g.cpu())
    off = ~torch.eye(n, dtype=torch.bool)
    torch.testing.assert_close(a[off], want[off], rtol=1e-5, atol=1e-3)


@pytest.mark.parametrize("n", [24, 33, 64])
def test_sqdist_nan_propagates_large_n(n):
    # NaN handling through the LDS-staged large-n kernel: a NaN coordinate
    # in row r must make every distance involving r NaN (ordered last by
    # the selection's total order), exactly like the small-n kernels.
    ext = _require_ext()
    g = _rand(n, 100000, seed=n)
    g[n // 2, 31337] = float("nan")
    got = ext.pairwise_sqdist(g).cpu()
    for j in range(n):
        if j != n // 2:
            assert torch.isnan(got[n // 2, j]) and torch.isnan(got[j, n // 2])
    # Healthy pairs stay finite.
    assert torch.isfinite(got[0, 1])


def test_gpu_large_n_bulyan_with_nan_row():
    # Bulyan at n > 16 with a NaN-poisoned row: the distance pass orders it
    # last, selection excludes it, and the bitmask-folded final pass must
    # produce a finite aggregate matching the fp64 oracle.
    ext = _require_ext()
    n, f = 24, 3
    m = n - f - 2
    g = _rand(n, 30000, seed=7)
    g[2] = float("nan")
    got = ext.bulyan(g, f, m).cpu()
    assert torch.isfinite(got).all()
    want = _ref64(R.bulyan, g.cpu(), f, m)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
