"""Pure-PyTorch fp32/fp64 reference implementations of every GAR.

These are the ground-truth oracles the HIP/CDNA4 kernels are validated
against, and the CPU execution path of every aggregator. They reproduce the
reference's algorithms exactly:

- Multi-Krum score/selection: /root/reference/native/op_krum/cpu.cpp:53-122
  and /root/reference/aggregators/krum.py:49-87 (score = sum of the
  ``n - f - 2`` smallest squared distances of each gradient; non-finite
  distances ordered last).
- Bulyan over Multi-Krum: /root/reference/native/op_bulyan/cpu.cpp:54-188
  (Krum pass with distance pruning, ``t = n - 2f - 2`` selection rounds with
  score decrement, final coordinate-wise averaged-median over the selected
  vectors with ``b = t - 2f``).
- Coordinate-wise median:
  /root/reference/aggregators/deprecated_native/native.cpp:678-704
  (``nth_element`` at index ``n // 2`` with non-finite values ordered last).
- Averaged-median: native.cpp:714-747 (mean of the ``beta`` coordinates
  closest to the coordinate median).
- NaN-skipping average: native.cpp:756-782.

Tie-breaking: where the reference leaves ``std::nth_element`` order
unspecified for equal keys, these implementations break ties by ascending
index (stable sort), which is a deterministic refinement of the reference's
behavior. Non-finite values (NaN and +/-inf) always order last, matching the
reference's ``isfinite`` comparators.
"""

import torch

__all__ = [
    "pairwise_sqdist", "krum_select", "krum", "bulyan", "median",
    "averaged_median", "average_nan", "average",
]


def _check_stacked(grads):
    assert grads.dim() == 2, f"expected stacked [n, d] gradients, got shape {tuple(grads.shape)}"
    return grads.shape


def pairwise_sqdist(grads):
    """All-pairs squared L2 distances of the n stacked gradients.

    Returns an ``[n, n]`` symmetric matrix with +inf on the diagonal (the
    diagonal is never a candidate; the reference stores ``T::max`` there,
    op_bulyan/cpu.cpp:75). Computed as a direct sum of squared differences,
    like the reference's ``reduce_sum_squared_difference``
    (native/include/operations.hpp:46-58) -- NOT via the Gram-matrix
    identity, so no catastrophic cancellation.
    """
    n, _ = _check_stacked(grads)
    diff = grads.unsqueeze(1) - grads.unsqueeze(0)          # [n, n, d]
    dist = (diff * diff).sum(dim=2)
    dist.fill_diagonal_(float("inf"))
    return dist


def _nonfinite_last_key(x):
    """Map values so that ascending sort orders non-finite entries last."""
    return torch.where(torch.isfinite(x), x, torch.full_like(x, float("inf")))


def krum_select(dist, f, m):
    """Multi-Krum scores and selection from a pairwise-distance matrix.

    Args:
      dist: [n, n] squared distances, diagonal +inf (or T::max).
      f:    number of declared Byzantine workers.
      m:    number of gradients to select (reference: ``n - f - 2``).
    Returns:
      (scores [n], selected indices [m] in ascending-score order).
    """
    n = dist.shape[0]
    nbinscore = n - f - 2
    assert 1 <= nbinscore <= n - 1, f"krum needs 1 <= n-f-2 <= n-1, got n={n} f={f}"
    assert 1 <= m <= n, f"krum selection count m={m} out of range for n={n}"
    key = _nonfinite_last_key(dist)
    # Score = sum of each row's nbinscore smallest distances (self excluded by
    # the +inf diagonal). The global-rank walk of op_krum/cpu.cpp:91-102 is
    # equivalent to each row's own ascending order.
    row_sorted, _ = torch.sort(key, dim=1)
    scores = row_sorted[:, :nbinscore].sum(dim=1)
    # Select the m smallest scores; stable sort = index tie-break.
    order = torch.argsort(_nonfinite_last_key(scores), stable=True)
    return scores, order[:m]


def krum(grads, f, m=None):
    """Multi-Krum GAR: average of the m smallest-scoring gradients."""
    n, _ = _check_stacked(grads)
    if m is None:
        m = n - f - 2
    dist = pairwise_sqdist(grads)
    _, selected = krum_select(dist, f, m)
    return grads[selected].mean(dim=0)


def bulyan(grads, f, m=None):
    """Bulyan over Multi-Krum GAR (op_bulyan/cpu.cpp:54-188 semantics).

    ``t = n - 2f - 2`` Krum selection rounds, each averaging the ``m - k``
    best and evicting the single best with score decrement by the *pruned*
    distance row; final coordinate-wise averaged-median with
    ``b = t - 2f``. Requires n >= 4f + 3.
    """
    n, d = _check_stacked(grads)
    if m is None:
        m = n - f - 2
    t = n - 2 * f - 2
    b = t - 2 * f
    assert t >= 1 and b >= 1, (
        f"bulyan needs n >= 4f+3 (n={n}, f={f} gives t={t}, b={b})")
    fmax = torch.finfo(grads.dtype).max

    dist = pairwise_sqdist(grads)                 # [n, n], diag=+inf
    key = _nonfinite_last_key(dist)
    # Per-row ascending order of the n-1 real distances (diag sorts last).
    row_order = torch.argsort(key, dim=1, stable=True)       # [n, n]
    nbinscore = n - f - 2
    row_sorted = torch.gather(key, 1, row_order)
    scores = row_sorted[:, :nbinscore].sum(dim=1)            # [n]
    # Distance pruning (op_bulyan/cpu.cpp:116-129): for each row i, keep only
    # the nbinscore closest distances; zero the f+1 remaining ones (and the
    # diagonal, which the eviction loop never reads for i == id).
    pruned = torch.zeros_like(dist)
    keep = row_order[:, :nbinscore]                          # [n, nbinscore]
    pruned.scatter_(1, keep, torch.gather(dist, 1, keep))
    pruned.fill_diagonal_(0)

    inters = torch.empty((t, d), dtype=grads.dtype, device=grads.device)
    scores = scores.clone()
    alive = torch.ones(n, dtype=torch.bool, device=grads.device)
    for k in range(t):
        order = torch.argsort(_nonfinite_last_key(scores), stable=True)
        sel = order[: m - k]
        inters[k] = grads[sel].mean(dim=0)
        if k + 1 >= t:
            break
        evicted = order[0]
        scores[evicted] = fmax
        alive[evicted] = False
        # scores[i] -= pruned[i, evicted] for all alive i (cpu.cpp:155-159
        # subtracts for every i != id; evicted rows hold fmax and a further
        # subtraction of a pruned distance cannot re-enter them before real
        # scores, so restricting to alive rows is equivalent and avoids
        # fmax arithmetic).
        scores[alive] -= pruned[alive, evicted]
    # Final coordinate-wise averaged-median over the t selected vectors
    # (cpu.cpp:163-187): median = element at index t//2, then average of the
    # b elements closest to it.
    return averaged_median(inters, b)


def median(grads):
    """Coordinate-wise median (native.cpp:678-704).

    Per coordinate: the element at rank ``n // 2`` under ascending order with
    non-finite values last.
    """
    n, _ = _check_stacked(grads)
    key = _nonfinite_last_key(grads)
    # kthvalue is 1-indexed; reference takes 0-indexed n//2 (upper median).
    idx = torch.argsort(key, dim=0, stable=True)[n // 2]     # [d]
    return torch.gather(grads, 0, idx.unsqueeze(0)).squeeze(0)


def averaged_median(grads, beta):
    """Coordinate-wise averaged-median (native.cpp:714-747).

    Per coordinate: ``zero`` = element at rank ``n // 2`` (non-finite last),
    then the mean of the ``beta`` elements closest to ``zero`` by absolute
    difference (ascending index on ties).
    """
    n, _ = _check_stacked(grads)
    assert 1 <= beta <= n, f"beta={beta} out of range for n={n}"
    key = _nonfinite_last_key(grads)
    med_idx = torch.argsort(key, dim=0, stable=True)[n // 2]
    zero = torch.gather(grads, 0, med_idx.unsqueeze(0)).squeeze(0)  # [d]
    delta = (grads - zero.unsqueeze(0)).abs()
    order = torch.argsort(_nonfinite_last_key(delta), dim=0, stable=True)
    chosen = order[:beta]                                    # [beta, d]
    return torch.gather(grads, 0, chosen).sum(dim=0) / float(beta)


def average_nan(grads):
    """Coordinate-wise mean skipping non-finite values (native.cpp:756-782).

    A coordinate with no finite value yields NaN (the reference's 0/0).
    """
    _check_stacked(grads)
    finite = torch.isfinite(grads)
    vals = torch.where(finite, grads, torch.zeros_like(grads))
    count = finite.sum(dim=0).to(grads.dtype)
    return vals.sum(dim=0) / count


def average(grads):
    """Plain arithmetic mean (reference aggregators/average.py:47-54)."""
    _check_stacked(grads)
    return grads.mean(dim=0)
