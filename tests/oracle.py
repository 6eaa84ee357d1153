"""Independent NumPy oracles for the GAR math.

Deliberately written loop-style from the algorithm definitions in the
reference (op_krum/cpu.cpp, op_bulyan/cpu.cpp, deprecated_native/native.cpp)
as a second, independent code path: the package's PyTorch implementations and
HIP kernels are both validated against these.
"""

import math

import numpy as np


def _lt_nonfinite_last(a, b):
    """The reference's isfinite comparator (op_krum/cpu.cpp:81-89)."""
    if not math.isfinite(a):
        return False
    if not math.isfinite(b):
        return True
    return a < b


def _sorted_idx(values):
    """Indices sorted ascending with non-finite last, ties by index."""
    key = [(0 if math.isfinite(v) else 1, v if math.isfinite(v) else 0.0, i)
           for i, v in enumerate(values)]
    key.sort()
    return [k[2] for k in key]


def pairwise_sqdist(grads):
    n = grads.shape[0]
    out = np.full((n, n), np.inf, dtype=np.float64)
    for i in range(n):
        for j in range(i + 1, n):
            diff = grads[i].astype(np.float64) - grads[j].astype(np.float64)
            d = float((diff * diff).sum())
            out[i, j] = d
            out[j, i] = d
    return out


def krum(grads, f, m=None):
    n, d = grads.shape
    if m is None:
        m = n - f - 2
    dist = pairwise_sqdist(grads)
    scores = np.empty(n, dtype=np.float64)
    nbinscore = n - f - 2
    for i in range(n):
        row = [dist[i, j] for j in range(n) if j != i]
        order = _sorted_idx(row)
        scores[i] = sum(row[k] for k in order[:nbinscore])
    sel = _sorted_idx(scores)[:m]
    acc = np.zeros(d, dtype=np.float64)
    for i in sel:
        acc += grads[i].astype(np.float64)
    return (acc / m).astype(grads.dtype)


def bulyan(grads, f, m=None):
    n, d = grads.shape
    if m is None:
        m = n - f - 2
    t = n - 2 * f - 2
    b = t - 2 * f
    assert t >= 1 and b >= 1
    dist = pairwise_sqdist(grads)
    nbinscore = n - f - 2
    scores = np.empty(n, dtype=np.float64)
    pruned = np.zeros((n, n), dtype=np.float64)
    for i in range(n):
        others = [j for j in range(n) if j != i]
        vals = [dist[i, j] for j in others]
        order = _sorted_idx(vals)
        scores[i] = sum(vals[k] for k in order[:nbinscore])
        for k in order[:nbinscore]:
            pruned[i, others[k]] = dist[i, others[k]]
    inters = np.zeros((t, d), dtype=np.float64)
    alive = [True] * n
    for k in range(t):
        order = _sorted_idx(scores)
        sel = order[: m - k]
        for i in sel:
            inters[k] += grads[i].astype(np.float64)
        inters[k] /= (m - k)
        if k + 1 >= t:
            break
        evicted = order[0]
        scores[evicted] = np.finfo(np.float64).max
        alive[evicted] = False
        for i in range(n):
            if i != evicted and alive[i]:
                scores[i] -= pruned[i, evicted]
    return averaged_median(inters.astype(grads.dtype), b)


def median(grads):
    n, d = grads.shape
    out = np.empty(d, dtype=grads.dtype)
    for x in range(d):
        col = grads[:, x].tolist()
        order = _sorted_idx(col)
        out[x] = col[order[n // 2]]
    return out


def averaged_median(grads, beta):
    n, d = grads.shape
    out = np.empty(d, dtype=grads.dtype)
    for x in range(d):
        col = grads[:, x].astype(np.float64).tolist()
        order = _sorted_idx(col)
        zero = col[order[n // 2]]
        deltas = [abs(v - zero) for v in col]
        chosen = _sorted_idx(deltas)[:beta]
        out[x] = sum(col[i] for i in chosen) / beta
    return out


def average_nan(grads):
    n, d = grads.shape
    out = np.empty(d, dtype=grads.dtype)
    for x in range(d):
        vals = [v for v in grads[:, x].tolist() if math.isfinite(v)]
        out[x] = (sum(vals) / len(vals)) if vals else np.nan
    return out
