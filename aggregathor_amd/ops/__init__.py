"""GAR compute ops: HIP/CDNA4 kernels on GPU, PyTorch oracle on CPU.

Every public function takes the stacked ``[n, d]`` gradient matrix and
dispatches on its device:

- CUDA (= ROCm/HIP) tensors run the hand-written gfx950 kernels from the
  in-tree ``_gar_hip`` torch extension (``aggregathor_amd/ops/csrc``). If the
  extension is not importable on a machine with a GPU, the op FAILS LOUDLY
  instead of silently falling back to eager PyTorch -- a silent fallback
  would invalidate every benchmark (the reference had the same class of trap:
  TF silently placing kernels on CPU cost "two orders of magnitude",
  /root/reference/README.md:174-182).
- CPU tensors run the pure-PyTorch reference implementations
  (:mod:`aggregathor_amd.ops.reference`), which are also the numerics oracle
  for the HIP kernels.

Set ``AGGREGATHOR_FORCE_EAGER=1`` to force the PyTorch path on GPU (debug
only; benchmarks refuse it).
"""

import os

import torch

from . import reference

_EXT = None
_EXT_ERROR = None


def _load_extension():
    """Import the in-tree HIP extension, caching the result."""
    global _EXT, _EXT_ERROR
    if _EXT is not None or _EXT_ERROR is not None:
        return _EXT
    try:
        from . import _gar_hip  # built in-tree by `python -m aggregathor_amd.ops.build`
        _EXT = _gar_hip
    except ImportError as e:
        _EXT_ERROR = e
    return _EXT


def hip_available():
    """Whether the HIP extension is importable."""
    return _load_extension() is not None


def itemize():
    """Names of the available GAR compute ops (the reference's
    ``native.itemize_op`` analog, native/__init__.py:374-377)."""
    return ["pairwise_sqdist", "krum", "bulyan", "median", "averaged_median",
            "average_nan", "average", "selection_average"]


def _want_hip(tensor):
    if tensor.device.type != "cuda":
        return False
    if os.environ.get("AGGREGATHOR_FORCE_EAGER") == "1":
        return False
    if _load_extension() is None:
        raise RuntimeError(
            "aggregathor_amd HIP extension '_gar_hip' is not built but a GPU "
            "tensor was passed; build it with "
            "`python -m aggregathor_amd.ops.build` (refusing to silently "
            f"fall back to eager PyTorch). Import error: {_EXT_ERROR}")
    return True


def _as_contig_2d(grads):
    assert grads.dim() == 2, f"expected [n, d] stacked gradients, got {tuple(grads.shape)}"
    return grads.contiguous()


# ---------------------------------------------------------------------------- #
# Public ops


def pairwise_sqdist(grads):
    """[n, n] squared L2 distance matrix, diagonal = +inf."""
    grads = _as_contig_2d(grads)
    if _want_hip(grads):
        return _EXT.pairwise_sqdist(grads)
    return reference.pairwise_sqdist(grads)


def average(grads):
    grads = _as_contig_2d(grads)
    return grads.mean(dim=0)  # memory-bound mean: PyTorch's reduction is optimal


def average_nan(grads):
    grads = _as_contig_2d(grads)
    if _want_hip(grads):
        return _EXT.average_nan(grads)
    return reference.average_nan(grads)


def median(grads):
    grads = _as_contig_2d(grads)
    if _want_hip(grads):
        return _EXT.median(grads)
    return reference.median(grads)


def averaged_median(grads, beta):
    grads = _as_contig_2d(grads)
    if _want_hip(grads):
        return _EXT.averaged_median(grads, int(beta))
    return reference.averaged_median(grads, beta)


def krum(grads, f, m=None):
    grads = _as_contig_2d(grads)
    if m is None:
        m = grads.shape[0] - f - 2
    if _want_hip(grads):
        return _EXT.krum(grads, int(f), int(m))
    return reference.krum(grads, f, m)


def bulyan(grads, f, m=None):
    grads = _as_contig_2d(grads)
    if m is None:
        m = grads.shape[0] - f - 2
    if _want_hip(grads):
        return _EXT.bulyan(grads, int(f), int(m))
    return reference.bulyan(grads, f, m)
