"""Gradient Aggregation Rule (GAR) plugin layer.

Re-expression of the reference's ``aggregators`` package
(/root/reference/aggregators/__init__.py:40-74): an abstract ``_GAR`` base
class, a named registry, and auto-import of every sibling module so each
plugin registers itself.

MI355X-native differences from the reference:
- ``aggregate`` takes the already-stacked ``[n, d]`` gradient matrix (the
  RCCL all-gather output resident in HBM), not a Python list of per-worker
  tensors, and runs as HIP kernels on GPU / the PyTorch oracle on CPU.
- There is a single implementation per rule instead of the reference's
  py/tf/co triplets; the reference's names (``krum-py``/``krum-tf``/
  ``krum-co``...) are registered as aliases for CLI compatibility.
"""

import os

from .. import tools

# ---------------------------------------------------------------------------- #
# GAR base class (reference aggregators/__init__.py:40-60)


class _GAR:
    """Base class of all gradient aggregation rules."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        """
        Args:
          nbworkers: total number of workers (n)
          nbbyzwrks: declared number of Byzantine workers (f)
          args:      list of "key:value" plugin arguments
        """
        raise NotImplementedError

    def aggregate(self, gradients):
        """Aggregate the stacked gradients.

        Args:
          gradients: [n, d] tensor, one flattened gradient per row.
        Returns:
          [d] aggregated gradient tensor (same dtype/device as the input).
        """
        raise NotImplementedError


# ---------------------------------------------------------------------------- #
# GAR registry (reference aggregators/__init__.py:66-69)

_register = tools.ClassRegister("GAR")


def itemize():
    return _register.itemize()


def register(name, cls):
    return _register.register(name, cls)


def instantiate(name, nbworkers, nbbyzwrks, args=None):
    return _register.instantiate(name, nbworkers, nbbyzwrks, args or [])


def get(name):
    return _register.get(name)


# ---------------------------------------------------------------------------- #
# Auto-import sibling plugin modules (reference aggregators/__init__.py:73-74)

tools.import_directory(__name__, os.path.dirname(__file__))
