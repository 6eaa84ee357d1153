"""Averaged-median GAR: mean of the beta = n - f coordinates closest to the
coordinate-wise median.

Reference: aggregators/averaged-median.py:40-67 (beta = nbworkers -
nbbyzwrks) + deprecated_native native.cpp:714-747.
"""

from . import _GAR, register
from .. import ops


class AveragedMedianGAR(_GAR):
    """NaN-tolerant averaged median."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        self._beta = nbworkers - nbbyzwrks

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.averaged_median(gradients, self._beta)


register("averaged-median", AveragedMedianGAR)
