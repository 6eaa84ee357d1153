"""Coordinate-wise median GAR.

Reference: aggregators/median.py:40-68 + deprecated_native
native.cpp:678-704 (element at rank n//2, non-finite values ordered last).
"""

from . import _GAR, register
from .. import ops


class MedianGAR(_GAR):
    """Coordinate-per-coordinate median (NaN sorted to +inf)."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        pass

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.median(gradients)


register("median", MedianGAR)
