"""NaN-skipping coordinate-wise mean GAR.

Reference: aggregators/average-nan.py:40-68 + deprecated_native
native.cpp:756-782. Pairs with the lossy-gradient (UDP-style) transport
injection: dropped chunks surface as NaN coordinates and are excluded from
the mean.
"""

from . import _GAR, register
from .. import ops


class AverageNaNGAR(_GAR):
    """Coordinate-wise mean over the finite values only."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        pass

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.average_nan(gradients)


register("average-nan", AverageNaNGAR)
