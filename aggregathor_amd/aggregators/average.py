"""Non-robust arithmetic-mean GAR (reference aggregators/average.py:47-60)."""

from . import _GAR, register
from .. import ops


class AverageGAR(_GAR):
    """Plain average of the n gradients -- the non-robust baseline."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        pass

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.average(gradients)


register("average", AverageGAR)
