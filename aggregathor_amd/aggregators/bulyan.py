"""Bulyan over Multi-Krum GAR.

Reference: aggregators/bulyan.py:43-94 and native/op_bulyan/cpu.cpp:54-188.
``t = n - 2f - 2`` Multi-Krum selection rounds (with distance pruning and
score decrement on eviction), then a coordinate-wise averaged-median with
``b = t - 2f`` over the selected vectors. Requires n >= 4f + 3.
"""

from . import _GAR, register
from .. import ops, tools


class BulyanGAR(_GAR):
    """Bulyan of Multi-Krum with m = n - f - 2."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        self._n = nbworkers
        self._f = nbbyzwrks
        self._m = nbworkers - nbbyzwrks - 2
        t = nbworkers - 2 * nbbyzwrks - 2
        b = t - 2 * nbbyzwrks
        if b < 1:
            raise tools.UserException(
                f"bulyan requires n >= 4f + 3 (got n={nbworkers}, f={nbbyzwrks}: "
                f"t={t}, beta={b})")

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.bulyan(gradients, self._f, self._m)


register("bulyan", BulyanGAR)
# Reference-name aliases (bulyan.py:90-92).
for _alias in ("bulyan-py", "bulyan-co"):
    register(_alias, BulyanGAR)
