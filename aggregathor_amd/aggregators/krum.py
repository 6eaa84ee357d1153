"""Multi-Krum GAR.

Reference: aggregators/krum.py:49-158 and native/op_krum/cpu.cpp:53-122.
Score of gradient i = sum of its ``n - f - 2`` smallest squared L2 distances
to the other gradients (non-finite distances ordered last); the aggregate is
the mean of the ``m = n - f - 2`` smallest-scoring gradients.

The reference ships three interchangeable implementations (krum-py /
krum-tf / krum-co); here one implementation serves all three names: HIP
CDNA4 kernels on GPU, the PyTorch oracle on CPU.
"""

from . import _GAR, register
from .. import ops, tools


class KrumGAR(_GAR):
    """Multi-Krum with m = n - f - 2 selected gradients."""

    def __init__(self, nbworkers, nbbyzwrks, args):
        self._n = nbworkers
        self._f = nbbyzwrks
        self._m = nbworkers - nbbyzwrks - 2
        if self._m < 1:
            raise tools.UserException(
                f"krum requires n - f - 2 >= 1 (got n={nbworkers}, f={nbbyzwrks})")

    def aggregate(self, gradients):
        assert len(gradients) > 0, "Empty list of gradient to aggregate"
        return ops.krum(gradients, self._f, self._m)


register("krum", KrumGAR)
# Reference-name aliases (krum.py:164-167) for CLI drop-in compatibility.
for _alias in ("krum-py", "krum-tf", "krum-co"):
    register(_alias, KrumGAR)
