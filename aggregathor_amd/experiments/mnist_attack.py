"""Data-poisoning MNIST experiment (reference experiments/mnistAttack.py).

Worker 0's training inputs are poisoned: scaled by -100 (``malformed-
severity:1``) or -1e12 (``malformed-severity:2``) and pixel-permuted
(reference mnistAttack.py:83-90). A robust GAR's accuracy curve vs
``average``'s under this experiment is the attack-validation check.
"""

import torch

from . import register
from .mnist import MNIST
from .. import tools


class MNISTAttack(MNIST):
    """MNIST with worker 0's data poisoned."""

    def __init__(self, args):
        args = list(args or [])
        parsed = tools.parse_keyval(args, defaults={"malformed-severity": 1})
        severity = parsed.pop("malformed-severity")
        if severity not in (1, 2):
            raise tools.UserException(
                f"malformed-severity must be 1 or 2, got {severity!r}")
        # Forward the remaining args to the MNIST base.
        base_args = [a for a in args if not a.startswith("malformed-severity:")]
        super().__init__(base_args)
        self.severity = severity
        gen = torch.Generator().manual_seed(0xBADD00D)
        self.permutation = torch.randperm(784, generator=gen)

    def train_batch(self, worker, step, device):
        x, y = super().train_batch(worker, step, device)
        if worker == 0:
            scale = -100.0 if self.severity == 1 else -1e12
            x = (x * scale)[:, self.permutation.to(x.device)]
        return x, y


register("mnistAttack", MNISTAttack)
