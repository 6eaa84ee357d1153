"""Gradient-space Byzantine attacks.

``reversal`` is the attack named by BASELINE.json's ResNet-50 config
("Multi-Krum f=2 + gradient-reversal attack"); the others are standard
Byzantine baselines used to validate GAR robustness.
"""

import torch

from . import _Attack, register
from .. import tools


class ReversalAttack(_Attack):
    """Submit ``-factor x`` the honest gradient (gradient ascent)."""

    graph_safe = True  # pure device math: hipGraph-capturable

    def __init__(self, args):
        self.args = tools.parse_keyval(args, defaults={"factor": 1.0})

    def craft(self, honest, worker, step):
        return honest * (-self.args["factor"])


class RandomAttack(_Attack):
    """Submit Gaussian noise of a chosen scale."""

    def __init__(self, args):
        self.args = tools.parse_keyval(args, defaults={"scale": 1.0, "seed": 666})

    def craft(self, honest, worker, step):
        gen = torch.Generator().manual_seed(
            (self.args["seed"] * 1000003 + worker * 7919 + step * 104729) & 0x7FFFFFFF)
        noise = torch.randn(honest.shape, generator=gen, dtype=torch.float32)
        return noise.to(honest.device, honest.dtype) * self.args["scale"]


class ZeroAttack(_Attack):
    """Submit the zero gradient (stalling attack)."""

    graph_safe = True

    def __init__(self, args):
        tools.parse_keyval(args)

    def craft(self, honest, worker, step):
        return torch.zeros_like(honest)


class MagnitudeAttack(_Attack):
    """Submit an arbitrarily-scaled honest gradient (the classic unbounded
    attack Krum was designed against)."""

    graph_safe = True

    def __init__(self, args):
        self.args = tools.parse_keyval(args, defaults={"factor": 1e6})

    def craft(self, honest, worker, step):
        return honest * self.args["factor"]


class NaNAttack(_Attack):
    """Submit all-NaN coordinates (tests non-finite handling end to end)."""

    graph_safe = True

    def __init__(self, args):
        tools.parse_keyval(args)

    def craft(self, honest, worker, step):
        return torch.full_like(honest, float("nan"))


register("reversal", ReversalAttack)
register("random", RandomAttack)
register("zero", ZeroAttack)
register("magnitude", MagnitudeAttack)
register("nan", NaNAttack)
