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


class ALIEAttack(_Attack):
    """'A Little Is Enough' (Baruch et al. 2019): submit mean + z * std of
    the honest gradients, with z small enough to evade distance-based
    selection yet consistently biased.

    Omniscient variant: when the attacker can see other workers' gradients
    (engine exposes the local honest rows via ``observe``), mean/std are
    estimated from them; otherwise the worker's own gradient serves as the
    mean estimate with zero std (degrading to a no-op).
    """

    graph_safe = True

    def __init__(self, args):
        self.args = tools.parse_keyval(args, defaults={"z": 1.0})
        self._rows = None

    def observe(self, honest_rows):
        self._rows = honest_rows

    def craft(self, honest, worker, step):
        if self._rows is not None and self._rows.shape[0] >= 2:
            mean = self._rows.mean(dim=0)
            std = self._rows.std(dim=0, unbiased=False)
            return mean + self.args["z"] * std
        return honest


class IPMAttack(_Attack):
    """Inner-product manipulation (Xie et al. 2020): submit -eps * mean of
    the honest gradients -- a negatively-aligned vector small enough to
    survive distance filters. Omniscient like ALIE."""

    graph_safe = True

    def __init__(self, args):
        self.args = tools.parse_keyval(args, defaults={"eps": 0.5})
        self._rows = None

    def observe(self, honest_rows):
        self._rows = honest_rows

    def craft(self, honest, worker, step):
        base = (self._rows.mean(dim=0)
                if self._rows is not None and self._rows.shape[0] >= 1
                else honest)
        return base * (-self.args["eps"])


register("alie", ALIEAttack)
register("ipm", IPMAttack)
