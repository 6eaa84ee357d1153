"""Experiment (model + dataset + loss + metric) plugin layer.

Re-expression of the reference's ``experiments`` package
(/root/reference/experiments/__init__.py:40-85): an abstract ``_Experiment``
base, a named registry, auto-import of sibling modules.

API differences from the reference (which built TF graph nodes): an
experiment owns a PyTorch model factory and a per-worker batch source; the
training engine calls ``loss`` per worker micro-batch and ``eval_batches``
for the accuracy service. Datasets default to deterministic synthetic data
(this environment has no network; BASELINE.json's big configs are synthetic
by design) with teacher-generated labels for the small models so
convergence/attack tests have real signal.
"""

import os

from .. import tools


class _Experiment:
    """Base class of all experiments."""

    #: Metric key reported by ``accuracy`` (reference experiments/mnist.py:148)
    metric_name = "top1-X-acc"

    def __init__(self, args):
        raise NotImplementedError

    def model(self):
        """Build and return a fresh randomly-initialized ``nn.Module``."""
        raise NotImplementedError

    def train_batch(self, worker, step, device):
        """Deterministic training micro-batch for the given worker and step.

        Returns (inputs, targets) on ``device``.
        """
        raise NotImplementedError

    def loss(self, model, batch):
        """Scalar training loss for a batch."""
        import torch.nn.functional as F
        inputs, targets = batch
        return F.cross_entropy(model(inputs), targets)

    def eval_batches(self, device):
        """Iterable of evaluation (inputs, targets) batches."""
        raise NotImplementedError

    def accuracy(self, model, device):
        """Top-1 accuracy over the evaluation set: {metric_name: value}."""
        import torch
        correct = total = 0
        was_training = model.training
        model.eval()
        with torch.no_grad():
            for inputs, targets in self.eval_batches(device):
                pred = model(inputs).argmax(dim=1)
                correct += (pred == targets).sum().item()
                total += targets.numel()
        if was_training:
            model.train()
        return {self.metric_name: correct / max(total, 1)}


# ---------------------------------------------------------------------------- #
# Registry (reference experiments/__init__.py:77-80)

_register = tools.ClassRegister("experiment")


def itemize():
    return _register.itemize()


def register(name, cls):
    return _register.register(name, cls)


def instantiate(name, args=None):
    return _register.instantiate(name, args or [])


def get(name):
    return _register.get(name)


tools.import_directory(__name__, os.path.dirname(__file__))
