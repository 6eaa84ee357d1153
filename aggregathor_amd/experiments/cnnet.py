"""CNNet experiment: the reference's custom CIFAR-10 CNN
(experiments/cnnet.py:58-95) on synthetic CIFAR-shaped data."""

from . import _Experiment, register
from .data import SyntheticClassification
from .. import tools
from ..models import CNNet


class CNNetExperiment(_Experiment):
    def __init__(self, args):
        args = tools.parse_keyval(args, defaults={
            "batch-size": 32, "eval-batch-size": 1024, "seed": 1234,
            "eval-examples": 1024})
        if args["batch-size"] <= 0:
            raise tools.UserException("Cannot make batches of non-positive size")
        self.args = args
        self._synth = SyntheticClassification(
            (3, 32, 32), 10, seed=args["seed"],
            eval_examples=args["eval-examples"])

    def model(self):
        return CNNet()

    def train_batch(self, worker, step, device):
        return self._synth.batch(self.args["batch-size"], worker, step, device)

    def eval_batches(self, device):
        yield from self._synth.eval_batches(self.args["eval-batch-size"], device)


register("cnnet", CNNetExperiment)
