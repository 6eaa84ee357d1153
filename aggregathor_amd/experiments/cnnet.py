"""CNNet experiment: the reference's custom CIFAR-10 CNN
(experiments/cnnet.py:58-95).

Data: real CIFAR-10 via ``data-dir:<path>`` pointing at the binary
distribution (the reference read real CIFAR-10 through a slim dataset
provider + queue pipeline, cnnet.py:115-146, with dataset-dir permission
checks, cnnet.py:187-196); synthetic CIFAR-shaped teacher data otherwise
(no dataset download exists in this environment).
"""

from . import _Experiment, register
from .data import SyntheticClassification
from .datasets import RealDataset
from .. import tools
from ..models import CNNet


class CNNetExperiment(_Experiment):
    def __init__(self, args):
        args = tools.parse_keyval(args, defaults={
            "batch-size": 32, "eval-batch-size": 1024, "seed": 1234,
            "eval-examples": 1024, "data-dir": "", "data-pool": 8,
            "signal": 0.5})
        if args["batch-size"] <= 0:
            raise tools.UserException("Cannot make batches of non-positive size")
        self.args = args
        self._real = None
        if args["data-dir"]:
            self._real = RealDataset.cifar10(args["data-dir"],
                                             seed=args["seed"])
        self._synth = SyntheticClassification(
            (3, 32, 32), 10, seed=args["seed"],
            eval_examples=args["eval-examples"],
            pool_size=args["data-pool"], signal=args["signal"])

    def model(self):
        return CNNet()

    def train_batch(self, worker, step, device):
        src = self._real if self._real is not None else self._synth
        return src.batch(self.args["batch-size"], worker, step, device)

    def eval_batches(self, device):
        src = self._real if self._real is not None else self._synth
        yield from src.eval_batches(self.args["eval-batch-size"], device)


register("cnnet", CNNetExperiment)
