"""MNIST experiment: 784-100-10 MLP (reference experiments/mnist.py).

Data is deterministic synthetic MNIST-shaped (784-dim, 10 classes) with
teacher labels -- there is no network to download the real dataset; pass
``data-dir:<path>`` pointing at raw MNIST idx files to use real data when
available.
"""

import pathlib
import struct

import numpy as np
import torch

from . import _Experiment, register
from .data import SyntheticClassification
from .. import tools
from ..models import MLP


def _load_idx_images(path):
    with open(path, "rb") as f:
        magic, count, rows, cols = struct.unpack(">IIII", f.read(16))
        assert magic == 2051, f"bad idx image magic in {path}"
        data = np.frombuffer(f.read(), dtype=np.uint8).reshape(count, rows * cols)
    return torch.from_numpy(data.astype(np.float32) / 255.0)


def _load_idx_labels(path):
    with open(path, "rb") as f:
        magic, count = struct.unpack(">II", f.read(8))
        assert magic == 2049, f"bad idx label magic in {path}"
        data = np.frombuffer(f.read(), dtype=np.uint8)
    return torch.from_numpy(data.astype(np.int64))


class MNIST(_Experiment):
    """784-100-10 MLP on (synthetic or idx-file) MNIST."""

    def __init__(self, args):
        args = tools.parse_keyval(args, defaults={
            "batch-size": 32, "eval-batch-size": 1024, "seed": 1234,
            "data-dir": "", "eval-examples": 1024, "data-pool": 8,
            "signal": 0.5})
        if args["batch-size"] <= 0:
            raise tools.UserException("Cannot make batches of non-positive size")
        self.args = args
        self._real = None
        data_dir = args["data-dir"]
        if data_dir:
            d = pathlib.Path(data_dir)
            try:
                self._real = {
                    "train": (_load_idx_images(d / "train-images-idx3-ubyte"),
                              _load_idx_labels(d / "train-labels-idx1-ubyte")),
                    "test": (_load_idx_images(d / "t10k-images-idx3-ubyte"),
                             _load_idx_labels(d / "t10k-labels-idx1-ubyte")),
                }
            except (OSError, AssertionError) as e:
                raise tools.UserException(f"Cannot load MNIST from {data_dir!r}: {e}")
        self._synth = SyntheticClassification(
            (784,), 10, seed=args["seed"], eval_examples=args["eval-examples"],
            pool_size=args["data-pool"], signal=args["signal"])

    def model(self):
        return MLP((784, 100, 10))

    def train_batch(self, worker, step, device):
        bs = self.args["batch-size"]
        if self._real is not None:
            x, y = self._real["train"]
            gen = torch.Generator().manual_seed(
                (self.args["seed"] * 1000003 + worker * 7919 + step * 104729) & 0x7FFFFFFF)
            idx = torch.randint(0, x.shape[0], (bs,), generator=gen)
            return x[idx].to(device), y[idx].to(device)
        return self._synth.batch(bs, worker, step, device)

    def eval_batches(self, device):
        bs = self.args["eval-batch-size"]
        if self._real is not None:
            x, y = self._real["test"]
            for i in range(0, x.shape[0], bs):
                yield x[i:i + bs].to(device), y[i:i + bs].to(device)
            return
        yield from self._synth.eval_batches(bs, device)


register("mnist", MNIST)
