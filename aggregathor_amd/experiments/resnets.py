"""ResNet experiment family.

Capability parity with the reference's slim cross-product registry
(experiments/slims.py:164-196, ``slim-<model>-<dataset>``): every ResNet in
the model zoo is registered against the ``imagenet`` (3x224x224, 1000
classes) and ``cifar10`` (3x32x32, 10 classes) dataset shapes, under both a
concise name (``resnet50-imagenet``) and the reference's slim alias
(``slim-resnet_v1_50-imagenet``). Data is synthetic (BASELINE.json's big
configs are synthetic/random-init by definition).
"""

from . import _Experiment, register
from .data import SyntheticClassification
from .datasets import RealDataset
from .. import tools
from ..models import NETWORKS

_DATASETS = {
    "imagenet": {"shape": (3, 224, 224), "classes": 1000},
    "cifar10": {"shape": (3, 32, 32), "classes": 10},
}


class ResNetExperiment(_Experiment):
    def __init__(self, args, arch="resnet50", dataset="imagenet"):
        args = tools.parse_keyval(args, defaults={
            "batch-size": 32, "eval-batch-size": 256, "seed": 1234,
            "eval-examples": 512, "image-size": 0, "data-dir": "",
            "data-pool": 8, "signal": 0.5})
        if args["batch-size"] <= 0:
            raise tools.UserException("Cannot make batches of non-positive size")
        self.args = args
        self.arch = arch
        spec = _DATASETS[dataset]
        shape = spec["shape"]
        if args["image-size"] > 0:
            shape = (shape[0], args["image-size"], args["image-size"])
        self.classes = spec["classes"]
        self._real = None
        if args["data-dir"]:
            # cifar10: the binary distribution; imagenet: npz tensor shards
            # (no JPEG decoder in this image -- datasets.py docstring).
            if dataset == "cifar10":
                self._real = RealDataset.cifar10(args["data-dir"],
                                                 seed=args["seed"])
            else:
                self._real = RealDataset.tensor_folder(args["data-dir"],
                                                       seed=args["seed"])
        self._synth = SyntheticClassification(
            shape, self.classes, seed=args["seed"],
            eval_examples=args["eval-examples"],
            pool_size=args["data-pool"], signal=args["signal"])

    def train_batch(self, worker, step, device):
        src = self._real if self._real is not None else self._synth
        return src.batch(self.args["batch-size"], worker, step, device)

    def model(self):
        return NETWORKS[self.arch](num_classes=self.classes)

    def eval_batches(self, device):
        src = self._real if self._real is not None else self._synth
        yield from src.eval_batches(self.args["eval-batch-size"], device)


def _make(arch, dataset):
    class _Bound(ResNetExperiment):
        def __init__(self, args):
            super().__init__(args, arch=arch, dataset=dataset)
    _Bound.__name__ = f"ResNet_{arch}_{dataset}"
    return _Bound


def _slim_alias(arch):
    """Reference slim-name alias (experiments/slims.py naming convention)."""
    if arch.startswith("resnet"):
        return f"resnet_v1_{arch[len('resnet'):]}"
    if arch.startswith("vgg"):
        return f"vgg_{arch[len('vgg'):]}"
    return arch


for _arch in NETWORKS:
    for _ds in _DATASETS:
        if min(_DATASETS[_ds]["shape"][1:]) < getattr(
                NETWORKS[_arch], "min_input", 0):
            continue  # e.g. inception_v3 needs >= 75 px (models/__init__)
        cls = _make(_arch, _ds)
        register(f"{_arch}-{_ds}", cls)
        alias = f"slim-{_slim_alias(_arch)}-{_ds}"
        if alias != f"{_arch}-{_ds}":
            register(alias, cls)
