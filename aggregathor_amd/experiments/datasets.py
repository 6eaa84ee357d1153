"""Real-dataset input pipeline.

The reference read real CIFAR-10 through a slim dataset provider + queue
pipeline (/root/reference/experiments/cnnet.py:115-146) and the slim
experiments read dataset directories (slims.py:164-196), both guarded by
recursive permission checks (cnnet.py:187-196). This module is the
MI355X-native equivalent:

* ``load_cifar10_binary`` reads the CIFAR-10 **binary** distribution
  (``data_batch_{1..5}.bin`` + ``test_batch.bin``, 10000 records each of
  1 label byte + 3072 CHW pixel bytes) with pure numpy.
* ``load_tensor_folder`` reads a directory of ``.npz`` shards
  (``images`` uint8 [N, C, H, W], ``labels`` int64 [N]; shards named
  ``train*.npz`` / ``test*.npz``) -- the stand-in for slim's TFRecord
  ImageNet dirs, since this image has no JPEG decoder (no PIL/cv2).
* ``RealDataset`` serves deterministic per-(worker, step) batches from the
  loaded tensors. MI355X-first: instead of a host-side queue pipeline, the
  ENTIRE dataset is moved to HBM once (CIFAR-10 is 0.7 GB as fp32 out of
  288 GB) and batches are device-side index selects -- no per-step H2D.
"""

import pathlib

import numpy as np
import torch

from .. import tools


def check_dataset_dir(path, what="dataset"):
    """Reference cnnet.py:187-196 semantics: the dataset dir must exist and
    be recursively readable."""
    p = pathlib.Path(path)
    if not p.is_dir():
        raise tools.UserException(
            f"{what} {str(path)!r} must be a directory")
    if not tools.can_access(p, read=True, recursive=True):
        raise tools.UserException(
            f"{what} {str(path) + '/*'!r} must be read-able")
    return p


def _read_cifar_file(path):
    raw = np.fromfile(str(path), dtype=np.uint8)
    if raw.size % 3073 != 0:
        raise tools.UserException(
            f"{str(path)!r} is not CIFAR-10 binary format "
            f"(size {raw.size} not a multiple of 3073)")
    raw = raw.reshape(-1, 3073)
    labels = raw[:, 0].astype(np.int64)
    images = raw[:, 1:].reshape(-1, 3, 32, 32)
    return images, labels


def load_cifar10_binary(data_dir):
    """Load the CIFAR-10 binary distribution from ``data_dir`` (accepts the
    dir itself or one containing the standard ``cifar-10-batches-bin``
    subdirectory). Returns (train_x u8 [N,3,32,32], train_y, test_x, test_y).
    """
    d = check_dataset_dir(data_dir, what="CIFAR-10 dataset")
    if not (d / "data_batch_1.bin").is_file() and \
            (d / "cifar-10-batches-bin").is_dir():
        d = d / "cifar-10-batches-bin"
    train_files = sorted(d.glob("data_batch_*.bin"))
    test_file = d / "test_batch.bin"
    if not train_files:
        raise tools.UserException(
            f"no data_batch_*.bin files in {str(d)!r} -- expected the "
            "CIFAR-10 BINARY distribution (cifar-10-binary.tar.gz)")
    xs, ys = zip(*(_read_cifar_file(f) for f in train_files))
    train_x = np.concatenate(xs)
    train_y = np.concatenate(ys)
    if test_file.is_file():
        test_x, test_y = _read_cifar_file(test_file)
    else:
        test_x, test_y = train_x[:0], train_y[:0]
    return train_x, train_y, test_x, test_y


def load_tensor_folder(data_dir):
    """Load ``train*.npz`` / ``test*.npz`` shards with ``images`` (uint8
    [N, C, H, W]) and ``labels`` (int64 [N]) arrays."""
    d = check_dataset_dir(data_dir, what="tensor dataset")
    splits = {}
    for split in ("train", "test"):
        shards = sorted(d.glob(f"{split}*.npz"))
        if not shards and split == "train":
            raise tools.UserException(
                f"no train*.npz shards in {str(d)!r} -- expected npz shards "
                "with 'images' (uint8 NCHW) and 'labels' (int64) arrays")
        xs, ys = [], []
        for s in shards:
            with np.load(str(s)) as z:
                if "images" not in z or "labels" not in z:
                    raise tools.UserException(
                        f"{str(s)!r} lacks 'images'/'labels' arrays")
                xs.append(np.ascontiguousarray(z["images"]))
                ys.append(z["labels"].astype(np.int64))
        if xs:
            splits[split] = (np.concatenate(xs), np.concatenate(ys))
        else:
            splits[split] = None
    return splits


class RealDataset:
    """Deterministic per-(worker, step) batch server over loaded tensors.

    Batch ``(worker, step)`` is a pure function of the seed (the same
    contract as SyntheticClassification, so GARs / attacks / tests see an
    identical data-stream abstraction). On GPU the uint8 dataset is moved
    to HBM once and batches are device-side gathers: normalization
    (x/255 - mean)/std runs on-device on the selected batch only.
    """

    # Standard CIFAR-10 channel statistics (of the real distribution).
    CIFAR10_MEAN = (0.4914, 0.4822, 0.4465)
    CIFAR10_STD = (0.2470, 0.2435, 0.2616)

    def __init__(self, train, test, seed=1234, normalize=None):
        """
        Args:
          train/test: (images uint8 [N, ...], labels int64 [N]) numpy pairs
                      (test may be None).
          normalize:  optional (mean, std) per-channel tuples.
        """
        tx, ty = train
        self.train_x = torch.from_numpy(np.ascontiguousarray(tx))
        self.train_y = torch.from_numpy(np.ascontiguousarray(ty))
        if test is not None and len(test[0]):
            self.test_x = torch.from_numpy(np.ascontiguousarray(test[0]))
            self.test_y = torch.from_numpy(np.ascontiguousarray(test[1]))
        else:
            self.test_x = self.test_y = None
        self.seed = seed
        if normalize is not None:
            c = len(normalize[0])
            self.mean = torch.tensor(normalize[0]).view(1, c, 1, 1) * 255.0
            self.std = torch.tensor(normalize[1]).view(1, c, 1, 1) * 255.0
        else:
            self.mean = self.std = None
        self._resident = {}  # device -> (train_x, train_y) on that device

    @classmethod
    def cifar10(cls, data_dir, seed=1234):
        tx, ty, ex, ey = load_cifar10_binary(data_dir)
        return cls((tx, ty), (ex, ey), seed=seed,
                   normalize=(cls.CIFAR10_MEAN, cls.CIFAR10_STD))

    @classmethod
    def tensor_folder(cls, data_dir, seed=1234, normalize=None):
        splits = load_tensor_folder(data_dir)
        return cls(splits["train"], splits["test"], seed=seed,
                   normalize=normalize)

    def _on_device(self, device):
        key = str(device)
        if key not in self._resident:
            self._resident[key] = (self.train_x.to(device),
                                   self.train_y.to(device))
        return self._resident[key]

    def _format(self, x_u8, device):
        x = x_u8.to(device=torch.device(device), dtype=torch.float32)
        if self.mean is not None:
            x = (x - self.mean.to(x.device)) / self.std.to(x.device)
        else:
            x = x / 255.0
        return x

    def batch(self, batch_size, worker, step, device="cpu"):
        """Training batch: deterministic sample with replacement (matches
        the reference's shuffled-queue semantics statistically while staying
        a pure function of (seed, worker, step))."""
        device = torch.device(device)
        gen = torch.Generator().manual_seed(
            (self.seed * 1000003 + worker * 7919 + step * 104729) & 0x7FFFFFFF)
        idx = torch.randint(0, self.train_x.shape[0], (batch_size,),
                            generator=gen)
        x_all, y_all = (self._on_device(device)
                        if device.type == "cuda" else
                        (self.train_x, self.train_y))
        idx = idx.to(x_all.device)
        return self._format(x_all[idx], device), y_all[idx].to(device)

    def eval_batches(self, batch_size, device="cpu"):
        x, y = (self.test_x, self.test_y)
        if x is None:  # no test split: hold out the training tail
            n = max(1, self.train_x.shape[0] // 10)
            x, y = self.train_x[-n:], self.train_y[-n:]
        for i in range(0, x.shape[0], batch_size):
            yield (self._format(x[i:i + batch_size], device),
                   y[i:i + batch_size].to(device))
