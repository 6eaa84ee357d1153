"""Real-dataset input pipeline (experiments/datasets.py): CIFAR-10 binary
format, npz tensor folders, permission checks, deterministic batch serving,
and end-to-end training through the experiments' ``data-dir:`` path
(reference: /root/reference/experiments/cnnet.py:115-146, 187-196)."""

import os

import numpy as np
import pytest
import torch

from aggregathor_amd import experiments, tools
from aggregathor_amd.experiments.datasets import (
    RealDataset, check_dataset_dir, load_cifar10_binary, load_tensor_folder)


def _write_cifar_binary(tmp_path, n_train_files=2, n_per_file=50, seed=7):
    """Write format-conformant CIFAR-10 binary files with known content."""
    rng = np.random.default_rng(seed)
    d = tmp_path / "cifar-10-batches-bin"
    d.mkdir()
    all_labels, all_images = [], []
    for i in range(1, n_train_files + 1):
        labels = rng.integers(0, 10, n_per_file, dtype=np.uint8)
        images = rng.integers(0, 256, (n_per_file, 3072), dtype=np.uint8)
        rec = np.concatenate([labels[:, None], images], axis=1)
        rec.tofile(str(d / f"data_batch_{i}.bin"))
        all_labels.append(labels)
        all_images.append(images)
    tl = rng.integers(0, 10, n_per_file, dtype=np.uint8)
    ti = rng.integers(0, 256, (n_per_file, 3072), dtype=np.uint8)
    np.concatenate([tl[:, None], ti], axis=1).tofile(str(d / "test_batch.bin"))
    return d, (np.concatenate(all_images), np.concatenate(all_labels)), (ti, tl)


def test_cifar10_binary_roundtrip(tmp_path):
    d, (imgs, labels), (ti, tl) = _write_cifar_binary(tmp_path)
    tx, ty, ex, ey = load_cifar10_binary(d)
    assert tx.shape == (100, 3, 32, 32) and tx.dtype == np.uint8
    assert (ty == labels).all()
    assert (tx.reshape(100, 3072) == imgs).all()
    assert ex.shape == (50, 3, 32, 32) and (ey == tl).all()
    # The parent dir (containing cifar-10-batches-bin/) also works.
    tx2, _, _, _ = load_cifar10_binary(tmp_path)
    assert (tx2 == tx).all()


def test_cifar10_binary_bad_format(tmp_path):
    d = tmp_path / "ds"
    d.mkdir()
    (d / "data_batch_1.bin").write_bytes(b"\x00" * 1000)  # not a 3073 multiple
    with pytest.raises(tools.UserException, match="binary format"):
        load_cifar10_binary(d)
    with pytest.raises(tools.UserException, match="directory"):
        load_cifar10_binary(tmp_path / "missing")


def test_dataset_dir_permission_check(tmp_path):
    d = tmp_path / "locked"
    d.mkdir()
    (d / "data_batch_1.bin").write_bytes(b"\x00" * 3073)
    os.chmod(d / "data_batch_1.bin", 0)
    if os.geteuid() == 0:  # root bypasses permission bits
        pytest.skip("permission bits not enforced for root")
    with pytest.raises(tools.UserException, match="read-able"):
        check_dataset_dir(d)


def test_tensor_folder(tmp_path):
    d = tmp_path / "inet"
    d.mkdir()
    rng = np.random.default_rng(3)
    for i in range(2):
        np.savez(str(d / f"train_{i}.npz"),
                 images=rng.integers(0, 256, (10, 3, 8, 8), dtype=np.uint8),
                 labels=rng.integers(0, 5, 10, dtype=np.int64))
    np.savez(str(d / "test_0.npz"),
             images=rng.integers(0, 256, (4, 3, 8, 8), dtype=np.uint8),
             labels=rng.integers(0, 5, 4, dtype=np.int64))
    splits = load_tensor_folder(d)
    assert splits["train"][0].shape == (20, 3, 8, 8)
    assert splits["test"][0].shape == (4, 3, 8, 8)
    with pytest.raises(tools.UserException, match="train"):
        e = tmp_path / "empty"
        e.mkdir()
        load_tensor_folder(e)


def test_real_dataset_deterministic_batches(tmp_path):
    d, _, _ = _write_cifar_binary(tmp_path)
    ds_a = RealDataset.cifar10(d, seed=42)
    ds_b = RealDataset.cifar10(d, seed=42)
    xa, ya = ds_a.batch(8, worker=1, step=3)
    xb, yb = ds_b.batch(8, worker=1, step=3)
    assert torch.equal(xa, xb) and torch.equal(ya, yb)
    xc, _ = ds_a.batch(8, worker=1, step=4)
    assert not torch.equal(xa, xc)
    # Normalized: roughly centered, not raw [0, 255].
    assert xa.abs().max() < 10.0
    evs = list(ds_a.eval_batches(32))
    assert sum(x.shape[0] for x, _ in evs) == 50


def test_cnnet_experiment_real_data_path(tmp_path):
    d, _, _ = _write_cifar_binary(tmp_path)
    exp = experiments.instantiate(
        "cnnet", [f"data-dir:{d}", "batch-size:4"])
    x, y = exp.train_batch(0, 0, "cpu")
    assert x.shape == (4, 3, 32, 32) and y.shape == (4,)
    # A training step runs end to end on the real-data path.
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    eng = Engine(exp, "krum", WorkerGroup(4), nbbyzwrks=1)
    loss = eng.step()
    assert loss == loss


def test_resnet_experiment_real_data_path(tmp_path):
    d, _, _ = _write_cifar_binary(tmp_path)
    exp = experiments.instantiate(
        "resnet18-cifar10", [f"data-dir:{d}", "batch-size:2"])
    x, y = exp.train_batch(0, 0, "cpu")
    assert x.shape == (2, 3, 32, 32)
    exp2 = experiments.instantiate("resnet18-imagenet", ["batch-size:2"])
    x2, _ = exp2.train_batch(0, 0, "cpu")
    assert x2.shape == (2, 3, 224, 224)


def test_data_pool_arg_controls_gpu_pooling():
    from aggregathor_amd.experiments.data import SyntheticClassification
    s = SyntheticClassification((8,), 2, pool_size=0)
    assert s.pool_size == 0
    # CPU path: always unpooled and a pure function of (worker, step).
    x1, _ = s.batch(4, 0, 123)
    x2, _ = s.batch(4, 0, 123)
    assert torch.equal(x1, x2)
    exp = experiments.instantiate("mnist", ["data-pool:0", "batch-size:4"])
    assert exp._synth.pool_size == 0
