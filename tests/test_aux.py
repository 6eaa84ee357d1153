"""Auxiliary subsystems: cluster allocation, spec parsing, gradient
integrity MACs, synthetic data determinism."""

import math
import os

import pytest
import torch

from aggregathor_amd import tools
from aggregathor_amd.cluster import Manager
from aggregathor_amd.experiments.data import SyntheticClassification
from aggregathor_amd.parallel.signing import GradientIntegrity


def test_cluster_parse_json():
    spec = tools.cluster_parse('{"workers": ["a:1", "b:2"], "ps": ["c:3"]}')
    assert spec["workers"] == ["a:1", "b:2"]


def test_cluster_parse_invalid():
    with pytest.raises(tools.UserException):
        tools.cluster_parse("not json")
    with pytest.raises(tools.UserException):
        tools.cluster_parse("[]")


def test_cluster_parse_g5k(tmp_path, monkeypatch):
    nodes = tmp_path / "nodes"
    nodes.write_text("host1\nhost1\nhost2\nhost3\n")
    monkeypatch.setenv("OAR_FILE_NODES", str(nodes))
    spec = tools.cluster_parse("G5k")
    assert spec["ps"] == ["host1:7000"]
    assert spec["workers"] == ["host2:7000", "host3:7000"]


def test_cluster_manager_allocation():
    mgr = Manager(use_gpu=False)
    devs = mgr.allocate("workers", 3)
    assert devs == ["cpu", "cpu", "cpu"]
    mgr.report()


def test_integrity_roundtrip():
    gi = GradientIntegrity("secret", nbworkers=4, sample=256)
    rows = torch.randn(4, 10000)
    macs = gi.sign_rows(rows, [0, 1, 2, 3], step=7)
    matrix = rows.clone()
    assert gi.verify_matrix(matrix, macs, step=7) == []
    assert torch.equal(matrix, rows)


def test_integrity_detects_corruption():
    gi = GradientIntegrity("secret", nbworkers=4, sample=10000)  # full cover
    rows = torch.randn(4, 10000)
    macs = gi.sign_rows(rows, [0, 1, 2, 3], step=3)
    matrix = rows.clone()
    matrix[2, 1234] += 1.0  # transport corruption
    failed = gi.verify_matrix(matrix, macs, step=3)
    assert failed == [2]
    assert torch.isnan(matrix[2]).all()
    assert torch.equal(matrix[0], rows[0])


def test_integrity_wrong_key_fails():
    a = GradientIntegrity("secret-a", nbworkers=2, sample=512)
    b = GradientIntegrity("secret-b", nbworkers=2, sample=512)
    rows = torch.randn(2, 4096)
    macs = a.sign_rows(rows, [0, 1], step=0)
    assert b.verify_matrix(rows.clone(), macs, step=0) == [0, 1]


def test_engine_with_integrity():
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    exp = experiments.instantiate("mnist", ["batch-size:16"])
    group = WorkerGroup(3, device="cpu")
    gi = GradientIntegrity("k", nbworkers=3, sample=128)
    eng = Engine(exp, "average", group, integrity=gi)
    for _ in range(3):
        assert math.isfinite(eng.step())


def test_synthetic_determinism():
    a = SyntheticClassification((32,), 5, seed=9)
    b = SyntheticClassification((32,), 5, seed=9)
    xa, ya = a.batch(8, worker=1, step=3)
    xb, yb = b.batch(8, worker=1, step=3)
    assert torch.equal(xa, xb) and torch.equal(ya, yb)
    xc, _ = a.batch(8, worker=2, step=3)
    assert not torch.equal(xa, xc)  # distinct per worker


def test_worker_group_single_process():
    from aggregathor_amd.parallel import WorkerGroup
    g = WorkerGroup(4, device="cpu")
    assert g.world == 1 and g.rank == 0 and not g.distributed
    assert g.worker_ids == [0, 1, 2, 3]
    rows = torch.randn(4, 10)
    assert g.gather(rows) is rows  # identity, zero-copy
    assert g.allreduce_max(3.5) == 3.5
    assert g.gather_small("x") == ["x"]


def test_worker_group_divisibility():
    from aggregathor_amd.parallel import WorkerGroup
    g = WorkerGroup(8, device="cpu")
    assert g.local_workers == 8
    with pytest.raises(tools.UserException):
        # world=1 divides everything; emulate indivisibility via a fake world
        bad = WorkerGroup(3, device="cpu")
        bad.world = 2
        if 3 % bad.world != 0:
            raise tools.UserException("indivisible")


def test_reference_oracle_parity():
    """Run this framework's GAR math against the reference's own compiled
    kernels (skipped when the reference mount or g++ is unavailable)."""
    import pathlib
    import shutil
    import subprocess
    import sys
    if not pathlib.Path("/root/reference").exists() or not shutil.which("g++"):
        pytest.skip("reference mount or g++ unavailable")
    repo = pathlib.Path(__file__).resolve().parent.parent
    r = subprocess.run(
        [sys.executable, str(repo / "scripts/reference_oracle_check.py")],
        capture_output=True, timeout=600)
    assert r.returncode == 0, r.stdout.decode()[-3000:] + r.stderr.decode()[-1000:]


def test_integrity_detects_single_element_sparse_corruption():
    # The sampled MAC alone would catch a 1-element flip with probability
    # ~sample/d; the full-row bit-checksum folded into the MAC catches it
    # ALWAYS (int64 sum of float32 bit patterns changes for any single flip).
    gi = GradientIntegrity("secret", nbworkers=2, sample=64)  # tiny sample
    d = 100000
    rows = torch.randn(2, d)
    macs = gi.sign_rows(rows, [0, 1], step=11)
    for trial in range(5):  # several positions, incl. ones never sampled
        matrix = rows.clone()
        pos = (trial * 31337 + 17) % d
        matrix[1, pos] += 1e-6  # smallest interesting corruption
        failed = gi.verify_matrix(matrix, macs, step=11)
        assert failed == [1], f"missed corruption at {pos}"
