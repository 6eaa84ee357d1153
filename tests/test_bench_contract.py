"""bench.py driver contract: flags, JSON schema, torchrun compatibility."""

import json
import os
import pathlib
import subprocess
import sys

REPO = pathlib.Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
                 "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                 "dtype", "data", "config"}


def _clean_env():
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK",
                        "MASTER_ADDR", "MASTER_PORT")}
    env["OMP_NUM_THREADS"] = "4"  # avoid CPU thrash with multiple ranks
    return env


def test_bench_json_contract():
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "resnet20", "--dataset",
         "cifar10", "--steps", "2", "--warmup", "1", "--batch-size", "2",
         "--device", "cpu"],
        capture_output=True, timeout=600, env=_clean_env(), cwd=str(REPO))
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    lines = [l for l in r.stdout.decode().splitlines() if l.startswith("{")]
    assert len(lines) == 1, "bench must print exactly one JSON line"
    d = json.loads(lines[0])
    assert REQUIRED_KEYS <= set(d), REQUIRED_KEYS - set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "strong"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert abs(d["value"] * d["ms_per_step"] / 1e3 - 1.0) < 1e-6
    cfg = d["config"]
    for k in ("model", "global_batch", "parallelism", "gar", "n_workers", "f"):
        assert k in cfg, k
    assert cfg["model"] == "resnet20" and cfg["n_workers"] == 8 and cfg["f"] == 2
    assert cfg["parallelism"] == "dp1"


def test_bench_under_torchrun_world2():
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29911", "bench.py", "--model", "resnet20",
         "--dataset", "cifar10", "--gpus", "2", "--steps", "2", "--warmup",
         "1", "--batch-size", "2", "--device", "cpu"],
        capture_output=True, timeout=900, env=_clean_env(), cwd=str(REPO))
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    lines = [l for l in r.stdout.decode().splitlines() if l.startswith("{")]
    assert len(lines) == 1  # rank 0 only
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["n_workers"] == 8  # strong scaling: n fixed


def test_bench_gar_all_emits_three_lines():
    # BASELINE.json's "Krum & Bulyan vs average" in one driver command:
    # one contract JSON line per GAR, same n, documented bulyan f bound.
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "resnet20", "--dataset",
         "cifar10", "--steps", "2", "--warmup", "1", "--batch-size", "2",
         "--device", "cpu", "--gar", "all"],
        capture_output=True, timeout=900, env=_clean_env(), cwd=str(REPO))
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    lines = [l for l in r.stdout.decode().splitlines() if l.startswith("{")]
    assert len(lines) == 3
    rows = [json.loads(l) for l in lines]
    assert [x["config"]["gar"] for x in rows] == ["krum", "bulyan", "average"]
    assert [x["config"]["f"] for x in rows] == [2, 1, 0]
    for x in rows:
        assert REQUIRED_KEYS <= set(x)
        assert x["config"]["n_workers"] == 8
