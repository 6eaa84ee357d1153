"""Multi-process distributed path over gloo (world_size 2, CPU).

Verifies the RCCL-replacing layout end to end: all-gather of worker
gradient rows, replicated GAR, deterministic apply -- every rank must end
with BIT-IDENTICAL parameters, equal to the single-process (loopback) run.
"""

import os
import pathlib
import subprocess
import sys

import pytest
import torch

REPO = pathlib.Path(__file__).resolve().parent.parent
WORKER = REPO / "tests" / "dist_worker.py"


def _run_world(tmp_path, world, steps=6, aggregator="krum", n=4, f=1,
               attack="", port=29600):
    procs = []
    outs = []
    for rank in range(world):
        out = tmp_path / f"rank{rank}.pt"
        outs.append(out)
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(WORKER), str(out), str(steps), aggregator,
             str(n), str(f), attack],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        stdout, stderr = p.communicate(timeout=300)
        assert p.returncode == 0, f"worker failed:\n{stderr.decode()[-2000:]}"
    return [torch.load(o, weights_only=False) for o in outs]


def _run_single(tmp_path, steps=6, aggregator="krum", n=4, f=1, attack=""):
    out = tmp_path / "single.pt"
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK",
                        "MASTER_ADDR", "MASTER_PORT")}
    p = subprocess.run(
        [sys.executable, str(WORKER), str(out), str(steps), aggregator,
         str(n), str(f), attack],
        env=env, capture_output=True, timeout=300)
    assert p.returncode == 0, f"single failed:\n{p.stderr.decode()[-2000:]}"
    return torch.load(out, weights_only=False)


@pytest.mark.parametrize("aggregator", ["average", "krum", "median"])
def test_world2_matches_single_process(tmp_path, aggregator):
    port = 29610 + hash(aggregator) % 50
    single = _run_single(tmp_path, aggregator=aggregator)
    r0, r1 = _run_world(tmp_path, 2, aggregator=aggregator, port=port)
    # Every rank ends bit-identical (replicated deterministic GAR + apply).
    assert torch.equal(r0["flat"], r1["flat"])
    # And the distributed run equals the loopback single-process run.
    assert torch.equal(r0["flat"], single["flat"])
    assert r0["meta"]["world"] == 2 and single["meta"]["world"] == 1


def test_world2_under_attack(tmp_path):
    # 1 real Byzantine worker mounting reversal; krum f=1 converges and all
    # ranks stay in lockstep.
    r0, r1 = _run_world(tmp_path, 2, steps=8, aggregator="krum", n=4, f=1,
                        attack="reversal", port=29661)
    assert torch.equal(r0["flat"], r1["flat"])
    assert all(l == l for l in r0["meta"]["losses"])  # finite


def test_world4_bulyan(tmp_path):
    # Bulyan needs n >= 4f+3: n=8 f=1 over 4 ranks (2 virtual workers each).
    outs = _run_world(tmp_path, 4, steps=4, aggregator="bulyan", n=8, f=1,
                      port=29671)
    base = outs[0]["flat"]
    for o in outs[1:]:
        assert torch.equal(base, o["flat"])


NAN_WORKER = REPO / "tests" / "dist_nan_worker.py"


def test_world2_collective_nan_abort(tmp_path):
    # Rank 1's loss goes non-finite at step 3: BOTH ranks must exit cleanly
    # at the same step (collective abort), not block in the next all-gather
    # until the 300 s process-group timeout.
    import json
    import time
    world = 2
    procs = []
    t0 = time.monotonic()
    for rank in range(world):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29685",
            "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
        })
        procs.append(subprocess.Popen(
            [sys.executable, str(NAN_WORKER), "1", "3"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    metas = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=120)
        assert p.returncode == 0, f"worker failed:\n{stderr.decode()[-2000:]}"
        metas.append(json.loads(stdout.decode().strip().splitlines()[-1]))
    wall = time.monotonic() - t0
    assert all(m["diverged"] for m in metas)
    # Same abort step on every rank (steps 0..2 healthy, abort during step 3).
    assert metas[0]["steps"] == metas[1]["steps"] == 4
    # Clean exit well under the process-group timeout.
    assert wall < 60, f"abort took {wall:.1f}s -- ranks likely hung in gather"


def test_world2_bucketed_gather_bit_identical(tmp_path):
    # Opt-in bucketed gather overlap (parallel/overlap.py): world-2, one
    # worker per rank, small buckets (many collectives per step). The
    # final parameters must be BITWISE identical to the serial-gather
    # world-2 run and to the single-process run.
    env_extra = {"AGGREGATHOR_BUCKET_MB": "0.05"}
    single = _run_single(tmp_path, steps=6, aggregator="average", n=2, f=0)
    r0, r1 = _run_world(tmp_path, 2, steps=6, aggregator="average", n=2,
                        f=0, port=29690)
    # Re-run world-2 WITH bucketed gather.
    procs, outs = [], []
    for rank in range(2):
        out = tmp_path / f"bucket_rank{rank}.pt"
        outs.append(out)
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "WORLD_SIZE": "2", "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29691",
            "GLOO_SOCKET_IFNAME": env.get("GLOO_SOCKET_IFNAME", "lo"),
        })
        env.update(env_extra)
        procs.append(subprocess.Popen(
            [sys.executable, str(WORKER), str(out), "6", "average", "2",
             "0", ""],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        stdout, stderr = p.communicate(timeout=300)
        assert p.returncode == 0, f"worker failed:\n{stderr.decode()[-2000:]}"
    b0 = torch.load(outs[0], weights_only=False)
    b1 = torch.load(outs[1], weights_only=False)
    assert torch.equal(b0["flat"], b1["flat"])
    assert torch.equal(b0["flat"], r0["flat"])
    assert torch.equal(b0["flat"], single["flat"])


def test_bucketed_gather_unit_world1():
    # Bucket layout + flush-at-finish semantics at world 1 (gloo identity
    # gather): finish() without any hook having fired must still gather
    # every bucket, and the scattered matrix must equal the row bitwise.
    import torch.distributed as dist
    import torch.nn as nn
    from aggregathor_amd.graph import bind_grad_views, flat_size
    from aggregathor_amd.parallel.overlap import BucketedGather
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29692")
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        class _G:
            distributed = True
            local_workers = 1
        model = nn.Sequential(nn.Linear(17, 33), nn.ReLU(),
                              nn.Linear(33, 5))
        params = list(model.parameters())
        d = flat_size(params)
        row = torch.zeros((1, d))
        matrix = torch.empty((1, d))
        bg = BucketedGather(_G(), params, row, matrix, bucket_bytes=64)
        assert len(bg.spans) >= 3  # tiny buckets -> several spans
        assert bg.spans[0][1] == d and bg.spans[-1][0] == 0  # full cover
        # Path 1: flush-only (no hooks fired).
        bg.begin_step()
        row.uniform_(-1, 1)
        out = bg.finish()
        assert torch.equal(out[0], row[0])
        # Path 2: hooks fire through a real backward.
        bg.begin_step()
        row.zero_()
        bind_grad_views(params, row[0])
        loss = model(torch.randn(4, 17)).square().mean()
        loss.backward()
        out = bg.finish()
        assert torch.equal(out[0], row[0])
        assert out.abs().sum() > 0
        bg.remove()
    finally:
        dist.destroy_process_group()


def test_bucketed_gather_gating(monkeypatch):
    # The overlap must NOT engage outside its v1 scope: single-process
    # runs, multiple local workers, or real-Byzantine attacks.
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    monkeypatch.setenv("AGGREGATHOR_BUCKET_MB", "1")
    exp = experiments.instantiate("mnist", ["batch-size:8"])
    # Single process: no distributed group -> no overlap.
    eng = Engine(exp, "average", WorkerGroup(4))
    assert eng.overlap is None
    assert eng.step() == eng.last_loss  # and training still works
