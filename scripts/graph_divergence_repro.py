#!/usr/bin/env python3
"""Bisection harness for the OPEN hipGraph divergence issue (NOTES.md).

Reproduces: resnet50-cifar10, n=8, lr 0.01, captured step -> loss goes
non-finite around step 150-250, while the eager step trains cleanly on the
identical config/seed. Runs a config matrix and reports the first
non-finite step plus the max |param| trajectory, to localize the corruption
(graphs on/off x eval service on/off x l2 on/off x find-mode).

Usage (on a GPU box):
  python scripts/graph_divergence_repro.py [--steps 400] [--case all]

ROOT CAUSE FOUND (round 2): MIOpen's ConvHipImplicitGemmGroup*Xdlops
solvers are not replay-pure under hipGraph capture (see
scripts/graph_purity_bisect.py and profiles/graph_purity_bisect.md).
The engine now runs a replay-purity self-check at capture and falls back
to eager on failure, so REPRODUCING the historical divergence requires
AGGREGATHOR_NO_PURITY_CHECK=1.
"""

import argparse
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

import torch

torch.backends.cudnn.benchmark = True
if os.environ.get("AGGREGATHOR_DETERMINISTIC_CONV") == "1":
    # Candidate fix for the capture-unsafe-solver hypothesis: deterministic
    # mode excludes atomic-accumulation conv algorithms.
    torch.backends.cudnn.deterministic = True


def run_case(name, graphs, eval_every, l2, steps, seed=1234):
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    exp = experiments.instantiate(
        "resnet50-cifar10", ["batch-size:32", "eval-examples:512"])
    eng = Engine(exp, "average", WorkerGroup(8, device="cuda:0"),
                 amp=True, seed=seed, use_graphs=graphs, graph_warmup=1,
                 l2_regularize=l2,
                 learning_rate_args=["initial-rate:0.01"])
    first_bad = None
    for i in range(steps):
        loss = eng.step()
        if not math.isfinite(loss) and first_bad is None:
            first_bad = i
            break
        if eval_every and i > 0 and i % eval_every == 0:
            acc = eng.evaluate()["top1-X-acc"]
            print(f"  [{name}] step {i}: loss={loss:.4f} acc={acc:.3f}",
                  flush=True)
        elif i % 100 == 0:
            mx = max(float(p.abs().max()) for p in eng.params)
            print(f"  [{name}] step {i}: loss={loss:.4f} max|p|={mx:.3f}",
                  flush=True)
    print(f"[{name}] graphs={graphs} eval_every={eval_every} l2={l2}: "
          f"{'DIVERGED at step %d' % first_bad if first_bad is not None else 'healthy'}"
          f" (ran {min(steps, first_bad or steps)} steps)", flush=True)
    return first_bad


CASES = {
    "graphs":           dict(graphs=True, eval_every=0, l2=-1.0),
    "graphs+eval":      dict(graphs=True, eval_every=150, l2=-1.0),
    "graphs+l2":        dict(graphs=True, eval_every=0, l2=0.001),
    "graphs+eval+l2":   dict(graphs=True, eval_every=150, l2=0.001),
    "eager+eval+l2":    dict(graphs=False, eval_every=150, l2=0.001),
    "eager":            dict(graphs=False, eval_every=0, l2=-1.0),
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=400)
    ap.add_argument("--case", type=str, default="all",
                    help="one of %s, 'purity', or 'all'" % ", ".join(CASES))
    args = ap.parse_args()
    if args.case == "purity":
        replay_purity_check()
        return
    names = list(CASES) if args.case == "all" else [args.case]
    results = {}
    for name in names:
        results[name] = run_case(name, steps=args.steps, **CASES[name])
    print("\n=== summary ===")
    for name, bad in results.items():
        print(f"{name:18s} {'diverged@%d' % bad if bad is not None else 'healthy'}")


def replay_purity_check(steps_before=5):
    """Decisive test for the stateful-kernel hypothesis: freeze the inputs
    and replay the captured local phase twice -- the produced gradient rows
    must be BITWISE identical. A drift implicates a capture-unsafe kernel
    (e.g. an MIOpen wrw solver accumulating into a lazily-zeroed workspace).
    Candidate fix to A/B if drift is found: torch.backends.cudnn
    .deterministic = True (excludes atomic-accumulation conv algorithms).
    """
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    exp = experiments.instantiate(
        "resnet50-cifar10", ["batch-size:32", "eval-examples:0"])
    eng = Engine(exp, "average", WorkerGroup(8, device="cuda:0"),
                 amp=True, use_graphs=True, graph_warmup=1,
                 learning_rate_args=["initial-rate:0.0"])  # lr 0: params frozen
    for _ in range(steps_before):
        eng.step()
    gs = eng._graphstep
    assert gs is not None and gs.ready, "graphs did not engage"
    # Stage once, replay the LOCAL graph twice on identical state.
    gs._stage_batches()
    gs.graph_local.replay()
    torch.cuda.synchronize()
    first = eng.local_rows.clone()
    gs.graph_local.replay()
    torch.cuda.synchronize()
    drift = (eng.local_rows - first).abs().max().item()
    same = torch.equal(eng.local_rows, first)
    print(f"replay purity: bitwise_equal={same} max_drift={drift:.3e}")
    return same


if __name__ == "__main__":
    main()
