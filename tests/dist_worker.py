"""Subprocess worker for the multi-process (gloo) distributed tests.

Runs a short training session as one rank of a world and dumps the final
parameters + losses for cross-rank / vs-single-process comparison.
"""

import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from aggregathor_amd import experiments
from aggregathor_amd.graph import Engine
from aggregathor_amd.parallel import WorkerGroup


def main():
    out_path = sys.argv[1]
    steps = int(sys.argv[2])
    aggregator = sys.argv[3]
    nbworkers = int(sys.argv[4])
    f = int(sys.argv[5])
    attack = sys.argv[6] if len(sys.argv) > 6 else ""

    exp = experiments.instantiate("mnist", ["batch-size:16"])
    group = WorkerGroup(nbworkers, device="cpu")
    eng = Engine(exp, aggregator, group, nbbyzwrks=f,
                 nb_real_byz=(f if attack else 0), attack=attack or None,
                 learning_rate="fixed", learning_rate_args=["initial-rate:0.1"])
    losses = [eng.step() for _ in range(steps)]
    flat = torch.cat([p.detach().reshape(-1) for p in eng.params])
    payload = {
        "rank": group.rank,
        "world": group.world,
        "losses": losses,
        "param_sum": float(flat.sum().item()),
        "param_digest": float(flat.abs().sum().item()),
    }
    torch.save({"meta": payload, "flat": flat}, out_path)
    print(json.dumps(payload))


if __name__ == "__main__":
    main()
