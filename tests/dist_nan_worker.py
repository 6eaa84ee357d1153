"""Subprocess worker for the collective divergence-abort test.

One designated rank's loss goes non-finite at a fixed step; the Trainer's
collective abort (runner_lib.py) must make EVERY rank exit cleanly at the
same step instead of leaving the healthy ranks blocked in the next
all-gather until the process-group timeout (reference NaN abort:
/root/reference/runner.py:570-574, which was single-process and never had
this problem).
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from aggregathor_amd import experiments
from aggregathor_amd.graph import Engine
from aggregathor_amd.parallel import WorkerGroup
from aggregathor_amd.runner_lib import Trainer


def main():
    poison_rank = int(sys.argv[1])
    poison_step = int(sys.argv[2])

    exp = experiments.instantiate("mnist", ["batch-size:8"])
    group = WorkerGroup(4, device="cpu")
    eng = Engine(exp, "average", group,
                 learning_rate="fixed", learning_rate_args=["initial-rate:0.1"])

    if group.rank == poison_rank:
        orig = exp.loss

        def poisoned(model, batch):
            loss = orig(model, batch)
            if eng.global_step >= poison_step:
                return loss * float("nan")
            return loss

        exp.loss = poisoned

    trainer = Trainer(eng, max_step=50, checkpoint_dir="",
                      evaluation_delta=-1, evaluation_period=-1,
                      summary_delta=-1, summary_period=-1,
                      evaluation_file="-", summary_dir="-")
    t0 = time.monotonic()
    report = trainer.train()
    print(json.dumps({
        "rank": group.rank,
        "diverged": report["diverged"],
        "steps": report["steps"],
        "train_wall_s": time.monotonic() - t0,
    }), flush=True)


if __name__ == "__main__":
    main()
