#!/usr/bin/env python3
"""Accuracy-under-attack evaluation: each GAR vs each attack.

Produces the reference's headline robustness artifact (top-1 accuracy under
Byzantine attack): for every (GAR, attack) pair, train the
MNIST MLP with n workers of which f_real are Byzantine, and report final
top-1 accuracy. Writes a markdown table + JSON to stdout / --out.

Usage: python scripts/attack_eval.py [--steps 300] [--out results.md]
"""

import argparse
import json
import math
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def run_config(gar, n, f, attack=None, attack_args=None, exp="mnist",
               exp_args=None, lossy=None, steps=300, device="cpu"):
    from aggregathor_amd import experiments
    from aggregathor_amd.attacks.lossy import LossyChannel
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    experiment = experiments.instantiate(
        exp, exp_args or ["batch-size:64"])
    group = WorkerGroup(n, device=device)
    eng = Engine(experiment, gar, group, nbbyzwrks=f,
                 nb_real_byz=(f if attack else 0), attack=attack,
                 attack_args=attack_args or [],
                 lossy=LossyChannel(lossy) if lossy else None,
                 learning_rate="fixed",
                 learning_rate_args=["initial-rate:0.3"],
                 amp=device.startswith("cuda"))
    loss = float("nan")
    for _ in range(steps):
        loss = eng.step()
        if not math.isfinite(loss):
            break
    acc = eng.evaluate()["top1-X-acc"]
    return {"gar": gar, "n": n, "f": f, "attack": attack or "-",
            "final_loss": loss if math.isfinite(loss) else "diverged",
            "top1_acc": round(acc, 4)}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--device", type=str,
                    default="cuda:0" if torch.cuda.is_available() else "cpu")
    ap.add_argument("--out", type=str, default="")
    args = ap.parse_args()

    configs = []
    # No attack: all GARs should learn.
    for gar in ("average", "krum", "median", "averaged-median", "bulyan"):
        n = 11 if gar == "bulyan" else 8
        configs.append(dict(gar=gar, n=n, f=2))
    # Gradient-reversal attack (BASELINE.json's named attack), f=2 real.
    for gar in ("average", "krum", "median", "averaged-median", "bulyan"):
        n = 11 if gar == "bulyan" else 8
        configs.append(dict(gar=gar, n=n, f=2, attack="reversal",
                            attack_args=["factor:10.0"]))
    # Unbounded magnitude attack.
    for gar in ("average", "krum", "bulyan"):
        n = 11 if gar == "bulyan" else 8
        configs.append(dict(gar=gar, n=n, f=2, attack="magnitude",
                            attack_args=["factor:1e6"]))
    # Omniscient attacks: ALIE (mean + z*std) and IPM (-eps * mean).
    for gar in ("average", "krum", "averaged-median"):
        configs.append(dict(gar=gar, n=8, f=2, attack="alie",
                            attack_args=["z:1.5"]))
        configs.append(dict(gar=gar, n=8, f=2, attack="ipm",
                            attack_args=["eps:1.0"]))
    # Data poisoning (mnistAttack severity 2, worker 0).
    for gar in ("average", "krum"):
        configs.append(dict(gar=gar, n=5, f=1, exp="mnistAttack",
                            exp_args=["batch-size:64", "malformed-severity:2"]))
    # UDP-style lossy channel + NaN-tolerant GARs. averaged-median needs
    # f > 0 under loss: beta = n - f must exclude the NaN-filled values
    # (beta = n would average them in, exactly as in the reference).
    configs.append(dict(gar="average-nan", n=8, f=0,
                        lossy=["drop-rate:0.1", "workers:0,1"]))
    configs.append(dict(gar="averaged-median", n=8, f=2,
                        lossy=["drop-rate:0.1", "workers:0,1"]))

    rows = []
    for cfg in configs:
        res = run_config(steps=args.steps, device=args.device, **cfg)
        rows.append(res)
        print(f"[attack_eval] {res}", file=sys.stderr)

    lines = ["| GAR | n | f | attack | top-1 acc | final loss |",
             "|---|---|---|---|---|---|"]
    for r in rows:
        lines.append(f"| {r['gar']} | {r['n']} | {r['f']} | {r['attack']} | "
                     f"{r['top1_acc']} | {r['final_loss']} |")
    table = "\n".join(lines)
    print(table)
    print(json.dumps(rows))
    if args.out:
        with open(args.out, "w") as f:
            f.write(table + "\n\n```json\n" + json.dumps(rows, indent=1)
                    + "\n```\n")


if __name__ == "__main__":
    main()
