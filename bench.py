#!/usr/bin/env python3
"""Flagship benchmark: ResNet-50, n=8 workers, f=2, Krum (vs Bulyan/average).

Driver contract:
  python bench.py --gpus N --steps K --warmup W
runs the BASELINE.json headline config -- steps/sec of Byzantine-resilient
data-parallel ResNet-50 training with n=8 workers and f=2 -- on N GPUs of one
node (one rank per GPU over RCCL when launched via torch.distributed.run;
single process when N=1). The total worker count stays n=8 at every N (the
GAR's n/f are properties of the training run, not the GPU count), so per-GPU
work shrinks as N grows: STRONG scaling, fixed global batch.

BASELINE.json's metric is "Krum & Bulyan vs average": `--gar all` runs the
three GARs back-to-back in one process and prints one contract JSON line per
GAR (krum f=2, bulyan f=1 -- Bulyan requires n >= 4f+3, so f=2 would need
n >= 11, indivisible by the GPU counts; keeping n=8 makes the three numbers
directly comparable -- and average f=0). The default (no --gar flag) stays
the single headline line: krum, n=8, f=2.

Synthetic ImageNet-shaped data (3x224x224, 1000 classes), random-init
weights, bf16 autocast compute with fp32 gradients/aggregation/optimizer
(`--no-amp` for fp32 end-to-end).

Rank 0 prints exactly one JSON line per benchmarked GAR with the whole-job
steps/sec (max-over-ranks timing, barrier+synchronize bracketed).
"""

import argparse
import gc
import json
import os
import sys
import time

# MIOpen configuration, measured on MI355X (profiles/gar_kernels.md):
# FAST find mode falls back to workspace-starved col2im conv-backward
# (~22 ms/step extra); normal find + torch benchmark mode + NHWC-friendly
# kernels take the ResNet-50 n=8 step from 150 to 86 ms. The find cost is
# paid once per conv config during the UNTIMED warmup steps.
os.environ.pop("MIOPEN_FIND_MODE", None)
os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

from aggregathor_amd.parallel.graphstep import enable_graph_safe_conv

# hipGraph capture safety: the capture-time replay-purity self-check
# (graphstep.py docstring, profiles/graph_purity_bisect.md) gates the
# graph fast path; AGGREGATHOR_SAFE_SOLVERS=1 additionally excludes the
# capture-unsafe MIOpen solver family (small-shape configs only).
enable_graph_safe_conv()

import torch

torch.backends.cudnn.benchmark = True
if os.environ.get("AGGREGATHOR_DETERMINISTIC_CONV") == "1":
    # Excludes atomic-accumulation conv algorithms (kept as a debug switch).
    torch.backends.cudnn.deterministic = True


def bench_one(args, gar, f, device, rank, n_gpus):
    """Build an engine for one GAR config, run warmup + timed steps, return
    the contract result dict (or None on non-zero ranks)."""
    from aggregathor_amd import experiments
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup

    image_size = args.image_size if args.dataset == "imagenet" else 0
    exp = experiments.instantiate(
        f"{args.model}-{args.dataset}",
        [f"batch-size:{args.batch_size}", f"image-size:{image_size}",
         "eval-examples:0"])
    group = WorkerGroup(args.workers, device=device)
    amp = not args.no_amp
    lossy = None
    if args.lossy:
        from aggregathor_amd.attacks.lossy import LossyChannel
        lossy = LossyChannel(args.lossy)
    engine = Engine(
        exp, gar, group, nbbyzwrks=f, amp=amp,
        nb_real_byz=(f if args.attack else 0), attack=args.attack,
        optimizer="sgd", learning_rate="fixed", graph_warmup=1, lossy=lossy)

    def sync():
        if device.startswith("cuda"):
            torch.cuda.synchronize()
        group.barrier()

    for _ in range(args.warmup):
        engine.step()
    # hipGraph capture must never land in the timed region: run extra
    # (untimed) steps until the capture has engaged, whatever warmup the
    # driver chose.
    extra = 0
    while engine.use_graphs and (engine._graphstep is None
                                 or not engine._graphstep.ready) and extra < 8:
        engine.step()
        extra += 1
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.step(sync_loss=False)
    sync()
    elapsed = time.perf_counter() - t0
    elapsed = group.allreduce_max(elapsed)  # worst rank defines job time
    # Validity: one synchronous loss read after timing -- a NaN here means
    # the measured steps were doing garbage math.
    final_loss = engine.step()
    if final_loss != final_loss:
        print(f"[bench] WARNING: post-run loss is NaN (rank {rank})",
              file=sys.stderr)

    steps_per_sec = args.steps / elapsed
    global_batch = args.batch_size * args.workers
    result = None
    if rank == 0:
        result = {
            "metric": f"steps/sec {args.model} n={args.workers} f={f} "
                      f"{gar}",
            "value": steps_per_sec,
            "unit": "steps/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,  # reference publishes no in-tree numbers
            "dtype": "bf16" if engine.amp else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": global_batch,
                "seq_len": args.image_size,
                "parallelism": f"dp{n_gpus}",
                "gar": gar,
                "n_workers": args.workers,
                "f": f,
                "attack": args.attack or None,
                "images_per_sec": steps_per_sec * global_batch,
                "final_loss": final_loss,
            },
        }
    # Release this engine's HBM (graphs, activations, gradient matrix)
    # before the next GAR config builds its own.
    del engine, exp, lossy
    gc.collect()
    if device.startswith("cuda"):
        torch.cuda.empty_cache()
    return result


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--model", type=str, default="resnet50")
    ap.add_argument("--dataset", type=str, default="imagenet",
                    help="dataset shape: imagenet | cifar10")
    ap.add_argument("--workers", type=int, default=8, help="total GAR workers n")
    ap.add_argument("--f", type=int, default=2, help="declared Byzantine workers")
    ap.add_argument("--gar", type=str, default="krum",
                    help="aggregation rule (krum | bulyan | average | ... | "
                         "all = the BASELINE comparison: krum f=2, bulyan "
                         "f=1 [n >= 4f+3 bound at n=8], average f=0)")
    ap.add_argument("--batch-size", type=int, default=32, help="per-worker batch")
    ap.add_argument("--image-size", type=int, default=224)
    ap.add_argument("--attack", type=str, default="",
                    help="optional attack (e.g. reversal) mounted by f workers")
    ap.add_argument("--lossy", nargs="*", default=None,
                    help="UDP-style lossy-channel injection key:value args")
    ap.add_argument("--device", type=str, default="",
                    help="override device (debug; cpu allowed)")
    ap.add_argument("--no-amp", action="store_true",
                    help="disable bf16 autocast (fp32 compute)")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)

    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}"
    else:
        print("[bench] WARNING: no GPU visible, running on CPU (debug only)",
              file=sys.stderr)
        device = "cpu"

    from aggregathor_amd import ops
    if device.startswith("cuda") and not ops.hip_available():
        raise RuntimeError("HIP extension _gar_hip not built -- run "
                           "`python -m aggregathor_amd.ops.build` first")

    if args.gar == "all":
        # The BASELINE.json comparison, one driver command: same n, same
        # per-step work, three GARs (Bulyan at f=1: n >= 4f+3 requires
        # n >= 11 for f=2, which no GPU count divides; see module docstring).
        configs = [("krum", args.f),
                   ("bulyan", min(args.f, max((args.workers - 3) // 4, 0))),
                   ("average", 0)]
    else:
        configs = [(args.gar, args.f)]

    for gar, f in configs:
        result = bench_one(args, gar, f, device, rank, n_gpus)
        if result is not None:
            print(json.dumps(result), flush=True)


if __name__ == "__main__":
    main()
