#!/usr/bin/env python3
"""Start/continue a Byzantine-resilient distributed training session.

MI355X-native re-expression of the reference's runner.py CLI
(/root/reference/runner.py:80-231): same flag surface, same plugin
key:value sub-arguments, same checkpoint/eval-file layout. The cluster
model changes: instead of a TF parameter server + gRPC workers, training
runs as one process per GPU over RCCL/xGMI (launched by deploy.py or
torch.distributed.run), or as a single process hosting all n virtual
workers (the moral equivalent of the reference's loopback cluster,
README.md:146).

Flag mapping notes:
- --server/--client are accepted for compatibility: a single process without
  torch.distributed env vars always acts as the full (loopback) cluster; with
  RANK/WORLD_SIZE set it joins the process group.
- --use-gpu selects cuda:LOCAL_RANK; --reuse-gpu is implied (virtual workers
  share their rank's GPU by design).
- --MPI is obsolete (RCCL over xGMI replaces grpc+mpi) and ignored with a
  warning.
"""

import argparse
import os
import sys

# MIOpen conv-algo selection: see bench.py (same measured rationale).
os.environ.setdefault("PYTORCH_MIOPEN_SUGGEST_NHWC", "1")

from aggregathor_amd import config, tools

# ---------------------------------------------------------------------------- #

parser = argparse.ArgumentParser(
    description="Start/continue a distributed training session.",
    formatter_class=argparse.RawTextHelpFormatter)
parser.add_argument("--client", type=str, default="",
                    help="Rendezvous address host:port to connect to (compat; "
                         "requires RANK/WORLD_SIZE in the environment)")
parser.add_argument("--server", type=str, default="",
                    help="Cluster specification (compat; single-process "
                         "loopback mode hosts every worker locally)")
parser.add_argument("--ps-job-name", type=str, default=config.default_ps_job_name)
parser.add_argument("--ev-job-name", type=str, default=config.default_ev_job_name)
parser.add_argument("--wk-job-name", type=str, default=config.default_wk_job_name)
parser.add_argument("--experiment", type=str, required=True,
                    help="Experiment to run")
parser.add_argument("--experiment-args", nargs="*",
                    help="Additional key:value arguments for the experiment")
parser.add_argument("--aggregator", type=str, required=True,
                    help="Gradient aggregation rule to use")
parser.add_argument("--aggregator-args", nargs="*",
                    help="Additional key:value arguments for the GAR")
parser.add_argument("--optimizer", type=str, default="sgd")
parser.add_argument("--optimizer-args", nargs="*")
parser.add_argument("--learning-rate", type=str, default="fixed",
                    help="Learning-rate decay: fixed | polynomial | exponential")
parser.add_argument("--learning-rate-args", nargs="*")
parser.add_argument("--l1-regularize", type=float, default=-1.)
parser.add_argument("--l2-regularize", type=float, default=-1.)
parser.add_argument("--nb-workers", type=int, required=True,
                    help="Total number of workers")
parser.add_argument("--nb-decl-byz-workers", type=int, default=0,
                    help="Declared Byzantine workers (the GAR's f)")
parser.add_argument("--nb-real-byz-workers", type=int, default=0,
                    help="Real Byzantine workers (mount --attack)")
parser.add_argument("--attack", type=str, default="",
                    help="Attack used by the real Byzantine workers")
parser.add_argument("--attack-args", nargs="*")
parser.add_argument("--lossy", nargs="*", default=None,
                    help="UDP-style lossy-channel injection key:value args "
                         "(e.g. drop-rate:0.01 workers:0 clever:1)")
parser.add_argument("--integrity-key", type=str, default="",
                    help="Enable per-step gradient integrity MACs with this "
                         "shared secret (the reference's message-signing "
                         "equivalent; corrupted rows are NaN-filled)")
parser.add_argument("--max-step", type=int, default=config.default_max_step,
                    help="Number of additional steps to perform before "
                         "stopping the training, non-positive for no limit")
parser.add_argument("--checkpoint-dir", type=str, default="")
parser.add_argument("--checkpoint-delta", type=int,
                    default=config.default_checkpoint_delta)
parser.add_argument("--checkpoint-period", type=float,
                    default=config.default_checkpoint_period)
parser.add_argument("--summary-dir", type=str, default="")
parser.add_argument("--summary-delta", type=float,
                    default=config.default_summary_delta)
parser.add_argument("--summary-period", type=float,
                    default=config.default_summary_period)
parser.add_argument("--evaluation-file", type=str, default="")
parser.add_argument("--evaluation-delta", type=int,
                    default=config.default_evaluation_delta)
parser.add_argument("--evaluation-period", type=float,
                    default=config.default_evaluation_period)
parser.add_argument("--use-gpu", action="store_true", default=False)
parser.add_argument("--reuse-gpu", action="store_true", default=False)
parser.add_argument("--use-tpu", action="store_true", default=False,
                    help="(compat; no TPU on this platform)")
parser.add_argument("--reuse-tpu", action="store_true", default=False)
parser.add_argument("--no-wait", action="store_true", default=False)
parser.add_argument("--trace", action="store_true", default=False,
                    help="Print a debugging message for every phase of the "
                         "step execution")
parser.add_argument("--amp", action="store_true", default=False,
                    help="bf16 autocast compute (fp32 gradients/aggregation)")
parser.add_argument("--graphs", type=str, default="auto",
                    choices=["auto", "on", "off"],
                    help="hipGraph step capture (default auto: on for "
                         "capturable configs, with capture-unsafe MIOpen "
                         "solvers excluded and a replay-purity self-check "
                         "at capture that falls back to eager on failure "
                         "-- the round-2 root-cause fix, NOTES.md)")
parser.add_argument("--seed", type=int, default=1234)
parser.add_argument("--profile-steps", type=int, default=0,
                    help="Profile this many steps with torch.profiler and "
                         "export a chrome trace (rank 0)")
parser.add_argument("--profile-dir", type=str, default="profile_trace")
parser.add_argument("--progress-every", type=int, default=100,
                    help="Print loss every K steps (0 = silent)")
parser.add_argument("--stdout-to", type=str, default="-")
parser.add_argument("--stderr-to", type=str, default="-")
parser.add_argument("--MPI", action="store_true", default=False)


def main():
    with tools.Context("args", "info"):
        args = parser.parse_args(sys.argv[1:])
        if args.stdout_to != "-":
            f = open(args.stdout_to, "w")
            sys.stdout = tools.MethodCallReplicator(sys.stdout, f)
        if args.stderr_to != "-":
            f = open(args.stderr_to, "w")
            sys.stderr = tools.MethodCallReplicator(sys.stderr, f)
        if args.MPI:
            tools.warning("--MPI is obsolete on MI355X: communication is RCCL "
                          "over xGMI (ignored)")
        if args.use_tpu or args.reuse_tpu:
            tools.warning("--use-tpu/--reuse-tpu: no TPU on this platform "
                          "(ignored)")
        if args.no_wait:
            # Reference semantics (runner.py:601-610): without --no-wait a
            # --server process parked as a cluster node after training. Here
            # no process outlives its training run (ranks exit together), so
            # --no-wait is always in effect.
            tools.warning("--no-wait: processes never park as cluster nodes "
                          "in the RCCL model (always in effect)")
        tools.print_args("experiment", args.experiment, args.experiment_args or [])
        tools.print_args("aggregator", args.aggregator, args.aggregator_args or [])

    if args.graphs in ("auto", "on"):
        # Must precede the first convolution: honors the opt-in
        # AGGREGATHOR_SAFE_SOLVERS exclusion (graphstep.py docstring); the
        # capture-time replay-purity self-check is the default gate.
        from aggregathor_amd.parallel.graphstep import enable_graph_safe_conv
        enable_graph_safe_conv()

    import torch
    from aggregathor_amd import experiments
    from aggregathor_amd.attacks.lossy import LossyChannel
    from aggregathor_amd.graph import Engine
    from aggregathor_amd.parallel import WorkerGroup
    from aggregathor_amd.runner_lib import Trainer

    with tools.Context("cluster", "info"):
        use_gpu = (args.use_gpu or args.reuse_gpu) and torch.cuda.is_available()
        if use_gpu:
            torch.backends.cudnn.benchmark = True
            if os.environ.get("AGGREGATHOR_DETERMINISTIC_CONV") == "1":
                # Excludes atomic-accumulation conv algorithms (candidate
                # fix for the hipGraph open issue, NOTES.md).
                torch.backends.cudnn.deterministic = True
        if (args.use_gpu or args.reuse_gpu) and not torch.cuda.is_available():
            tools.warning("--use-gpu requested but no GPU is available; "
                          "falling back to CPU")
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        if use_gpu:
            torch.cuda.set_device(local_rank)
            device = f"cuda:{local_rank}"
        else:
            device = "cpu"
        group = WorkerGroup(args.nb_workers, device=device)
        tools.info(f"world={group.world} rank={group.rank} device={device} "
                   f"workers={group.worker_ids}")

    with tools.Context("graph", "info"):
        experiment = experiments.instantiate(
            args.experiment, args.experiment_args or [])
        lossy = None
        if args.lossy:
            lossy = LossyChannel(args.lossy)
        integrity = None
        if args.integrity_key:
            from aggregathor_amd.parallel.signing import GradientIntegrity
            integrity = GradientIntegrity(args.integrity_key, args.nb_workers)
        engine = Engine(
            experiment, args.aggregator, group,
            nbbyzwrks=args.nb_decl_byz_workers,
            aggregator_args=args.aggregator_args or [],
            optimizer=args.optimizer, optimizer_args=args.optimizer_args or [],
            learning_rate=args.learning_rate,
            learning_rate_args=args.learning_rate_args or [],
            l1_regularize=args.l1_regularize, l2_regularize=args.l2_regularize,
            nb_real_byz=args.nb_real_byz_workers, attack=args.attack,
            attack_args=args.attack_args or [], lossy=lossy, amp=args.amp,
            trace=args.trace, seed=args.seed, integrity=integrity,
            use_graphs=("auto" if args.graphs == "auto" else args.graphs == "on"))
        tools.info(f"model d = {engine.d} parameters, GAR = {args.aggregator}")

    with tools.Context("session", "info"):
        trainer = Trainer(
            engine, max_step=args.max_step,
            checkpoint_dir=args.checkpoint_dir,
            checkpoint_delta=args.checkpoint_delta,
            checkpoint_period=args.checkpoint_period,
            summary_dir=args.summary_dir or None,
            summary_delta=args.summary_delta,
            summary_period=args.summary_period,
            evaluation_file=args.evaluation_file or None,
            evaluation_delta=args.evaluation_delta,
            evaluation_period=args.evaluation_period,
            profile_steps=args.profile_steps, profile_dir=args.profile_dir)
        report = trainer.train(progress_every=args.progress_every)
        if report["diverged"]:
            # Skip the torch/HIP C++ teardown: after an aborted captured-graph
            # session it intermittently calls std::terminate from a reaper
            # thread (SIGABRT), clobbering the exit status the caller needs.
            sys.stdout.flush()
            sys.stderr.flush()
            os._exit(1)


if __name__ == "__main__":
    main()
