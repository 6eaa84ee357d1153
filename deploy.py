#!/usr/bin/env python3
"""Deploy a multi-process (one rank per GPU) training session.

MI355X-native re-expression of the reference's deploy.py
(/root/reference/deploy.py): where the reference piped its own source over
SSH to bring up tf.train.Server processes and assembled mpirun command
lines, this deployer spawns one runner.py process per GPU of a node (local
mode), or over SSH for multi-node specs, wiring the torch.distributed
rendezvous (RANK / WORLD_SIZE / MASTER_ADDR / MASTER_PORT) -- the RCCL
process group replaces the reference's gRPC/MPI cluster entirely.

Examples:
  # 8 ranks on the local node, Krum f=2, ResNet-50:
  python deploy.py --nproc 8 -- \
      --experiment resnet50-imagenet --aggregator krum \
      --nb-workers 8 --nb-decl-byz-workers 2 --use-gpu --amp --max-step 1000

  # explicit cluster spec (JSON like the reference's):
  python deploy.py --cluster '{"workers": ["host1:29517", "host2:29517"]}' \
      --nproc-per-node 8 -- <runner args>
"""

import argparse
import os
import pathlib
import shlex
import signal
import subprocess
import sys

from aggregathor_amd import config, tools

REPO = pathlib.Path(__file__).resolve().parent


def parse_cluster(spec):
    """Parse a cluster spec -- JSON {"workers": ["host:port", ...]} or a
    special value like "G5k" (reference tools/cluster.py:81-91; the ps/eval
    jobs have no MI355X equivalent -- every rank is a worker)."""
    data = tools.cluster_parse(spec)
    hosts = []
    for job in ("workers", "ps", "local"):
        for entry in data.get(job, []):
            host = entry.rsplit(":", 1)[0]
            if host not in hosts:
                hosts.append(host)
    if not hosts:
        raise tools.UserException(f"empty cluster specification {spec!r}")
    return hosts


def main():
    ap = argparse.ArgumentParser(
        description="Deploy a multi-process training session.")
    ap.add_argument("--cluster", type=str, default="",
                    help="JSON cluster spec (reference format); default: "
                         "local node only")
    ap.add_argument("--nproc", "--nproc-per-node", dest="nproc", type=int,
                    default=0, help="ranks per node (default: #GPUs, else 1)")
    ap.add_argument("--master-addr", type=str, default=config.default_master_addr)
    ap.add_argument("--master-port", type=int, default=config.default_master_port)
    ap.add_argument("--ssh", type=str, default="ssh",
                    help="remote shell command for multi-node deployment")
    ap.add_argument("--repo-path", type=str, default=str(REPO),
                    help="repo path on remote nodes")
    ap.add_argument("runner_args", nargs=argparse.REMAINDER,
                    help="arguments forwarded to runner.py (prefix with --)")
    args = ap.parse_args()

    runner_args = args.runner_args
    if runner_args and runner_args[0] == "--":
        runner_args = runner_args[1:]

    nproc = args.nproc
    if nproc <= 0:
        try:
            import torch
            nproc = max(torch.cuda.device_count(), 1)
        except Exception:
            nproc = 1

    hosts = parse_cluster(args.cluster) if args.cluster else ["localhost"]
    world = nproc * len(hosts)

    procs = []

    def shutdown(signum=None, frame=None):
        for p in procs:
            if p.poll() is None:
                p.terminate()

    signal.signal(signal.SIGINT, shutdown)
    signal.signal(signal.SIGTERM, shutdown)

    with tools.Context("deploy", "info"):
        tools.info(f"world size {world}: {len(hosts)} node(s) x {nproc} rank(s)")
        rank = 0
        for node_i, host in enumerate(hosts):
            for local in range(nproc):
                env_pairs = {
                    "RANK": str(rank), "LOCAL_RANK": str(local),
                    "WORLD_SIZE": str(world),
                    "MASTER_ADDR": args.master_addr,
                    "MASTER_PORT": str(args.master_port),
                    "HSA_ENABLE_IPC_MODE_LEGACY":
                        os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY", "0"),
                }
                cmd = [sys.executable, str(REPO / "runner.py")] + runner_args
                if host in ("localhost", "127.0.0.1"):
                    env = dict(os.environ)
                    env.update(env_pairs)
                    p = subprocess.Popen(cmd, env=env)
                else:
                    env_str = " ".join(f"{k}={v}" for k, v in env_pairs.items())
                    remote = (f"cd {shlex.quote(args.repo_path)} && {env_str} "
                              + " ".join(shlex.quote(c) for c in
                                         ([sys.executable, "runner.py"]
                                          + runner_args)))
                    p = subprocess.Popen([args.ssh, host, remote])
                procs.append(p)
                tools.info(f"rank {rank} -> {host} (local_rank {local}, "
                           f"pid {p.pid})")
                rank += 1
        rcs = [p.wait() for p in procs]
        bad = [i for i, rc in enumerate(rcs) if rc != 0]
        if bad:
            tools.error(f"ranks {bad} exited non-zero: "
                        f"{[rcs[i] for i in bad]}")
            sys.exit(1)
        tools.success("all ranks completed")


if __name__ == "__main__":
    main()
