"""AggregaThor-AMD: MI355X-native Byzantine-resilient distributed SGD.

A from-scratch re-design of LPD-EPFL/AggregaThor (SysML'19) for AMD Instinct
MI355X (gfx950): PyTorch-ROCm models and autograd, hand-written HIP/CDNA4
kernels for the robust Gradient Aggregation Rules (Krum, Multi-Krum, Bulyan,
coordinate-wise median / averaged-median / NaN-mean), and RCCL over xGMI
(torch.distributed) replacing the reference's gRPC/MPI parameter server:
each of the 8 GPUs on a node is one worker whose flattened gradient is
all-gathered every step, the GAR runs replicated on every rank, and the
update is applied locally and deterministically.

Package layout:
  tools/        logging, registries, checkpoints (ref: tools/)
  config        defaults (ref: config.py)
  aggregators/  GAR plugins (ref: aggregators/)
  ops/          GAR compute: HIP gfx950 kernels + PyTorch oracles (ref: native/)
  models/       model zoo: MLP, CNNet, ResNet families (ref: experiments/ + external/slim)
  experiments/  experiment plugins incl. data poisoning (ref: experiments/)
  attacks/      Byzantine gradient attacks + UDP-style lossy injection
                (ref: --attack flags + tf_patches UDP transport)
  parallel/     torch.distributed/RCCL worker group (ref: tf_patches comms)
  graph         training engine (ref: graph.py)
  runner_lib    train loop + eval/checkpoint/summary services (ref: runner.py)
"""

__version__ = "0.1.0"
