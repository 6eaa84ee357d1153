"""Distributed worker group: RCCL over xGMI / gloo over CPU.

Replaces the reference's parameter-server communication stack (stock TF gRPC
+ the grpc+mpi / UDP patches) with the MI355X-native layout:
one process per GPU via ``torch.distributed`` (backend "nccl" IS RCCL on
ROCm), each rank hosting one or more *virtual workers*. Per step the n
flattened worker gradients are all-gathered into the resident [n, d] matrix
over xGMI, the GAR runs REPLICATED on every rank (deterministic kernels on
identical input -> identical aggregate), and the optimizer applies locally.
Replicated aggregation beats the reference's PS round trip: the GAR is
memory-bound (~2 passes over n x d at ~8 TB/s per GPU) which is cheaper than
broadcasting the d-vector over one ~153 GB/s xGMI link.
"""

from .worker_group import WorkerGroup

__all__ = ["WorkerGroup"]
