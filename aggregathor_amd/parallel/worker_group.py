"""Worker-group abstraction over torch.distributed (RCCL / gloo)."""

import datetime
import os

import torch
import torch.distributed as dist

from .. import tools


class WorkerGroup:
    """n total workers distributed over the process group's ranks.

    Single-process mode (no torch.distributed rendezvous): all n workers are
    *virtual* workers of this process -- the moral equivalent of the
    reference's single-machine loopback cluster (README.md:146). Multi-
    process mode: n must be divisible by the world size; each rank hosts
    n/world consecutive worker ids. ``gather`` produces the same [n, d]
    matrix on every rank.
    """

    def __init__(self, nbworkers, device="cpu", backend=None):
        self.nbworkers = nbworkers
        self.device = torch.device(device)
        if dist.is_available() and dist.is_initialized():
            self.world = dist.get_world_size()
            self.rank = dist.get_rank()
            self.backend = dist.get_backend()
        elif "RANK" in os.environ and "WORLD_SIZE" in os.environ:
            if backend is None:
                backend = "nccl" if self.device.type == "cuda" else "gloo"
            dist.init_process_group(
                backend=backend, timeout=datetime.timedelta(seconds=300))
            self.world = dist.get_world_size()
            self.rank = dist.get_rank()
            self.backend = backend
        else:
            self.world = 1
            self.rank = 0
            self.backend = None
        if nbworkers % self.world != 0:
            raise tools.UserException(
                f"nb-workers ({nbworkers}) must be divisible by the world size "
                f"({self.world})")
        self.local_workers = nbworkers // self.world
        self.worker_ids = list(range(self.rank * self.local_workers,
                                     (self.rank + 1) * self.local_workers))
        self._flag_buf = None  # lazy; see any_rank

    @property
    def distributed(self):
        return self.world > 1

    def gather(self, local_rows, out=None):
        """All-gather local worker gradient rows into the full [n, d] matrix.

        Args:
          local_rows: [local_workers, d] contiguous tensor.
          out: optional preallocated [n, d] output (avoids reallocation; on
               GPU this matrix stays resident in HBM across steps).
        Returns:
          [n, d] matrix, identical on every rank (row w = worker w).
        """
        if self.world == 1:
            return local_rows
        n = self.nbworkers
        d = local_rows.shape[1]
        if out is None:
            out = torch.empty((n, d), dtype=local_rows.dtype,
                              device=local_rows.device)
        if self.backend == "nccl":
            dist.all_gather_into_tensor(out.view(-1), local_rows.reshape(-1))
        else:
            chunks = list(out.view(self.world, self.local_workers * d).unbind(0))
            dist.all_gather(chunks, local_rows.reshape(-1))
        return out

    def barrier(self):
        if self.world > 1:
            dist.barrier()

    def broadcast_model(self, model):
        """Broadcast rank-0's parameters/buffers (identical init guarantee)."""
        if self.world == 1:
            return
        for t in list(model.parameters()) + list(model.buffers()):
            dist.broadcast(t.data, src=0)

    def gather_small(self, obj):
        """All-gather a small per-rank object (e.g. MAC vectors); returns
        the rank-ordered list."""
        if self.world == 1:
            return [obj]
        objs = [None] * self.world
        dist.all_gather_object(objs, obj)
        return objs

    def allreduce_max(self, value):
        """Max over ranks of a Python float (used for worst-rank step time)."""
        if self.world == 1:
            return value
        # RCCL reduces device tensors; gloo reduces host tensors.
        dev = self.device if self.backend == "nccl" else "cpu"
        t = torch.tensor([value], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return t.item()

    def any_rank(self, flag):
        """True iff `flag` is True on ANY rank (MAX all-reduce of a bit).

        Used for the collective divergence abort: a rank whose local loss
        goes non-finite must not exit alone -- the others would block in the
        next ``gather`` until the process-group timeout. Every rank folds
        its local flag in every step so all ranks agree on the abort step
        (local losses differ per rank, so local detection steps would too).

        The flag buffer is PREALLOCATED: this runs every step, and steady-
        state device allocations are the trigger class that corrupts live
        captured graphs (profiles/graph_purity_bisect.md).
        """
        if self.world == 1:
            return bool(flag)
        if self._flag_buf is None:
            dev = self.device if self.backend == "nccl" else "cpu"
            self._flag_buf = torch.zeros(1, device=dev)
        self._flag_buf.fill_(1.0 if flag else 0.0)
        dist.all_reduce(self._flag_buf, op=dist.ReduceOp.MAX)
        return bool(self._flag_buf.item())
