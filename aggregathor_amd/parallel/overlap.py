"""Bucketed gather overlap: hide the RCCL all-gather behind backward.

At N GPUs with one worker per rank, the serial step is
``backward -> all_gather([n, d]) -> GAR``: the gather (~1 ms for the
ResNet-50 n=8 matrix over xGMI) sits on the critical path. This module
overlaps it with the backward pass, the same idea as the reference's
dedicated-communication-stream rationale (tf_patches/kernels/
mpi_ops.cc:231-247) re-expressed for torch.distributed:

* the flat gradient row is partitioned into BUCKETS aligned to parameter
  boundaries (default 25 MB);
* a ``register_post_accumulate_grad_hook`` on every parameter marks its
  bucket complete as backward produces gradients (tail of the model
  first, since the grad views make ``param.grad`` writes land directly
  in the row);
* complete buckets are issued as ASYNC ``all_gather_into_tensor``
  collectives **in a fixed bucket order** (reverse flat order, the
  order backward completes): hooks only mark readiness, and an issuing
  cursor launches ready buckets strictly in that order -- so every rank
  issues the same collective sequence regardless of scheduling jitter
  (the NCCL ordering requirement), and a late bucket merely delays
  overlap, never correctness;
* ``finish()`` flushes any never-signalled buckets (e.g. parameters an
  attack-free loss did not touch), waits for all works, and scatters the
  per-bucket staging buffers into the [n, d] matrix columns.

Scope (v1, opt-in via ``AGGREGATHOR_BUCKET_MB``): distributed runs with
one local worker per rank (the N=n scaling configuration), eager local
phase (full-graph capture cannot fire Python hooks during replay), no
real-Byzantine attack (the attack rewrites the row AFTER backward, i.e.
after buckets would already be in flight). The engine falls back to the
serial gather whenever the scope does not apply. Deterministic: bucket
boundaries, issue order, and staging layout are pure functions of the
parameter list, so the gathered matrix is bitwise identical to the
serial path (asserted by tests/test_distributed.py over gloo).
"""

import torch
import torch.distributed as dist


class BucketedGather:
    """Overlapped all-gather of the local gradient row."""

    def __init__(self, group, params, row, matrix, bucket_bytes=25 << 20):
        """
        Args:
          group:  WorkerGroup (distributed, local_workers == 1)
          params: engine parameter list (flat order defines offsets)
          row:    the [1, d] local gradient row (grad views bound into it)
          matrix: the [n, d] gathered matrix to fill
        """
        assert group.distributed and group.local_workers == 1
        self.group = group
        self.row = row
        self.matrix = matrix
        n, d = matrix.shape

        # Bucket layout: parameter-aligned spans of the flat row, built
        # back to front so bucket 0 is the TAIL of the model (the first
        # gradients backward produces).
        spans = []
        off = 0
        self.param_bucket = {}
        bounds = []
        acc = 0
        for p in params:
            bounds.append((off, p))
            off += p.numel()
        assert off == d
        cur_end = d
        cur_params = []
        for start, p in reversed(bounds):
            cur_params.append(p)
            acc += p.numel() * 4
            if acc >= bucket_bytes or start == 0:
                spans.append((start, cur_end))
                for q in cur_params:
                    self.param_bucket[id(q)] = len(spans) - 1
                cur_end = start
                cur_params = []
                acc = 0
        self.spans = spans  # bucket i: row[start:end), issued in order i
        self.stage = [torch.empty((n, e - s), dtype=row.dtype,
                                  device=row.device)
                      for s, e in spans]
        self._pending = [0] * len(spans)
        self._param_count = [0] * len(spans)
        for p in params:
            self._param_count[self.param_bucket[id(p)]] += 1
        self._ready = [False] * len(spans)
        self._issued = [None] * len(spans)
        self._cursor = 0
        self._active = False

        self._hooks = []
        for p in params:
            self._hooks.append(p.register_post_accumulate_grad_hook(
                self._on_grad_ready))

    # ------------------------------------------------------------------ #

    def _on_grad_ready(self, param):
        if not self._active:
            return
        b = self.param_bucket[id(param)]
        self._pending[b] -= 1
        if self._pending[b] == 0:
            self._ready[b] = True
            self._issue_ready()

    def _issue_ready(self):
        """Launch ready buckets strictly in fixed order (cursor)."""
        while self._cursor < len(self.spans) and self._ready[self._cursor]:
            b = self._cursor
            s, e = self.spans[b]
            self._issued[b] = dist.all_gather_into_tensor(
                self.stage[b].view(-1), self.row[0, s:e].contiguous(),
                async_op=True)
            self._cursor += 1

    def begin_step(self):
        """Arm the hooks for this step's backward."""
        for b in range(len(self.spans)):
            self._pending[b] = self._param_count[b]
            self._ready[b] = False
            self._issued[b] = None
        self._cursor = 0
        self._active = True

    def finish(self):
        """Flush + wait all buckets; scatter into the matrix; return it."""
        self._active = False
        for b in range(len(self.spans)):  # flush never-signalled buckets
            if not self._ready[b]:
                self._ready[b] = True
        self._issue_ready()
        for b, (s, e) in enumerate(self.spans):
            self._issued[b].wait()
            self.matrix[:, s:e].copy_(self.stage[b])
        return self.matrix

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []
