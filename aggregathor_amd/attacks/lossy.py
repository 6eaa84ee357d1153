"""UDP-style lossy-gradient channel injection.

Reproduces the reference's unreliable-UDP transport semantics as a
deterministic injection on the gathered gradient matrix:
large worker->PS gradients traveled as 65000-byte chunks
(tf_patches/patches/mpi_rendezvous_mgr.patch:563-592); lost or forged chunks
were back-filled with NaN bytes, or with the last known-good copy of the
same chunk when ``CLEVER=1`` (mpi_rendezvous_mgr.patch:735-860). On MI355X
the physical transport (RCCL over xGMI) is reliable, so loss is *injected*,
not suffered -- preserving the average-nan / averaged-median use case.

This is NOT an attack plugin (it models the channel, not a worker): the
training engine applies it to the gathered [n, d] matrix each step.
"""

import torch

from .. import config, tools


class LossyChannel:
    """Deterministic per-(step, worker, chunk) packet-loss injection."""

    def __init__(self, args=None, nbworkers=None):
        args = tools.parse_keyval(args, defaults={
            "drop-rate": 0.0,      # probability each chunk of a worker's gradient is lost
            "workers": "",         # comma-separated worker ids subject to loss ("" = all)
            "clever": 0,           # 1 = back-fill with last known-good chunk, 0 = NaN fill
            "seed": 20190331,
            "chunk-bytes": config.lossy_chunk_bytes,
        })
        self.drop_rate = float(args["drop-rate"])
        self.clever = bool(int(args["clever"]))
        self.seed = int(args["seed"])
        self.chunk_bytes = int(args["chunk-bytes"])
        if args["workers"]:
            self.workers = set(int(w) for w in str(args["workers"]).split(","))
        else:
            self.workers = None  # all workers
        self._last_good = None  # [n, d] copy of the last fully-received matrix

    def chunk_elems(self, dtype):
        return max(1, self.chunk_bytes // torch.empty((), dtype=dtype).element_size())

    def drop_mask(self, n, d, dtype, step, device):
        """[n, n_chunks] bool mask of dropped chunks for this step."""
        ce = self.chunk_elems(dtype)
        n_chunks = (d + ce - 1) // ce
        gen = torch.Generator().manual_seed(
            (self.seed * 1000003 + step * 104729) & 0x7FFFFFFF)
        mask = torch.rand((n, n_chunks), generator=gen) < self.drop_rate
        if self.workers is not None:
            keep = torch.ones(n, dtype=torch.bool)
            for w in self.workers:
                if 0 <= w < n:
                    keep[w] = False
            mask[keep] = False
        return mask.to(device)

    def inject(self, matrix, step):
        """Apply chunk loss to the gathered [n, d] matrix (returns it, modified
        in place). NaN fill, or last-good substitution with ``clever:1``."""
        if self.drop_rate <= 0.0:
            return matrix
        n, d = matrix.shape
        ce = self.chunk_elems(matrix.dtype)
        mask = self.drop_mask(n, d, matrix.dtype, step, matrix.device)  # [n, nc]
        # Expand chunk mask to element granularity.
        elem_mask = mask.repeat_interleave(ce, dim=1)[:, :d]
        if self.clever:
            # Last-known-good substitution; before any copy exists, fill with
            # zero (a neutral gradient contribution) instead of the
            # reference's boot-time NaN -- average would otherwise diverge on
            # the very first step with nothing gained.
            if self._last_good is None:
                matrix[elem_mask] = 0.0
            else:
                matrix[elem_mask] = self._last_good[elem_mask]
        else:
            matrix[elem_mask] = float("nan")
        if self.clever:
            # Retain the latest value of every *received* chunk (the reference
            # kept the last-known-good copy per chunk, patch :833-835).
            if self._last_good is None:
                self._last_good = torch.where(
                    elem_mask, torch.zeros_like(matrix), matrix).clone()
            else:
                self._last_good = torch.where(elem_mask, self._last_good, matrix)
        return matrix
