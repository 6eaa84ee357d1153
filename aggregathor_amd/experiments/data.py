"""Deterministic synthetic classification data.

There is no dataset download in this environment; the reference's benchmark
configs (BASELINE.json) explicitly use synthetic data with random-init
weights for the large models. For the small models (MNIST MLP, CIFAR CNN) a
fixed random "teacher" projection generates learnable labels so convergence
and accuracy-under-attack curves are meaningful.

Determinism: batch ``(worker, step)`` is a pure function of the experiment
seed, so every rank regenerates identical data for identical worker ids and
runs are exactly reproducible (the reference relied on tf.data shuffling
seeds; here determinism is a first-class contract used by the tests).
"""

import torch


class SyntheticClassification:
    """Deterministic synthetic (inputs, labels) source."""

    def __init__(self, shape, classes, seed=1234, teacher=None,
                 eval_examples=1024, scale=1.0):
        """
        Args:
          shape:   per-example input shape, e.g. (784,) or (3, 32, 32)
          classes: number of classes
          seed:    base seed; all data derives deterministically from it
          teacher: generate labels from a fixed random linear teacher when the
                   flat dim is small (default: dim <= 8192); False = uniform
                   random labels (throughput benchmarking)
          eval_examples: size of the held-out deterministic eval set
        """
        self.shape = tuple(shape)
        self.classes = classes
        self.seed = seed
        self.scale = scale
        self.eval_examples = eval_examples
        dim = 1
        for s in self.shape:
            dim *= s
        self.dim = dim
        self.teacher = (dim <= 8192) if teacher is None else teacher
        if self.teacher:
            gen = torch.Generator().manual_seed(seed ^ 0x7EAC4E12)
            self.teacher_w = torch.randn(dim, classes, generator=gen)
        else:
            self.teacher_w = None
        # Device-resident batch pool: on GPU, per-(worker, step) batches are
        # served from a once-generated pool of `pool_size` distinct batches
        # (batch(worker, step) == cpu batch(worker, step % pool_size)) --
        # generating 20 MB of fresh host randoms per micro-batch would
        # bottleneck an MI355X training step on the host RNG + H2D copy.
        self.pool_size = 8
        self._pools = {}

    def _labels(self, flat_inputs, gen):
        if self.teacher:
            return (flat_inputs @ self.teacher_w).argmax(dim=1)
        return torch.randint(0, self.classes, (flat_inputs.shape[0],), generator=gen)

    def _raw_batch(self, batch_size, worker, step):
        gen = torch.Generator().manual_seed(
            (self.seed * 1000003 + worker * 7919 + step * 104729) & 0x7FFFFFFF)
        x = torch.randn((batch_size, *self.shape), generator=gen) * self.scale
        y = self._labels(x.flatten(1), gen)
        return x, y

    def batch(self, batch_size, worker, step, device="cpu"):
        """Training batch for (worker, step): pure function of the seed."""
        device = torch.device(device)
        if device.type == "cuda":
            key = (batch_size, worker, str(device))
            pool = self._pools.get(key)
            if pool is None:
                pool = []
                for s in range(self.pool_size):
                    x, y = self._raw_batch(batch_size, worker, s)
                    pool.append((x.to(device), y.to(device)))
                self._pools[key] = pool
            return pool[step % self.pool_size]
        x, y = self._raw_batch(batch_size, worker, step)
        return x.to(device, non_blocking=True), y.to(device, non_blocking=True)

    def eval_batches(self, batch_size, device="cpu"):
        """Fixed deterministic eval set, independent of the training stream."""
        gen = torch.Generator().manual_seed(self.seed ^ 0x5EED5EED)
        remaining = self.eval_examples
        while remaining > 0:
            bs = min(batch_size, remaining)
            x = torch.randn((bs, *self.shape), generator=gen) * self.scale
            y = self._labels(x.flatten(1), gen)
            yield x.to(device), y.to(device)
            remaining -= bs
