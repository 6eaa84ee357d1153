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
                 eval_examples=1024, scale=1.0, signal=0.5, pool_size=8):
        """
        Args:
          shape:   per-example input shape, e.g. (784,) or (3, 32, 32)
          classes: number of classes
          seed:    base seed; all data derives deterministically from it
          teacher: True = labeled task (default when flat dim <= 8192):
                   x = noise + signal * pattern[y] with one fixed random
                   pattern per class -- a global-template detection task
                   every architecture (MLPs AND pooling convnets) can learn.
                   False = uniform random labels (throughput benchmarking).
          signal:  class-pattern amplitude (task difficulty knob)
          eval_examples: size of the held-out deterministic eval set
          pool_size: GPU batch-pool size (see below); 0 disables pooling --
                   every (worker, step) batch is freshly generated and
                   identical to the CPU stream. Use 0 (or a large pool) for
                   accuracy / attack-convergence evaluations; the small
                   default pool is for throughput benchmarking, where
                   per-step host RNG + H2D would bottleneck the MI355X step.
        """
        self.shape = tuple(shape)
        self.classes = classes
        self.seed = seed
        self.scale = scale
        self.signal = signal
        self.eval_examples = eval_examples
        dim = 1
        for s in self.shape:
            dim *= s
        self.dim = dim
        self.teacher = (dim <= 8192) if teacher is None else teacher
        if self.teacher:
            gen = torch.Generator().manual_seed(seed ^ 0x7EAC4E12)
            self.patterns = torch.randn((classes, *self.shape), generator=gen)
        else:
            self.patterns = None
        # Device-resident batch pool: on GPU, per-(worker, step) batches are
        # served from a once-generated pool of `pool_size` distinct batches
        # (batch(worker, step) == cpu batch(worker, step % pool_size)) --
        # generating 20 MB of fresh host randoms per micro-batch would
        # bottleneck an MI355X training step on the host RNG + H2D copy.
        # pool_size=0 disables pooling (GPU stream == CPU stream).
        self.pool_size = pool_size
        self._pools = {}

    def _make(self, batch_size, gen):
        """Generate (inputs, labels) from a generator."""
        x = torch.randn((batch_size, *self.shape), generator=gen) * self.scale
        if self.teacher:
            y = torch.randint(0, self.classes, (batch_size,), generator=gen)
            x = x + self.signal * self.patterns[y]
        else:
            y = torch.randint(0, self.classes, (batch_size,), generator=gen)
        return x, y

    def _raw_batch(self, batch_size, worker, step):
        gen = torch.Generator().manual_seed(
            (self.seed * 1000003 + worker * 7919 + step * 104729) & 0x7FFFFFFF)
        return self._make(batch_size, gen)

    def batch(self, batch_size, worker, step, device="cpu"):
        """Training batch for (worker, step): pure function of the seed."""
        device = torch.device(device)
        if device.type == "cuda" and self.pool_size > 0:
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
            x, y = self._make(bs, gen)
            yield x.to(device), y.to(device)
            remaining -= bs
