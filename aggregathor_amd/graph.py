"""Training engine: the MI355X-native equivalent of the reference's
``graph.Manager`` (/root/reference/graph.py:204-315).

Where the reference built one big TF graph (per-worker replica losses,
gradient flatten, GAR node, inflate + apply on the parameter server), this
engine drives the same pipeline imperatively per step:

  1. per local worker: micro-batch forward + backward, with ``param.grad``
     bound as views into that worker's row of the local gradient matrix --
     autograd writes the flattened gradient IN PLACE (zero-copy flatten,
     replacing graph.py:144-168's reshape/concat),
  2. real-Byzantine workers' rows replaced by the attack plugin,
  3. RCCL all-gather of the [n, d] matrix over xGMI (WorkerGroup.gather),
  4. optional lossy-channel injection (UDP-semantics reproduction),
  5. GAR kernel, replicated deterministically on every rank,
  6. ``param.grad`` rebound as views into the aggregated flat gradient
     (zero-copy inflate, graph.py:182-199) + optimizer step.

Learning-rate schedules (fixed / polynomial / exponential) and the optimizer
set (adadelta / adagrad / adam / rmsprop / sgd) mirror graph.py:51-66 with
the same sub-argument names and defaults.
"""

import torch

from . import aggregators as aggregators_mod
from . import attacks as attacks_mod
from . import config, tools

# ---------------------------------------------------------------------------- #
# Learning-rate schedules (reference graph.py:51-57)

learning_rates = {
    "fixed": {
        "args": {"initial-rate": config.default_learning_rate},
        "fn": lambda step, a: a["initial-rate"],
    },
    "polynomial": {
        "args": {"initial-rate": config.default_learning_rate,
                 "end-rate": config.default_end_learning_rate,
                 "decay-step": config.default_decay_step,
                 "power": 1.},
        # tf.train.polynomial_decay with cycle=False (step clamped).
        "fn": lambda step, a: (
            (a["initial-rate"] - a["end-rate"])
            * (1 - min(step, a["decay-step"]) / a["decay-step"]) ** a["power"]
            + a["end-rate"]),
    },
    "exponential": {
        "args": {"initial-rate": config.default_learning_rate,
                 "decay-step": config.default_decay_step,
                 "decay-rate": config.default_decay_rate},
        # tf.train.exponential_decay, staircase=False.
        "fn": lambda step, a: a["initial-rate"] * a["decay-rate"] ** (step / a["decay-step"]),
    },
}

# Optimizers (reference graph.py:58-66); defaults follow the reference's
# TF-1 hyper-parameters (e.g. adadelta eps=1.0, rmsprop decay=0.9 eps=1e-10).
optimizers = {
    "adadelta": {
        "args": {"adadelta-rho": 0.95, "opt-epsilon": 1.},
        "fn": lambda params, lr, a: torch.optim.Adadelta(
            params, lr=lr, rho=a["adadelta-rho"], eps=a["opt-epsilon"]),
    },
    "adagrad": {
        "args": {"initial-accumulator-value": 0.1},
        "fn": lambda params, lr, a: torch.optim.Adagrad(
            params, lr=lr, initial_accumulator_value=a["initial-accumulator-value"]),
    },
    "adam": {
        "args": {"adam-beta1": 0.9, "adam-beta2": 0.999},
        "fn": lambda params, lr, a: torch.optim.Adam(
            params, lr=lr, betas=(a["adam-beta1"], a["adam-beta2"])),
    },
    "rmsprop": {
        "args": {},
        "fn": lambda params, lr, a: torch.optim.RMSprop(
            params, lr=lr, alpha=0.9, eps=1e-10),
    },
    "sgd": {
        "args": {},
        "fn": lambda params, lr, a: torch.optim.SGD(params, lr=lr),
    },
}


def build_lr_schedule(name, args):
    if name not in learning_rates:
        raise tools.UserException(
            f"Unknown learning rate {name!r}, expected one of: "
            + ", ".join(sorted(learning_rates)))
    spec = learning_rates[name]
    parsed = tools.parse_keyval(args, defaults=spec["args"])
    known = {k: parsed[k] for k in spec["args"]}
    return lambda step: spec["fn"](step, known)


def build_optimizer(name, args, params, lr):
    if name not in optimizers:
        raise tools.UserException(
            f"Unknown optimizer {name!r}, expected one of: "
            + ", ".join(sorted(optimizers)))
    spec = optimizers[name]
    parsed = tools.parse_keyval(args, defaults=spec["args"])
    known = {k: parsed[k] for k in spec["args"]}
    return spec["fn"](params, lr, known)


# ---------------------------------------------------------------------------- #
# Flatten/inflate helpers (reference graph.py:144-199), view-based.


def flat_size(params):
    return sum(p.numel() for p in params)


def bind_grad_views(params, flat):
    """Bind each param's ``.grad`` as a view into the flat [d] buffer.

    The view adopts the PARAM's stride layout (`as_strided`), so channels-
    last conv weights get channels-last grad views: autograd accumulates
    without a layout conversion, and the flat buffer simply stores that
    param's gradient in NHWC element order -- identical on every rank, so
    all GAR semantics are unchanged (a consistent coordinate permutation).
    """
    off = 0
    for p in params:
        n = p.numel()
        p.grad = flat[off:off + n].as_strided(p.shape, p.stride())
        off += n
    assert off == flat.numel()


# ---------------------------------------------------------------------------- #
# Engine


class Engine:
    """One rank's training engine (all ranks run the same engine)."""

    def __init__(self, experiment, aggregator, group,
                 nbbyzwrks=0, aggregator_args=None,
                 optimizer="sgd", optimizer_args=None,
                 learning_rate="fixed", learning_rate_args=None,
                 l1_regularize=-1., l2_regularize=-1.,
                 nb_real_byz=0, attack=None, attack_args=None,
                 lossy=None, amp=False, trace=False, seed=1234,
                 use_graphs="auto", graph_warmup=3, integrity=None):
        """
        Args:
          experiment: an instantiated _Experiment
          aggregator: GAR name (see aggregathor_amd.aggregators.itemize())
          group:      WorkerGroup (defines n and this rank's worker ids)
          nbbyzwrks:  declared f for the GAR
          nb_real_byz / attack / attack_args: real Byzantine workers
                      (ids 0..nb_real_byz-1) and the attack they mount
          lossy:      optional attacks.lossy.LossyChannel
          amp:        bf16 autocast for forward/backward (fp32 gradients)
        """
        self.experiment = experiment
        self.group = group
        self.device = group.device
        self.n = group.nbworkers
        self.f = nbbyzwrks
        self.amp = amp and self.device.type == "cuda"
        self.trace = trace
        self.lossy = lossy
        self.integrity = integrity
        self.global_step = 0

        torch.manual_seed(seed)
        self.model = experiment.model().to(self.device)
        self.channels_last = (self.device.type == "cuda" and
                              any(p.dim() == 4 for p in self.model.parameters()))
        if self.channels_last:
            # NHWC end to end: MIOpen's NHWC conv kernels skip the
            # batched-transpose passes. Grad views adopt the param strides
            # (bind_grad_views), so accumulation stays layout-matched.
            self.model = self.model.to(memory_format=torch.channels_last)
        group.broadcast_model(self.model)
        self.params = [p for p in self.model.parameters() if p.requires_grad]
        for p in self.params:
            # Grad views cover [0, numel) through the param's strides; that
            # requires a dense layout (contiguous or channels-last).
            if not (p.is_contiguous() or
                    (p.dim() == 4 and
                     p.is_contiguous(memory_format=torch.channels_last))):
                raise tools.UserException(
                    f"parameter of shape {tuple(p.shape)} has a non-dense "
                    "layout; the flattened-gradient views require dense "
                    "parameters")
        self.d = flat_size(self.params)

        self.gar = aggregators_mod.instantiate(
            aggregator, self.n, nbbyzwrks, aggregator_args or [])
        self.l1 = l1_regularize
        self.l2 = l2_regularize
        self.lr_schedule_name = learning_rate
        self.lr_fn = build_lr_schedule(learning_rate, learning_rate_args or [])
        self.optimizer = build_optimizer(
            optimizer, optimizer_args or [], self.params, self.lr_fn(0))

        self.phase_times = {}
        self._eval_model = None
        self._eval_stream = None
        # Serializes training steps against the service threads' device work
        # (evaluate / state_dict): concurrent eval-model builds + MIOpen
        # benchmark-find + forwards interleaving with hipGraph replays were
        # observed to corrupt training (loss collapse ~20 steps after the
        # first eval fire; one segfault). The reference evaluated on separate
        # devices; on one GPU, serialization is the sound choice.
        import threading
        self.lock = threading.Lock()
        self.nb_real_byz = nb_real_byz
        self.attack = None
        if nb_real_byz > 0 and attack:
            self.attack = attacks_mod.instantiate(attack, attack_args or [])

        # Resident buffers: local gradient rows, gathered matrix, aggregate.
        lw = group.local_workers
        self.local_rows = torch.zeros((lw, self.d), dtype=torch.float32,
                                      device=self.device)
        if group.distributed:
            self.matrix = torch.empty((self.n, self.d), dtype=torch.float32,
                                      device=self.device)
        else:
            self.matrix = self.local_rows
        self.agg_flat = torch.zeros(self.d, dtype=torch.float32,
                                    device=self.device)
        self.last_loss = float("nan")

        # hipGraph step capture (parallel/graphstep.py): auto-enabled on GPU
        # for capturable configs, after `graph_warmup` eager steps.
        from .parallel.graphstep import CapturedStep
        self._graph_cls = CapturedStep
        self.graph_warmup = graph_warmup
        if use_graphs == "auto":
            self.use_graphs = CapturedStep.supported(self)
        else:
            self.use_graphs = bool(use_graphs) and CapturedStep.supported(self)
        if self.use_graphs:
            # No conv has executed yet (first forward is in step()): the
            # capture-unsafe solver exclusion can still take effect.
            from .parallel.graphstep import enable_graph_safe_conv
            enable_graph_safe_conv()
        self._graphstep = None

        # Opt-in bucketed gather overlap (parallel/overlap.py): hide the
        # all-gather behind backward. Scope v1: one worker per rank,
        # eager local phase, no real-Byzantine attack (it rewrites the
        # row after backward, when buckets would already be in flight).
        self.overlap = None
        import os as _os
        _mb = _os.environ.get("AGGREGATHOR_BUCKET_MB")
        if (_mb and group.distributed and group.local_workers == 1
                and not self.use_graphs and self.attack is None):
            from .parallel.overlap import BucketedGather
            self.overlap = BucketedGather(
                group, self.params, self.local_rows, self.matrix,
                bucket_bytes=int(float(_mb) * (1 << 20)))

    # ------------------------------------------------------------------ #

    def _trace(self, msg):
        if self.trace:
            tools.trace(f"[step {self.global_step}] {msg}")

    def _phase(self, name, t0):
        """Accumulate per-phase wall time (enabled with --trace; each phase
        boundary synchronizes the device, so this perturbs throughput --
        it is the reference's tf.Print phase bracketing analog)."""
        import time
        if self.device.type == "cuda":
            torch.cuda.synchronize()
        t1 = time.monotonic()
        self.phase_times[name] = self.phase_times.get(name, 0.0) + (t1 - t0)
        return t1

    def _format_batch(self, batch):
        if self.channels_last:
            x, y = batch
            if x.dim() == 4:
                return x.to(memory_format=torch.channels_last), y
        return batch

    def _regularization(self):
        """Reference graph.py:125-139: l1 = sum |w|; l2 = sqrt(sum w^2)."""
        reg = None
        if self.l1 > 0:
            reg = self.l1 * sum(p.abs().sum() for p in self.params)
        if self.l2 > 0:
            l2 = self.l2 * torch.sqrt(sum((p * p).sum() for p in self.params))
            reg = l2 if reg is None else reg + l2
        return reg

    def compute_local_gradients(self):
        """Fill ``local_rows`` with this rank's worker gradients; returns the
        mean loss over the local workers (as a tensor)."""
        import contextlib
        self.model.train()
        losses = []
        # One autocast region for the whole worker loop: the bf16 weight
        # casts are cached across the local workers' forwards instead of
        # being re-materialized per micro-batch.
        if self.overlap is not None:
            self.overlap.begin_step()
        amp_ctx = (torch.autocast(device_type="cuda", dtype=torch.bfloat16)
                   if self.amp else contextlib.nullcontext())
        with amp_ctx:
            for li, worker in enumerate(self.group.worker_ids):
                row = self.local_rows[li]
                row.zero_()
                bind_grad_views(self.params, row)
                batch = self._format_batch(self.experiment.train_batch(
                    worker, self.global_step, self.device))
                self._trace(f"worker {worker}: forward")
                loss = self.experiment.loss(self.model, batch)
                if self.l1 > 0 or self.l2 > 0:
                    with torch.autocast(device_type="cuda", enabled=False) \
                            if self.amp else contextlib.nullcontext():
                        loss = loss.float() + self._regularization()
                self._trace(f"worker {worker}: backward")
                loss.backward()
                losses.append(loss.detach())
        self._apply_attack()
        return torch.stack(losses).mean()

    def _apply_attack(self):
        """Replace the local real-Byzantine workers' rows with crafted ones.

        Runs AFTER all local gradients are computed so omniscient attacks
        (ALIE/IPM) can observe this rank's honest rows. Byzantine worker ids
        are the global prefix 0..nb_real_byz-1, so locally they are the
        first rows and the honest remainder is a zero-copy slice.
        """
        if self.attack is None:
            return
        n_local_byz = sum(1 for w in self.group.worker_ids
                          if w < self.nb_real_byz)
        if n_local_byz == 0:
            return
        if hasattr(self.attack, "observe"):
            self.attack.observe(self.local_rows[n_local_byz:])
        for li in range(n_local_byz):
            worker = self.group.worker_ids[li]
            self._trace(f"worker {worker}: byzantine craft")
            row = self.local_rows[li]
            row.copy_(self.attack.craft(row.clone(), worker,
                                        self.global_step))

    def _gather_matrix(self):
        if self.overlap is not None:
            return self.overlap.finish()
        return self.group.gather(
            self.local_rows, out=self.matrix if self.group.distributed else None)

    def aggregate(self, matrix=None):
        """Gather all rows (unless pre-gathered), verify integrity, inject
        channel loss, run the GAR."""
        if matrix is None:
            self._trace("gather")
            matrix = self._gather_matrix()
        if self.integrity is not None:
            self._trace("integrity check")
            macs_local = self.integrity.sign_rows(
                self.local_rows, self.group.worker_ids, self.global_step)
            macs = torch.cat(self.group.gather_small(macs_local), dim=0)
            failed = self.integrity.verify_matrix(matrix, macs,
                                                  self.global_step)
            if failed:
                tools.warning(f"integrity check failed for workers {failed} "
                              f"at step {self.global_step}; rows NaN-filled")
        if self.lossy is not None:
            self._trace("lossy inject")
            matrix = self.lossy.inject(matrix, self.global_step)
        self._trace(f"aggregate ({type(self.gar).__name__})")
        return self.gar.aggregate(matrix)

    def apply(self, aggregated):
        """Apply the aggregated gradient (identical on every rank)."""
        self._trace("apply")
        self.agg_flat.copy_(aggregated)
        bind_grad_views(self.params, self.agg_flat)
        lr = self.lr_fn(self.global_step)
        for pg in self.optimizer.param_groups:
            pg["lr"] = lr
        self.optimizer.step()
        self.global_step += 1

    def _apply_from_matrix(self):
        """Aggregate ``self.matrix`` and apply -- the capturable tail of a
        step (no lr update, no step counter; the caller owns those)."""
        aggregated = self.gar.aggregate(self.matrix)
        self.agg_flat.copy_(aggregated)
        bind_grad_views(self.params, self.agg_flat)
        self.optimizer.step()

    def step(self, sync_loss=True):
        """One full training step; returns the local mean worker loss.

        ``sync_loss=False`` skips the device->host loss read (no implicit
        synchronization -- benchmark hot loops use this; the NaN-divergence
        check then only sees the value when one is requested).
        """
        with self.lock:
            return self._step_locked(sync_loss)

    def _step_locked(self, sync_loss):
        if self.use_graphs and self._graphstep is None \
                and self.global_step >= self.graph_warmup:
            self._trace("engaging hipGraph step capture")
            # Freeze the (fixed) lr into the captured optimizer step.
            lr = self.lr_fn(self.global_step)
            for pg in self.optimizer.param_groups:
                pg["lr"] = lr
            self._graphstep = self._graph_cls(self)
        if self._graphstep is not None:
            try:
                loss = self._graphstep.step_once()
            except RuntimeError as e:
                # Capture can fail on exotic configs; training must not.
                # Fall back to eager permanently (the warm steps it already
                # ran were full training steps, so the trajectory is intact).
                tools.warning(f"hipGraph step failed ({e}); falling back to "
                              f"eager execution")
                self.use_graphs = False
                self._graphstep = None
                loss = self.compute_local_gradients()
                aggregated = self.aggregate()
                self.apply(aggregated)
        elif self.trace:
            import time
            t = time.monotonic()
            loss = self.compute_local_gradients()
            t = self._phase("local_gradients", t)
            matrix = self._gather_matrix()
            t = self._phase("gather", t)
            aggregated = self.aggregate(matrix)
            t = self._phase("aggregate", t)
            self.apply(aggregated)
            self._phase("apply", t)
        else:
            loss = self.compute_local_gradients()
            aggregated = self.aggregate()
            self.apply(aggregated)
        if sync_loss:
            self.last_loss = loss.item()
            return self.last_loss
        return None

    # ------------------------------------------------------------------ #

    def evaluate(self):
        """Top-1 accuracy on the experiment's eval set.

        Runs on a cached REPLICA of the model: the evaluation service thread
        must neither flip the shared model's train/eval mode while the
        training thread is mid-forward, nor issue collectives (it runs on
        rank 0 only -- a broadcast here would deadlock the other ranks).
        BN running stats are the local rank's, exactly like the reference's
        eval replicas reading the PS variables concurrently with updates.

        Default: serialized with training via the engine lock (pauses
        training for the eval duration -- the sound choice on a shared
        GPU). AGGREGATHOR_CONCURRENT_EVAL=1 restores the reference's
        concurrent-eval capability (runner.py:318-330 allocated separate
        eval DEVICES) on a dedicated HIP stream WITHOUT the lock -- the
        eval replica may then read mid-step (torn) parameters, exactly
        like the reference's eval replicas reading PS variables during
        updates. Gated to runs with no live captured graphs: eval-side
        device allocations permanently corrupt captured MIOpen kernels
        (the round-1 eval-corruption finding, root-caused in
        profiles/graph_purity_bisect.md)."""
        import os
        concurrent = (os.environ.get("AGGREGATHOR_CONCURRENT_EVAL") == "1"
                      and not self.use_graphs and self._graphstep is None)
        if not concurrent:
            with self.lock:
                if self.device.type == "cuda":
                    torch.cuda.synchronize()
                if self._eval_model is None:
                    self._eval_model = self.experiment.model().to(self.device)
                self._eval_model.load_state_dict(self.model.state_dict())
                return self.experiment.accuracy(self._eval_model, self.device)
        if self._eval_model is None:
            self._eval_model = self.experiment.model().to(self.device)
        if self.device.type != "cuda":
            self._eval_model.load_state_dict(self.model.state_dict())
            return self.experiment.accuracy(self._eval_model, self.device)
        if self._eval_stream is None:
            self._eval_stream = torch.cuda.Stream()
        with torch.cuda.stream(self._eval_stream):
            self._eval_model.load_state_dict(self.model.state_dict())
            metrics = self.experiment.accuracy(self._eval_model, self.device)
        self._eval_stream.synchronize()
        return metrics

    def state_dict(self):
        with self.lock:
            if self.device.type == "cuda":
                torch.cuda.synchronize()
            return {
                "step": self.global_step,
                "model": self.model.state_dict(),
                "optimizer": self.optimizer.state_dict(),
            }

    def load_state_dict(self, state):
        self.global_step = state["step"]
        self.model.load_state_dict(state["model"])
        self.optimizer.load_state_dict(state["optimizer"])
        # A restore swaps optimizer-state storages: any captured graph holds
        # stale pointers and must be rebuilt.
        self._graphstep = None
