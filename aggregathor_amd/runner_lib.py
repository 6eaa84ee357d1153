"""Training-session driver: the loop + services of the reference's
``runner.py`` session phase (/root/reference/runner.py:497-610).

Provides the hot training loop with NaN-divergence abort
(runner.py:570-574), the evaluation / checkpoint / summary service threads
with step-delta and wall-clock-period dual triggers (runner.py:356-494,
poll delay ``config.thread_idle_delay``), the TSV evaluation file format
(``wall-time \\t step \\t name:value``, runner.py:385-399), and the final
performance self-report (in-step vs off-step time and steps/s,
runner.py:586-598).
"""

import json
import math
import pathlib
import threading
import time

from . import config, tools


class _ServiceThread(threading.Thread):
    """Delta/period dual-trigger service thread (runner.py:417-452 pattern)."""

    def __init__(self, name, fn, get_step, delta, period, stop_event):
        super().__init__(name=name, daemon=True)
        self.fn = fn
        self.get_step = get_step
        self.delta = delta
        self.period = period
        self.stop_event = stop_event
        self.last_step = 0
        self.last_time = time.monotonic()

    def due(self):
        step = self.get_step()
        if self.delta is not None and self.delta >= 0 and step - self.last_step >= self.delta:
            return True
        if self.period is not None and self.period >= 0 and \
                time.monotonic() - self.last_time >= self.period:
            return True
        return False

    def fire(self, final=False):
        step = self.get_step()
        try:
            self.fn(step)
        except Exception as e:  # service failures must not kill training
            tools.warning(f"service {self.name!r} failed at step {step}: {e}")
        self.last_step = step
        self.last_time = time.monotonic()

    def run(self):
        enabled = (self.delta is not None and self.delta >= 0) or \
                  (self.period is not None and self.period >= 0)
        if not enabled:
            return
        while not self.stop_event.wait(config.thread_idle_delay):
            if self.due():
                self.fire()
        self.fire(final=True)  # final service call on shutdown (runner.py:448-452)


class Trainer:
    """Runs an Engine to max_step with eval/checkpoint/summary services."""

    def __init__(self, engine, max_step=config.default_max_step,
                 checkpoint_dir="", checkpoint_delta=config.default_checkpoint_delta,
                 checkpoint_period=config.default_checkpoint_period,
                 summary_dir=None, summary_delta=config.default_summary_delta,
                 summary_period=config.default_summary_period,
                 evaluation_file=None, evaluation_delta=config.default_evaluation_delta,
                 evaluation_period=config.default_evaluation_period,
                 rank0_only_services=True,
                 profile_steps=0, profile_dir="profile_trace"):
        self.engine = engine
        # Reference semantics (runner.py --max-step help): the number of
        # ADDITIONAL steps to perform -- on resume, training continues for
        # max_step more steps past the restored global step.
        self.max_step = max_step
        self.offstep = 0  # step at (re)start
        self.is_rank0 = engine.group.rank == 0
        services = self.is_rank0 or not rank0_only_services

        self.checkpoints = None
        if checkpoint_dir:
            self.checkpoints = tools.Checkpoints(checkpoint_dir)
            if self.checkpoints.can_restore():
                with tools.Context("checkpoint", "info"):
                    payload = self.checkpoints.restore()
                    engine.load_state_dict(payload)
                    tools.info(f"Restored checkpoint at step {engine.global_step}")
            self.offstep = engine.global_step

        # Evaluation TSV file (runner.py:184-187 format).
        self.eval_path = None
        if services:
            if evaluation_file == "-":
                pass
            elif evaluation_file:
                self.eval_path = pathlib.Path(evaluation_file)
            elif checkpoint_dir:
                self.eval_path = pathlib.Path(checkpoint_dir) / config.default_evaluation_file_name
            if self.eval_path:
                self.eval_path.parent.mkdir(parents=True, exist_ok=True)

        # Summary JSONL (replaces TF summaries; scalar series over step).
        self.summary_path = None
        if services:
            if summary_dir == "-":
                pass
            elif summary_dir:
                self.summary_path = pathlib.Path(summary_dir) / "summary.jsonl"
            elif checkpoint_dir:
                self.summary_path = pathlib.Path(checkpoint_dir) / "summary.jsonl"
            if self.summary_path:
                self.summary_path.parent.mkdir(parents=True, exist_ok=True)

        # torch.profiler integration (per-phase profiler spans
        # complement the rocprofv3 kernel evidence): profile `profile_steps`
        # steps after a short warmup and export a chrome trace on rank 0.
        self.profile_steps = profile_steps if self.is_rank0 else 0
        self.profile_dir = profile_dir

        self._stop = threading.Event()
        self._threads = []
        if services:
            if self.eval_path is not None or evaluation_delta >= 0 or evaluation_period >= 0:
                self._threads.append(_ServiceThread(
                    "evaluation", self._do_eval, lambda: engine.global_step,
                    evaluation_delta, evaluation_period, self._stop))
            if self.checkpoints is not None:
                self._threads.append(_ServiceThread(
                    "checkpoint", self._do_checkpoint, lambda: engine.global_step,
                    checkpoint_delta, checkpoint_period, self._stop))
            if self.summary_path is not None:
                self._threads.append(_ServiceThread(
                    "summary", self._do_summary, lambda: engine.global_step,
                    summary_delta, summary_period, self._stop))

    # ------------------------------------------------------------------ #
    # Services

    def _do_eval(self, step):
        metrics = self.engine.evaluate()
        if self.eval_path is not None:
            line = f"{time.time():.3f}\t{step}" + "".join(
                f"\t{k}:{v}" for k, v in metrics.items())
            with self.eval_path.open("a") as f:
                f.write(line + "\n")
        with tools.Context("eval", "info"):
            tools.info(f"Step {step}: " + ", ".join(
                f"{k} = {v:.4f}" for k, v in metrics.items()))
        return metrics

    def _do_checkpoint(self, step):
        path = self.checkpoints.save(self.engine.state_dict(), step)
        with tools.Context("checkpoint", "info"):
            tools.info(f"Saved {path}")

    def _do_summary(self, step):
        rec = {"time": time.time(), "step": step,
               "lr": self.engine.lr_fn(step), "loss": self.engine.last_loss}
        with self.summary_path.open("a") as f:
            f.write(json.dumps(rec) + "\n")

    # ------------------------------------------------------------------ #

    def train(self, progress_every=0):
        """Run to max_step; returns the perf report dict (runner.py:586-598)."""
        engine = self.engine
        profiler = None
        if self.profile_steps > 0:
            import torch.profiler as tp
            profiler = tp.profile(
                activities=[tp.ProfilerActivity.CPU,
                            tp.ProfilerActivity.CUDA],
                schedule=tp.schedule(wait=1, warmup=2,
                                     active=self.profile_steps, repeat=1),
                on_trace_ready=tp.tensorboard_trace_handler(self.profile_dir))
            profiler.__enter__()
        for t in self._threads:
            t.start()
        first_step_time = None
        in_step = 0.0
        steps_done = 0
        t_total0 = time.monotonic()
        diverged = False
        target = self.offstep + self.max_step if self.max_step > 0 else -1
        try:
            while target < 0 or engine.global_step < target:
                t0 = time.monotonic()
                loss = engine.step()
                dt = time.monotonic() - t0
                steps_done += 1
                if first_step_time is None:
                    first_step_time = dt
                in_step += dt
                if profiler is not None:
                    profiler.step()
                if progress_every and steps_done % progress_every == 0 and self.is_rank0:
                    tools.info(f"step {engine.global_step}  loss {loss:.5f}  "
                               f"({dt * 1e3:.1f} ms/step)")
                # NaN-divergence abort (runner.py:570-574). In distributed
                # mode the abort is COLLECTIVE: local losses differ per
                # rank, so a rank-local break would leave the other ranks
                # blocked in the next all-gather until the process-group
                # timeout -- every rank folds its flag in every step and all
                # abort together at the same step.
                local_bad = not math.isfinite(loss)
                if engine.group.any_rank(local_bad):
                    diverged = True
                    detail = f"loss = {loss}" if local_bad else \
                        "a peer rank's loss went non-finite"
                    tools.error(f"Training diverged ({detail}) at step "
                                f"{engine.global_step}; aborting on all ranks")
                    break
        finally:
            if profiler is not None:
                profiler.__exit__(None, None, None)
            self._stop.set()
            for t in self._threads:
                t.join(timeout=30.)
        total = time.monotonic() - t_total0
        report = {
            "steps": steps_done,
            "diverged": diverged,
            "total_time_s": total,
            "in_step_time_s": in_step,
            "off_step_time_s": total - in_step,
            "steps_per_sec_all": steps_done / in_step if in_step > 0 else 0.0,
            "steps_per_sec_excl_first": ((steps_done - 1) / (in_step - first_step_time)
                                         if steps_done > 1 and in_step > first_step_time
                                         else 0.0),
        }
        if self.is_rank0:
            with tools.Context("perf", "info"):
                tools.info(f"In-step time:  {in_step:.3f} s "
                           f"({100 * in_step / total if total else 0:.1f}% of total)")
                tools.info(f"Off-step time: {report['off_step_time_s']:.3f} s")
                tools.info(f"Steps/s (all steps): {report['steps_per_sec_all']:.3f}")
                tools.info(f"Steps/s (excluding first): {report['steps_per_sec_excl_first']:.3f}")
                if getattr(engine, "phase_times", None):
                    for name, secs in engine.phase_times.items():
                        tools.info(f"Phase {name}: {secs:.3f} s "
                                   f"({1e3 * secs / max(steps_done, 1):.2f} ms/step)")
        report["phase_times"] = dict(getattr(engine, "phase_times", {}))
        return report
