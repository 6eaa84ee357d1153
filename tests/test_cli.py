"""CLI integration: runner.py flag surface, checkpoint/eval layout, resume,
deploy.py local multi-process launch."""

import json
import os
import pathlib
import subprocess
import sys

REPO = pathlib.Path(__file__).resolve().parent.parent


def _run(cmd, timeout=300, env_extra=None):
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK",
                        "MASTER_ADDR", "MASTER_PORT")}
    if env_extra:
        env.update(env_extra)
    return subprocess.run([sys.executable] + cmd, capture_output=True,
                          timeout=timeout, env=env, cwd=str(REPO))


def test_runner_mnist_average(tmp_path):
    ckpt = tmp_path / "ckpt"
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator", "average",
              "--nb-workers", "3", "--max-step", "12",
              "--experiment-args", "batch-size:16",
              "--checkpoint-dir", str(ckpt),
              "--checkpoint-delta", "5", "--checkpoint-period", "-1",
              "--evaluation-delta", "6", "--evaluation-period", "-1",
              "--summary-delta", "6", "--summary-period", "-1",
              "--progress-every", "5"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    out = r.stdout.decode()
    assert "Steps/s" in out
    # Checkpoint layout: model-<step>.ckpt files.
    files = sorted(ckpt.glob("model-*.ckpt"))
    assert files, "no checkpoints written"
    # Eval TSV: wall-time \t step \t top1-X-acc:value
    eval_file = ckpt / "eval"
    assert eval_file.exists()
    line = eval_file.read_text().strip().split("\n")[0].split("\t")
    assert len(line) >= 3 and line[2].startswith("top1-X-acc:")
    # Summary JSONL
    recs = [json.loads(l) for l in (ckpt / "summary.jsonl").read_text().splitlines()]
    assert all("loss" in r and "lr" in r for r in recs)


def test_runner_resume(tmp_path):
    ckpt = tmp_path / "ckpt"
    base = ["runner.py", "--experiment", "mnist", "--aggregator", "krum",
            "--nb-workers", "5", "--nb-decl-byz-workers", "1",
            "--experiment-args", "batch-size:16",
            "--checkpoint-dir", str(ckpt),
            "--checkpoint-delta", "4", "--checkpoint-period", "-1",
            "--evaluation-delta", "-1", "--evaluation-period", "-1",
            "--progress-every", "0"]
    r = _run(base + ["--max-step", "8"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    r = _run(base + ["--max-step", "14"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    assert "Restored checkpoint" in r.stdout.decode()
    steps = sorted(int(p.stem.split("-")[1]) for p in ckpt.glob("model-*.ckpt"))
    assert steps[-1] >= 12  # continued past the first run's 8


def test_runner_attack_flags(tmp_path):
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator", "krum",
              "--nb-workers", "5", "--nb-decl-byz-workers", "1",
              "--nb-real-byz-workers", "1", "--attack", "reversal",
              "--attack-args", "factor:2.0",
              "--experiment-args", "batch-size:16",
              "--max-step", "6", "--evaluation-delta", "-1",
              "--evaluation-period", "-1", "--progress-every", "0"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]


def test_runner_lossy_flags(tmp_path):
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator",
              "average-nan", "--nb-workers", "4",
              "--lossy", "drop-rate:0.2", "workers:0", "chunk-bytes:4096",
              "--experiment-args", "batch-size:16",
              "--max-step", "6", "--evaluation-delta", "-1",
              "--evaluation-period", "-1", "--progress-every", "0"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]


def test_runner_rejects_unknown_aggregator():
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator", "nope",
              "--nb-workers", "2", "--max-step", "1"])
    assert r.returncode != 0
    assert b"Unknown GAR" in r.stderr or b"Unknown GAR" in r.stdout


def test_deploy_local_two_ranks(tmp_path):
    r = _run(["deploy.py", "--nproc", "2", "--master-port", "29701", "--",
              "--experiment", "mnist", "--aggregator", "median",
              "--nb-workers", "4", "--experiment-args", "batch-size:16",
              "--max-step", "6", "--evaluation-delta", "-1",
              "--evaluation-period", "-1", "--progress-every", "0"],
             timeout=420)
    assert r.returncode == 0, (r.stdout.decode()[-1500:]
                               + r.stderr.decode()[-1500:])
    assert "all ranks completed" in r.stdout.decode()


def test_runner_torch_profiler(tmp_path):
    prof = tmp_path / "trace"
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator", "average",
              "--nb-workers", "2", "--experiment-args", "batch-size:16",
              "--max-step", "8", "--evaluation-delta", "-1",
              "--evaluation-period", "-1", "--progress-every", "0",
              "--profile-steps", "3", "--profile-dir", str(prof)])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    assert list(prof.glob("*.json*")), "no profiler trace exported"


def test_reference_readme_loopback_invocation(tmp_path):
    """The reference README's documented local-deployment command line
    (reference README.md:146) runs verbatim (with max-step shortened):
    every flag is accepted with the same spelling and semantics."""
    r = _run(["runner.py",
              "--server", '{"local": ["127.0.0.1:7000"]}',
              "--ps-job-name", "local", "--wk-job-name", "local",
              "--ev-job-name", "local",
              "--experiment", "mnist",
              "--learning-rate-args", "initial-rate:0.05",
              "--aggregator", "average", "--nb-workers", "4",
              "--reuse-gpu", "--max-step", "10",
              "--evaluation-period", "-1", "--checkpoint-period", "-1",
              "--summary-period", "-1", "--evaluation-delta", "100",
              "--checkpoint-delta", "-1", "--summary-delta", "-1",
              "--no-wait", "--progress-every", "0"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    assert "Steps/s" in r.stdout.decode()


def test_experiments_sh_smoke(tmp_path):
    """The experiment batch driver runs end to end (shortened steps)."""
    env = _clean = {k: v for k, v in os.environ.items()
                    if k not in ("RANK", "WORLD_SIZE", "LOCAL_RANK")}
    env["MAX_STEP"] = "3"
    r = subprocess.run(["bash", "experiments.sh", str(tmp_path / "out")],
                       capture_output=True, timeout=600, env=env,
                       cwd=str(REPO))
    assert r.returncode == 0, r.stderr.decode()[-1500:]
    out = r.stdout.decode()
    assert "all done" in out
    # Every run produced its redirected stdout log and checkpoint dir.
    logs = list((tmp_path / "out").glob("*.stdout"))
    assert len(logs) == 5, [p.name for p in (tmp_path / "out").iterdir()]


def test_runner_trace_phases(tmp_path):
    r = _run(["runner.py", "--experiment", "mnist", "--aggregator", "median",
              "--nb-workers", "3", "--experiment-args", "batch-size:16",
              "--max-step", "3", "--evaluation-delta", "-1",
              "--evaluation-period", "-1", "--progress-every", "0",
              "--trace"])
    assert r.returncode == 0, r.stderr.decode()[-2000:]
    out = r.stdout.decode()
    for marker in ("forward", "backward", "gather", "aggregate", "apply",
                   "Phase local_gradients"):
        assert marker in out, marker
