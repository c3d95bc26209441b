"""Subprocess smoke tests for the command-line entry points.

Functions are unit-tested elsewhere; these guard the argv wiring and
module `__main__` blocks the way a user actually invokes them.
"""

import os
import subprocess
import sys

REPO = os.path.join(os.path.dirname(__file__), "..")


def _run(args, timeout=240):
    r = subprocess.run(
        args, cwd=REPO, capture_output=True, text=True, timeout=timeout,
        env={**os.environ, "PYTHONPATH": REPO},
    )
    assert r.returncode == 0, r.stderr[-1500:]
    return r.stdout


def test_eval_policy_cli(tmp_path):
    out = _run([
        sys.executable, "-m",
        "nn_distributed_training_amd.rl.eval_policy",
        "examples/rl_trained/ppo_actors_tag_cadmm_0.pth",
        "--episodes", "1",
        "--plot", str(tmp_path / "traj.png"),
    ])
    assert "episodic rewards" in out
    assert (tmp_path / "traj.png").exists()


def test_plot_results_cli_on_committed_artifacts(tmp_path):
    out = _run([
        sys.executable, "visualization/plot_results.py",
        "examples/scaling_gpu/run", "--out", str(tmp_path),
        "--rl", "examples/rl_trained",
    ])
    assert "figures ->" in out
    assert (tmp_path / "consensus.png").exists()
    assert (tmp_path / "rl_rewards.png").exists()


def test_sdf_preproc_cli(tmp_path):
    out = _run([
        sys.executable, "-m",
        "nn_distributed_training_amd.data.sdf_preproc",
        "--out", str(tmp_path / "sdf"), "--count", "2", "--size", "64",
    ])
    assert "wrote 2 SDFs" in out


def test_bench_help():
    out = _run([sys.executable, "bench.py", "--help"])
    assert "--workload" in out and "--gpus" in out
