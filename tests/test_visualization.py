"""Figure/animation generation from synthetic run artifacts.

The reference ships its figures as Jupyter notebooks
(visualization/*.ipynb + visualization/animations/*.ipynb);
visualization/plot_results.py is this framework's headless equivalent
— these tests pin its contract against the metric-file formats the
problems actually write.
"""

import os
import sys

import numpy as np
import pytest
import torch

sys.path.insert(
    0, os.path.join(os.path.dirname(__file__), "..", "visualization")
)
import plot_results  # noqa: E402


def _fake_run(tmp_path, evals=4, nodes=3):
    metrics = {
        "top1_accuracy": [torch.rand(nodes) for _ in range(evals)],
        "validation_loss": [torch.rand(nodes) for _ in range(evals)],
        "consensus_error": [
            (torch.rand(nodes, nodes), torch.rand(nodes, 1))
            for _ in range(evals)
        ],
        "mesh_grid_density": [
            torch.rand(nodes, 144, 1) for _ in range(evals)
        ],
        "current_position": [
            np.random.rand(nodes, 2) * 96 for _ in range(evals)
        ],
    }
    torch.save(metrics, tmp_path / "dinno_results.pt")
    return tmp_path


def test_plot_run_dir(tmp_path):
    run = tmp_path / "run"
    os.makedirs(run)
    _fake_run(run)
    out = tmp_path / "figs"
    plot_results.plot_run_dir(str(run), str(out))
    for f in ("accuracy.png", "val_loss.png", "consensus.png"):
        assert (out / f).exists(), f


def test_animate_run_dir(tmp_path):
    run = tmp_path / "run"
    os.makedirs(run)
    _fake_run(run)
    out = tmp_path / "figs"
    plot_results.animate_run_dir(str(run), str(out))
    frames = list((out / "dinno_frames").glob("*.png"))
    assert len(frames) == 4
    assert (out / "dinno_mesh.gif").exists()


def test_plot_rl_dir(tmp_path):
    tag = "tag_cadmm_0"
    np.save(tmp_path / f"avg_ep_rews_{tag}.npy", np.random.rand(5))
    np.save(tmp_path / f"timesteps_{tag}.npy", np.arange(5) * 100)
    # [iters, N, N] pairwise agreement matrices, as _PPOBase.save writes
    np.savez(
        tmp_path / f"agreements_{tag}.npz",
        actor=np.random.rand(5, 3, 3) + 1e-3,
        critic=np.random.rand(5, 3, 3) + 1e-3,
    )
    out = tmp_path / "figs"
    plot_results.plot_rl_dir(str(tmp_path), str(out))
    assert (out / "rl_rewards.png").exists()
    assert (out / "rl_agreements.png").exists()
