"""Figure generation from experiment result files.

The reference produces its paper figures from Jupyter notebooks
(visualization/*.ipynb) reading ``<problem>_results.pt``; this script
covers the same plots headlessly:

  * top-1 accuracy band (min-max over nodes) vs communication rounds
  * validation-loss band
  * consensus error (log scale) vs rounds
  * scaling summary (rounds/sec and convergence vs N) from
    dist_mnist_scaling's ``scaling_summary.pt``
  * RL reward / agreement curves from the trained/ npy+npz artifacts

Usage:
  python visualization/plot_results.py <run_dir> [--out figs/]
  python visualization/plot_results.py --rl trained/ [--out figs/]
"""

from __future__ import annotations

import argparse
import glob
import os

import matplotlib

matplotlib.use("Agg")
import matplotlib.pyplot as plt  # noqa: E402
import numpy as np  # noqa: E402
import torch  # noqa: E402


def _band(ax, xs, series, label):
    arr = torch.stack(
        [torch.as_tensor(s, dtype=torch.float64) for s in series]
    )
    lo = arr.amin(dim=1)
    hi = arr.amax(dim=1)
    mid = arr.mean(dim=1)
    ax.plot(xs, mid, label=label)
    ax.fill_between(xs, lo, hi, alpha=0.25)


def plot_run_dir(run_dir: str, out_dir: str, eval_every: int = 20):
    os.makedirs(out_dir, exist_ok=True)
    results = sorted(
        rp
        for rp in glob.glob(os.path.join(run_dir, "*_results.pt"))
        # baselines have scalar-per-epoch curves, not per-node rows
        if os.path.basename(rp) not in ("solo_results.pt",
                                        "centralized_results.pt")
    )
    if not results:
        print(f"no *_results.pt under {run_dir}")
        return

    for metric, fname, ylabel, logy in (
        ("top1_accuracy", "accuracy.png", "top-1 accuracy", False),
        ("validation_loss", "val_loss.png", "validation loss", False),
    ):
        fig, ax = plt.subplots(figsize=(6, 4))
        drew = False
        for rp in results:
            name = os.path.basename(rp).replace("_results.pt", "")
            res = torch.load(rp, weights_only=False)
            if metric not in res or not res[metric]:
                continue
            xs = np.arange(len(res[metric])) * eval_every
            _band(ax, xs, res[metric], name)
            drew = True
        if drew:
            ax.set_xlabel("communication rounds")
            ax.set_ylabel(ylabel)
            if logy:
                ax.set_yscale("log")
            ax.legend()
            fig.tight_layout()
            fig.savefig(os.path.join(out_dir, fname), dpi=130)
        plt.close(fig)

    # consensus error (max pairwise distance, log scale)
    fig, ax = plt.subplots(figsize=(6, 4))
    drew = False
    for rp in results:
        name = os.path.basename(rp).replace("_results.pt", "")
        res = torch.load(rp, weights_only=False)
        if "consensus_error" not in res or not res["consensus_error"]:
            continue
        vals = [
            float(torch.as_tensor(c[1]).amax())
            for c in res["consensus_error"]
        ]
        xs = np.arange(len(vals)) * eval_every
        ax.plot(xs, np.maximum(vals, 1e-12), label=name)
        drew = True
    if drew:
        ax.set_xlabel("communication rounds")
        ax.set_ylabel("max consensus error")
        ax.set_yscale("log")
        ax.legend()
        fig.tight_layout()
        fig.savefig(os.path.join(out_dir, "consensus.png"), dpi=130)
    plt.close(fig)

    # mnist_four-style comparability figure: centralized (pooled-data
    # upper bound) + per-node solo baselines + all decentralized
    # curves on one accuracy axis (reference
    # visualization/mnist_four.ipynb cells 1-5)
    cent_path = os.path.join(run_dir, "centralized_results.pt")
    solo_path = os.path.join(run_dir, "solo_results.pt")
    if os.path.exists(cent_path) or os.path.exists(solo_path):
        for metric, fname, ylabel in (
            ("top1_accuracy", "four_accuracy.png", "top-1 accuracy"),
            ("validation_loss", "four_val_loss.png",
             "validation loss"),
        ):
            fig, ax = plt.subplots(figsize=(6.5, 4))
            drew = False
            for rp in results:
                name = os.path.basename(rp).replace("_results.pt", "")
                res = torch.load(rp, weights_only=False)
                if metric not in res or not res[metric]:
                    continue
                xs = np.arange(len(res[metric])) * eval_every
                _band(ax, xs, res[metric], name)
                drew = True
            key = ("validation_accuracy"
                   if metric == "top1_accuracy" else metric)
            if os.path.exists(cent_path):
                cent = torch.load(cent_path, weights_only=False)
                if key in cent and cent[key]:
                    ax.axhline(cent[key][-1], color="k", ls="--",
                               label="centralized (pooled)")
                    drew = True
            if os.path.exists(solo_path):
                solo = torch.load(solo_path, weights_only=False)
                vals = [
                    v[key] for v in solo.values() if key in v
                ]
                if vals:
                    ax.axhline(np.mean(vals), color="gray", ls=":",
                               label="solo mean")
                    drew = True
            if drew:
                ax.set_xlabel("communication rounds")
                ax.set_ylabel(ylabel)
                ax.legend(fontsize=8)
                fig.tight_layout()
                fig.savefig(os.path.join(out_dir, fname), dpi=130)
            plt.close(fig)

    summary = os.path.join(run_dir, "scaling_summary.pt")
    if os.path.exists(summary):
        s = torch.load(summary, weights_only=False)
        fig, ax = plt.subplots(figsize=(6, 4))
        by_alg = {}
        for (t, alg), d in s.items():
            by_alg.setdefault(alg, []).append(
                (d["N"], d["rounds_per_sec"])
            )
        for alg, pts in by_alg.items():
            pts.sort()
            ax.plot([p[0] for p in pts], [p[1] for p in pts], "o-",
                    label=alg)
        ax.set_xlabel("graph nodes N")
        ax.set_ylabel("comm rounds / sec")
        ax.legend()
        fig.tight_layout()
        fig.savefig(os.path.join(out_dir, "scaling.png"), dpi=130)
        plt.close(fig)
    print(f"figures -> {out_dir}")


def plot_rl_dir(rl_dir: str, out_dir: str):
    os.makedirs(out_dir, exist_ok=True)
    fig, ax = plt.subplots(figsize=(6, 4))
    for rews_f in sorted(glob.glob(
        os.path.join(rl_dir, "avg_ep_rews_*.npy")
    )):
        tag = os.path.basename(rews_f)[len("avg_ep_rews_"):-4]
        rews = np.load(rews_f)
        ts_f = os.path.join(rl_dir, f"timesteps_{tag}.npy")
        xs = np.load(ts_f) if os.path.exists(ts_f) \
            else np.arange(len(rews))
        ax.plot(xs, rews, label=tag)
    ax.set_xlabel("environment timesteps")
    ax.set_ylabel("avg episodic reward")
    ax.legend(fontsize=7)
    fig.tight_layout()
    fig.savefig(os.path.join(out_dir, "rl_rewards.png"), dpi=130)
    plt.close(fig)

    fig, ax = plt.subplots(figsize=(6, 4))
    for ag_f in sorted(glob.glob(
        os.path.join(rl_dir, "agreements_*.npz")
    )):
        tag = os.path.basename(ag_f)[len("agreements_"):-4]
        ag = np.load(ag_f)
        ax.plot(ag["actor"].max(axis=(1, 2)), label=f"{tag} actor")
        ax.plot(ag["critic"].max(axis=(1, 2)), "--",
                label=f"{tag} critic")
    ax.set_xlabel("iteration")
    ax.set_ylabel("max agreement distance")
    ax.set_yscale("log")
    ax.legend(fontsize=7)
    fig.tight_layout()
    fig.savefig(os.path.join(out_dir, "rl_agreements.png"), dpi=130)
    plt.close(fig)
    print(f"RL figures -> {out_dir}")


def animate_run_dir(run_dir: str, out_dir: str, node: int = 0,
                    fps: int = 8, world_size: float = 256.0):
    """Render the per-eval mesh frames of an anim-config run.

    Covers the reference's visualization/animations/{density_anim,
    mnist_anim}.ipynb role: for every ``<problem>_results.pt`` whose
    metrics carry per-eval ``mesh_grid_density`` frames (configs with
    ``mesh_only_at_end: false``, e.g. configs/dist_online_dense_anim.
    yaml), write one PNG per evaluation for ``node`` (robot positions
    overlaid when ``current_position`` was recorded) plus an animated
    GIF. Frames hold the per-rank LOCAL nodes — run the anim configs
    single-rank for all-node frames.
    """
    os.makedirs(out_dir, exist_ok=True)
    for res in sorted(glob.glob(os.path.join(run_dir, "*_results.pt"))):
        name = os.path.basename(res)[: -len("_results.pt")]
        metrics = torch.load(res, map_location="cpu", weights_only=False)
        frames = [
            f for f in metrics.get("mesh_grid_density", [])
            if f.numel() > 0
        ]
        if len(frames) < 2:
            continue  # PAPER configs only record the final mesh
        positions = metrics.get("current_position", [])
        side = int(round(frames[0].shape[1] ** 0.5))
        fdir = os.path.join(out_dir, f"{name}_frames")
        os.makedirs(fdir, exist_ok=True)
        paths = []
        half = world_size / 2.0  # lidar world box is [-nx/2, nx/2]
        for t, fr in enumerate(frames):
            img = fr[node].reshape(side, side).numpy()
            fig, ax = plt.subplots(figsize=(4, 4))
            ax.imshow(img, origin="lower", cmap="viridis",
                      vmin=0.0, vmax=1.0,
                      extent=[-half, half, -half, half])
            if t < len(positions) and positions[t] is not None:
                pos = np.asarray(positions[t])  # world coords
                ax.scatter(pos[:, 0], pos[:, 1], c="red", s=12,
                           marker="o")
            ax.set_title(f"{name} node {node} eval {t}")
            ax.set_axis_off()
            fig.tight_layout()
            path = os.path.join(fdir, f"{t:04d}.png")
            fig.savefig(path, dpi=110)
            plt.close(fig)
            paths.append(path)
        try:
            from PIL import Image

            ims = [Image.open(p) for p in paths]
            gif = os.path.join(out_dir, f"{name}_mesh.gif")
            ims[0].save(gif, save_all=True, append_images=ims[1:],
                        duration=int(1000 / fps), loop=0)
            print(f"{len(paths)} frames -> {fdir}, gif -> {gif}")
        except ImportError:
            print(f"{len(paths)} frames -> {fdir} (no pillow: gif "
                  "skipped)")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("run_dir", nargs="?", default=None)
    p.add_argument("--rl", default=None)
    p.add_argument("--animate", default=None,
                   help="run dir of an anim config; renders mesh frames")
    p.add_argument("--node", type=int, default=0)
    p.add_argument("--world-size", type=float, default=256.0,
                   help="lidar world box side (= floorplan_size)")
    p.add_argument("--out", default="./figs")
    p.add_argument("--eval-every", type=int, default=20)
    args = p.parse_args()
    if args.rl:
        plot_rl_dir(args.rl, args.out)
    if args.animate:
        animate_run_dir(args.animate, args.out, node=args.node,
                        world_size=args.world_size)
    if args.run_dir:
        plot_run_dir(args.run_dir, args.out, args.eval_every)
