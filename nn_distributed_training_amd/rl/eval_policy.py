"""Policy evaluation / rollout visualization for trained tag policies.

Capability parity with the reference's ``RL/dist_rl/eval_policy.py``:
load saved actor state dicts, roll deterministic episodes, report
per-episode rewards; optional matplotlib trajectory plot instead of the
reference's live renderer.
"""

from __future__ import annotations

import argparse

import numpy as np
import torch

from ..models.mlp import FFReLUNet
from .envs import SimpleTagEnv


def load_actors(path, env, hidden=(64, 64, 64), device="cpu"):
    states = torch.load(path, map_location=device, weights_only=False)
    actors = {}
    for i, sd in states.items():
        a = FFReLUNet([env.obs_dim, *hidden, env.act_dim]).to(device)
        a.load_state_dict(sd)
        actors[int(i)] = a
    return actors


def eval_episodes(actors, env, episodes=5, max_steps=100,
                  record_traj=False):
    rews, trajs = [], []
    for _ in range(episodes):
        obs = env.reset()
        total = 0.0
        traj = []
        for _ in range(max_steps):
            obs_t = torch.as_tensor(
                obs, dtype=torch.get_default_dtype()
            )
            with torch.no_grad():
                acts = torch.stack(
                    [actors[i](obs_t[i]) for i in range(env.n)]
                )
            obs, r, done, _ = env.step(acts.numpy())
            total += float(np.mean(r))
            if record_traj:
                traj.append(
                    (env.pred_pos.copy(), env.prey_pos.copy())
                )
            if done:
                break
        rews.append(total)
        trajs.append(traj)
    return rews, trajs


def plot_trajectories(traj, env, out_path):
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(5, 5))
    preds = np.stack([p for p, _ in traj])  # [T, n, 2]
    prey = np.stack([q for _, q in traj])
    for i in range(preds.shape[1]):
        ax.plot(preds[:, i, 0], preds[:, i, 1], "-", lw=1,
                label=f"pred {i}")
    ax.plot(prey[:, 0], prey[:, 1], "k--", lw=1.5, label="prey")
    for o in env.obst_pos:
        ax.add_patch(plt.Circle(o, env.obst_size, color="gray",
                                alpha=0.4))
    ax.set_xlim(-1.3, 1.3)
    ax.set_ylim(-1.3, 1.3)
    ax.legend(fontsize=7)
    fig.savefig(out_path, dpi=120)
    plt.close(fig)


def animate_rollout(traj, env, out_path, interval_ms=80):
    """Render a rollout as an animated GIF (headless stand-in for the
    reference's live pyglet renderer, RL/dist_rl/eval_policy.py —
    predators red, prey green, obstacles gray, MPE colors)."""
    import matplotlib

    matplotlib.use("Agg")
    import matplotlib.animation as animation
    import matplotlib.pyplot as plt

    fig, ax = plt.subplots(figsize=(5, 5))
    ax.set_xlim(-1.4, 1.4)
    ax.set_ylim(-1.4, 1.4)
    ax.set_aspect("equal")
    for o in env.obst_pos:
        ax.add_patch(plt.Circle(o, env.obst_size,
                                color=(0.25, 0.25, 0.25), alpha=0.6))
    pred_c = [
        plt.Circle((0, 0), env.pred_size, color=(0.85, 0.35, 0.35))
        for _ in range(env.n)
    ]
    prey_c = plt.Circle((0, 0), env.prey_size,
                        color=(0.35, 0.85, 0.35))
    for c in pred_c + [prey_c]:
        ax.add_patch(c)
    title = ax.set_title("")

    def update(f):
        preds, prey = traj[f]
        for i, c in enumerate(pred_c):
            c.center = tuple(preds[i])
        prey_c.center = tuple(prey)
        title.set_text(f"step {f}")
        return pred_c + [prey_c, title]

    ani = animation.FuncAnimation(
        fig, update, frames=len(traj), blit=False
    )
    ani.save(out_path, writer=animation.PillowWriter(
        fps=max(1, int(1000 / interval_ms))))
    plt.close(fig)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("actors_path")
    p.add_argument("--episodes", type=int, default=5)
    p.add_argument("--predators", type=int, default=3)
    p.add_argument("--plot", default=None)
    p.add_argument("--animate", default=None,
                   help="write a rollout animation GIF here")
    args = p.parse_args(argv)

    env = SimpleTagEnv(num_predators=args.predators)
    actors = load_actors(args.actors_path, env)
    rews, trajs = eval_episodes(
        actors, env, episodes=args.episodes,
        record_traj=args.plot is not None or args.animate is not None,
    )
    print("episodic rewards:", [f"{r:.2f}" for r in rews])
    print(f"mean: {np.mean(rews):.2f}")
    if args.plot:
        plot_trajectories(trajs[0], env, args.plot)
        print("trajectory plot ->", args.plot)
    if args.animate:
        animate_rollout(trajs[0], env, args.animate)
        print("rollout animation ->", args.animate)


if __name__ == "__main__":
    main()
