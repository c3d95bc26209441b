"""Single-agent PPO CLI (train/test modes).

Capability parity with the reference's ``RL/main.py`` + ``RL/arguments.py``:
train a centralized PPO baseline on the single-predator tag game, save
actor/critic, or roll deterministic evaluation episodes from saved
weights.

    python -m nn_distributed_training_amd.rl.main --mode train \
        --timesteps 50000 --out ./trained_solo
    python -m nn_distributed_training_amd.rl.main --mode test \
        --actor ./trained_solo/ppo_actor.pth
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import torch

from .envs import SimpleTagEnv
from .ppo import PPO


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--mode", default="train", choices=["train", "test"])
    p.add_argument("--timesteps", type=int, default=50_000)
    p.add_argument("--out", default="./trained_solo")
    p.add_argument("--actor", default=None,
                   help="actor .pth for --mode test")
    p.add_argument("--episodes", type=int, default=5)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)

    torch.manual_seed(args.seed)
    env = SimpleTagEnv(num_predators=1, seed=args.seed)

    if args.mode == "train":
        agent = PPO(env)
        rews = agent.learn(args.timesteps)
        os.makedirs(args.out, exist_ok=True)
        torch.save(agent.actor.state_dict(),
                   os.path.join(args.out, "ppo_actor.pth"))
        torch.save(agent.critic.state_dict(),
                   os.path.join(args.out, "ppo_critic.pth"))
        np.save(os.path.join(args.out, "avg_ep_rews.npy"),
                np.asarray(rews))
        print(f"saved actor/critic + reward curve -> {args.out}")
        return

    assert args.actor, "--mode test needs --actor"
    agent = PPO(env)
    agent.actor.load_state_dict(
        torch.load(args.actor, map_location="cpu", weights_only=False)
    )
    totals = []
    for _ in range(args.episodes):
        obs = env.reset()
        tot = 0.0
        for _ in range(agent.h["max_timesteps_per_episode"]):
            obs_t = torch.as_tensor(
                obs[0], dtype=torch.get_default_dtype()
            )
            with torch.no_grad():
                a = agent.actor(obs_t)
            obs, r, done, _ = env.step(a.numpy().reshape(1, -1))
            tot += float(r[0])
            if done:
                break
        totals.append(tot)
    print("episodic rewards:", [f"{t:.2f}" for t in totals])
    print(f"mean: {np.mean(totals):.2f}")


if __name__ == "__main__":
    main()
