"""Multi-agent decentralized PPO driver.

Capability parity with the reference's
``RL/dist_rl/train_{cadmm,dsgd,dsgt}_multi.py`` (one file per algorithm
with hardcoded configs; here one driver + --alg flag and the same
defaults: 3 predators on a wheel graph, heuristic prey, obstacles).
Run: python -m nn_distributed_training_amd.rl.train_multi --alg dinno
"""

from __future__ import annotations

import argparse

import networkx as nx
import torch

from .dist_ppo import DistPPOProblem
from .envs import SimpleTagEnv
from .ppo_optimizers import build_ppo_optimizer


def default_conf(alg: str) -> dict:
    conf = {
        "timesteps_per_batch": 600,
        "max_timesteps_per_episode": 100,
        "gamma": 0.99,
        "clip": 0.2,
        "cov": 0.5,
        "max_rl_timesteps": 100_000,
        "save_freq": 10,
        "verbose": True,
        "run_id": "0",
        "output_dir": "./trained",
    }
    if alg in ("dinno", "cadmm"):
        conf.update(
            rho_init=0.3, rho_scaling=1.0, primal_iterations=5,
            primal_lr_start=1e-3, primal_lr_finish=1e-4,
            lr_decay_type="log", expected_iterations=200,
        )
    elif alg == "dsgd":
        conf.update(alpha0=1e-3, mu=1e-3)
    elif alg == "dsgt":
        conf.update(alpha_actor=1e-3, alpha_critic=1e-3)
    return conf


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--alg", default="dinno",
                   choices=["dinno", "cadmm", "dsgd", "dsgt"])
    p.add_argument("--predators", type=int, default=3)
    p.add_argument("--obstacles", type=int, default=2)
    p.add_argument("--max-timesteps", type=int, default=None)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--out", default="./trained")
    args = p.parse_args(argv)

    torch.manual_seed(args.seed)
    conf = default_conf(args.alg)
    conf["output_dir"] = args.out
    if args.max_timesteps:
        conf["max_rl_timesteps"] = args.max_timesteps

    env = SimpleTagEnv(
        num_predators=args.predators, num_obstacles=args.obstacles,
        seed=args.seed,
    )
    graph = nx.wheel_graph(args.predators)
    device = torch.device(
        "cuda" if torch.cuda.is_available() else "cpu"
    )
    pr = DistPPOProblem(graph, env, device, conf)
    opt = build_ppo_optimizer(args.alg, pr, conf)
    opt.train()
    print(f"final eval (deterministic): {pr.evaluate():.2f}")


if __name__ == "__main__":
    main()
