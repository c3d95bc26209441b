"""Distributed multi-agent PPO problem (predator-prey tag).

Capability parity with the reference's ``RL/dist_rl/dist_ppo.py``: N
predators = N graph nodes, each holding an actor+critic pair; rollouts
step ONE shared environment with every node's current actor; rewards-to-
go, advantage normalization, clipped-surrogate + MSE-critic loss, and
Gaussian action sampling with a fixed covariance (cov 0.5, reference
dist_ppo.py:325-351). Hyperparameters come from a plain config dict (the
reference's ``exec``-based injection, dist_ppo.py:426-427, is not
reproduced).
"""

from __future__ import annotations

import copy

import numpy as np
import torch
from torch.distributions import MultivariateNormal

from ..models.mlp import FFReLUNet
from .envs import SimpleTagEnv

DEFAULTS = dict(
    timesteps_per_batch=600,
    max_timesteps_per_episode=100,
    gamma=0.99,
    clip=0.2,
    critic_coef=0.5,
    cov=0.5,
    hidden=(64, 64, 64),
)


class DistPPOProblem:
    def __init__(self, graph, env: SimpleTagEnv, device, conf: dict):
        self.graph = graph
        self.env = env
        self.device = torch.device(device)
        self.conf = {**DEFAULTS, **conf}

        self.N = graph.number_of_nodes()
        assert self.N == env.n, "one predator per graph node"

        obs_dim, act_dim = env.obs_dim, env.act_dim
        hid = list(self.conf["hidden"])
        base_actor = FFReLUNet([obs_dim, *hid, act_dim])
        base_critic = FFReLUNet([obs_dim, *hid, 1])
        self.actors = {
            i: copy.deepcopy(base_actor).to(self.device)
            for i in range(self.N)
        }
        self.critics = {
            i: copy.deepcopy(base_critic).to(self.device)
            for i in range(self.N)
        }
        self.n_actor = sum(p.numel() for p in base_actor.parameters())
        self.n_critic = sum(p.numel() for p in base_critic.parameters())
        self.n = self.n_actor + self.n_critic

        cov = self.conf["cov"]
        self.cov_mat = torch.eye(act_dim, device=self.device) * cov

        # rollout buffers (per node), populated by rollout()
        self.buf = None
        self.total_timesteps = 0
        self.ep_rew_history = []  # mean episodic reward per batch

    # ------------------------------------------------------------------
    def node_vector(self, i) -> torch.Tensor:
        """Concatenated [actor | critic] flat parameters of node i."""
        return torch.cat(
            [
                torch.nn.utils.parameters_to_vector(
                    self.actors[i].parameters()
                ),
                torch.nn.utils.parameters_to_vector(
                    self.critics[i].parameters()
                ),
            ]
        )

    def set_node_vector(self, i, vec: torch.Tensor):
        torch.nn.utils.vector_to_parameters(
            vec[: self.n_actor], self.actors[i].parameters()
        )
        torch.nn.utils.vector_to_parameters(
            vec[self.n_actor :], self.critics[i].parameters()
        )

    def node_parameters(self, i):
        return list(self.actors[i].parameters()) + list(
            self.critics[i].parameters()
        )

    # ------------------------------------------------------------------
    def get_action(self, i, obs_t: torch.Tensor):
        mean = self.actors[i](obs_t)
        dist = MultivariateNormal(mean, self.cov_mat)
        act = dist.sample()
        return act, dist.log_prob(act)

    # ------------------------------------------------------------------
    def rollout(self):
        """Collect one on-policy batch by stepping the shared env.

        Parity with reference split_rollout_marl (dist_ppo.py:171-293):
        serial env stepping, per-predator trajectories, rewards-to-go,
        advantage = rtg - V(obs) normalized per node.
        """
        conf = self.conf
        T = conf["timesteps_per_batch"]
        obs_buf = [[] for _ in range(self.N)]
        act_buf = [[] for _ in range(self.N)]
        logp_buf = [[] for _ in range(self.N)]
        rews_eps = [[] for _ in range(self.N)]  # list of per-ep lists
        ep_total_rews = []

        t = 0
        while t < T:
            obs = self.env.reset()
            ep_rews = [[] for _ in range(self.N)]
            for _ in range(conf["max_timesteps_per_episode"]):
                t += 1
                obs_t = torch.as_tensor(
                    obs, dtype=torch.get_default_dtype(),
                    device=self.device,
                )
                acts, logps = [], []
                with torch.no_grad():
                    for i in range(self.N):
                        a, lp = self.get_action(i, obs_t[i])
                        acts.append(a)
                        logps.append(lp)
                a_np = torch.stack(acts).cpu().numpy()
                nobs, rews, done, _ = self.env.step(a_np)
                for i in range(self.N):
                    obs_buf[i].append(obs_t[i])
                    act_buf[i].append(acts[i])
                    logp_buf[i].append(logps[i])
                    ep_rews[i].append(float(rews[i]))
                obs = nobs
                if done or t >= T:
                    break
            for i in range(self.N):
                rews_eps[i].append(ep_rews[i])
            ep_total_rews.append(
                float(np.mean([sum(r) for r in ep_rews]))
            )

        self.total_timesteps += t
        self.ep_rew_history.append(float(np.mean(ep_total_rews)))

        gamma = conf["gamma"]
        self.buf = {}
        for i in range(self.N):
            rtgs = []
            for ep in rews_eps[i]:
                run = 0.0
                ep_rtgs = []
                for r in reversed(ep):
                    run = r + gamma * run
                    ep_rtgs.append(run)
                rtgs.extend(reversed(ep_rtgs))
            obs_i = torch.stack(obs_buf[i])
            acts_i = torch.stack(act_buf[i])
            logp_i = torch.stack(logp_buf[i])
            rtgs_i = torch.as_tensor(
                rtgs, dtype=torch.get_default_dtype(),
                device=self.device,
            )
            with torch.no_grad():
                v = self.critics[i](obs_i).squeeze(-1)
            adv = rtgs_i - v
            adv = (adv - adv.mean()) / (adv.std() + 1e-10)
            self.buf[i] = dict(
                obs=obs_i, acts=acts_i, logp=logp_i, rtgs=rtgs_i,
                adv=adv,
            )

    # ------------------------------------------------------------------
    def local_batch_loss(self, i):
        """Clipped PPO surrogate + critic MSE on node i's rollout
        (reference ev_ppo_loss, dist_ppo.py:128-156)."""
        b = self.buf[i]
        mean = self.actors[i](b["obs"])
        dist = MultivariateNormal(mean, self.cov_mat)
        logp = dist.log_prob(b["acts"])
        ratios = torch.exp(logp - b["logp"])
        clip = self.conf["clip"]
        s1 = ratios * b["adv"]
        s2 = torch.clamp(ratios, 1 - clip, 1 + clip) * b["adv"]
        actor_loss = -torch.min(s1, s2).mean()
        v = self.critics[i](b["obs"]).squeeze(-1)
        critic_loss = torch.nn.functional.mse_loss(v, b["rtgs"])
        return actor_loss + self.conf["critic_coef"] * critic_loss

    # ------------------------------------------------------------------
    def agreement(self):
        """Normalized pairwise parameter distances (actor, critic)."""
        with torch.no_grad():
            av = torch.stack(
                [
                    torch.nn.utils.parameters_to_vector(
                        self.actors[i].parameters()
                    )
                    for i in range(self.N)
                ]
            )
            cv = torch.stack(
                [
                    torch.nn.utils.parameters_to_vector(
                        self.critics[i].parameters()
                    )
                    for i in range(self.N)
                ]
            )
            an = torch.nn.functional.normalize(av, dim=1)
            cn = torch.nn.functional.normalize(cv, dim=1)
            return (
                torch.cdist(an, an).cpu(),
                torch.cdist(cn, cn).cpu(),
            )

    # ------------------------------------------------------------------
    def evaluate(self, episodes=3):
        """Mean episodic reward with deterministic (mean) actions."""
        totals = []
        for _ in range(episodes):
            obs = self.env.reset()
            tot = 0.0
            for _ in range(self.conf["max_timesteps_per_episode"]):
                obs_t = torch.as_tensor(
                    obs, dtype=torch.get_default_dtype(),
                    device=self.device,
                )
                with torch.no_grad():
                    a = torch.stack(
                        [
                            self.actors[i](obs_t[i])
                            for i in range(self.N)
                        ]
                    )
                obs, rews, done, _ = self.env.step(a.cpu().numpy())
                tot += float(np.mean(rews))
                if done:
                    break
            totals.append(tot)
        return float(np.mean(totals))
