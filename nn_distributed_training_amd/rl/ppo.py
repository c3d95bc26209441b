"""Single-agent PPO baseline (centralized comparison curves).

Capability parity with the reference's ``RL/ppo.py`` (the classic
from-scratch PPO: rollout -> rewards-to-go -> n update epochs of clipped
surrogate + critic MSE with separate Adam optimizers). The default
environment is the tag game with a single learning predator.
"""

from __future__ import annotations

import numpy as np
import torch
from torch.distributions import MultivariateNormal

from ..models.mlp import FFReLUNet
from .envs import SimpleTagEnv


class PPO:
    def __init__(self, env=None, device="cpu", **hyperparams):
        self.env = env or SimpleTagEnv(num_predators=1)
        self.device = torch.device(device)
        h = {
            "timesteps_per_batch": 800,
            "max_timesteps_per_episode": 100,
            "n_updates_per_iteration": 5,
            "gamma": 0.99,
            "clip": 0.2,
            "lr": 3e-3,
            "cov": 0.5,
            "hidden": (64, 64, 64),
            "save_freq": 0,
            "verbose": True,
        }
        h.update(hyperparams)
        self.h = h

        obs_dim, act_dim = self.env.obs_dim, self.env.act_dim
        hid = list(h["hidden"])
        self.actor = FFReLUNet([obs_dim, *hid, act_dim]).to(self.device)
        self.critic = FFReLUNet([obs_dim, *hid, 1]).to(self.device)
        self.actor_opt = torch.optim.Adam(
            self.actor.parameters(), lr=h["lr"]
        )
        self.critic_opt = torch.optim.Adam(
            self.critic.parameters(), lr=h["lr"]
        )
        self.cov_mat = torch.eye(act_dim, device=self.device) * h["cov"]
        self.avg_ep_rews = []

    # ------------------------------------------------------------------
    def get_action(self, obs_t):
        mean = self.actor(obs_t)
        dist = MultivariateNormal(mean, self.cov_mat)
        a = dist.sample()
        return a, dist.log_prob(a)

    def rollout(self):
        h = self.h
        obs_b, act_b, logp_b, rews_eps = [], [], [], []
        ep_totals = []
        t = 0
        while t < h["timesteps_per_batch"]:
            obs = self.env.reset()
            ep = []
            for _ in range(h["max_timesteps_per_episode"]):
                t += 1
                obs_t = torch.as_tensor(
                    obs[0], dtype=torch.get_default_dtype(),
                    device=self.device,
                )
                with torch.no_grad():
                    a, lp = self.get_action(obs_t)
                obs, rews, done, _ = self.env.step(
                    a.cpu().numpy().reshape(1, -1)
                )
                obs_b.append(obs_t)
                act_b.append(a)
                logp_b.append(lp)
                ep.append(float(rews[0]))
                if done or t >= h["timesteps_per_batch"]:
                    break
            rews_eps.append(ep)
            ep_totals.append(sum(ep))
        self.avg_ep_rews.append(float(np.mean(ep_totals)))
        rtgs = []
        for ep in rews_eps:
            run = 0.0
            out = []
            for r in reversed(ep):
                run = r + self.h["gamma"] * run
                out.append(run)
            rtgs.extend(reversed(out))
        return (
            torch.stack(obs_b),
            torch.stack(act_b),
            torch.stack(logp_b),
            torch.as_tensor(
                rtgs, dtype=torch.get_default_dtype(),
                device=self.device,
            ),
            t,
        )

    # ------------------------------------------------------------------
    def learn(self, total_timesteps):
        t_done = 0
        it = 0
        while t_done < total_timesteps:
            obs, acts, logp, rtgs, t = self.rollout()
            t_done += t
            it += 1
            with torch.no_grad():
                v = self.critic(obs).squeeze(-1)
            adv = rtgs - v
            adv = (adv - adv.mean()) / (adv.std() + 1e-10)

            for _ in range(self.h["n_updates_per_iteration"]):
                mean = self.actor(obs)
                dist = MultivariateNormal(mean, self.cov_mat)
                cur_logp = dist.log_prob(acts)
                ratios = torch.exp(cur_logp - logp)
                clip = self.h["clip"]
                s1 = ratios * adv
                s2 = torch.clamp(ratios, 1 - clip, 1 + clip) * adv
                actor_loss = -torch.min(s1, s2).mean()
                v = self.critic(obs).squeeze(-1)
                critic_loss = torch.nn.functional.mse_loss(v, rtgs)

                self.actor_opt.zero_grad()
                actor_loss.backward()
                self.actor_opt.step()
                self.critic_opt.zero_grad()
                critic_loss.backward()
                self.critic_opt.step()

            if self.h["verbose"]:
                print(
                    f"[ppo] iter {it} t={t_done} "
                    f"avg_ep_rew={self.avg_ep_rews[-1]:.2f}",
                    flush=True,
                )
        return self.avg_ep_rews
