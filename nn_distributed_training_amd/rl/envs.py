"""Minimal multi-agent particle environments (MPE-style).

The reference vendors PettingZoo 1.10's MPE ``simple_tag`` (an external
dependency, RL/README.md:31-33 there); this environment is unavailable
offline, so a compact native implementation of the same game ships here:
N predators (the learning graph nodes), one heuristic-evading prey, and
circular obstacles in a bounded 2-D world with MPE dynamics (force
actions, damping, elastic collision forces).

Observation layout per predator mirrors MPE simple_tag's adversary
observation: [self_vel(2), self_pos(2), obstacle_rel(2*n_obst),
other_predators_rel(2*(n_pred-1)), prey_rel(2), prey_vel(2)].
Actions are MPE's 5-dim continuous force parameterization
(u_x = a[1]-a[2], u_y = a[3]-a[4]; a[0] is a no-op channel).
"""

from __future__ import annotations

import numpy as np

DT = 0.1
DAMPING = 0.25
CONTACT_MARGIN = 0.001
CONTACT_FORCE = 100.0


class SimpleTagEnv:
    """Predator-prey tag. Predators are externally controlled (one per
    graph node); the prey runs a fixed evader heuristic (parity with the
    reference's heuristic evader, RL/dist_rl/dist_ppo.py:79-126)."""

    def __init__(self, num_predators=3, num_obstacles=2, seed=0,
                 max_steps=100):
        self.n = num_predators
        self.n_obst = num_obstacles
        self.rng = np.random.default_rng(seed)
        self.max_steps = max_steps
        self.pred_size = 0.075
        self.prey_size = 0.05
        self.obst_size = 0.2
        self.pred_accel = 3.0
        self.prey_accel = 4.0
        self.pred_max_speed = 1.0
        self.prey_max_speed = 1.3
        self.obs_dim = 2 + 2 + 2 * self.n_obst + 2 * (self.n - 1) + 2 + 2
        self.act_dim = 5
        self.reset()

    # ------------------------------------------------------------------
    def reset(self):
        self.t = 0
        self.pred_pos = self.rng.uniform(-1, 1, size=(self.n, 2))
        self.pred_vel = np.zeros((self.n, 2))
        self.prey_pos = self.rng.uniform(-1, 1, size=2)
        self.prey_vel = np.zeros(2)
        self.obst_pos = self.rng.uniform(-0.9, 0.9,
                                         size=(self.n_obst, 2))
        return self._observations()

    # ------------------------------------------------------------------
    def _observations(self):
        obs = []
        for i in range(self.n):
            parts = [self.pred_vel[i], self.pred_pos[i]]
            for o in range(self.n_obst):
                parts.append(self.obst_pos[o] - self.pred_pos[i])
            for j in range(self.n):
                if j != i:
                    parts.append(self.pred_pos[j] - self.pred_pos[i])
            parts.append(self.prey_pos - self.pred_pos[i])
            parts.append(self.prey_vel)
            obs.append(np.concatenate(parts))
        return np.stack(obs)

    # ------------------------------------------------------------------
    def _prey_heuristic_action(self):
        """Evade: accelerate away from the nearest predator, repelled
        from walls."""
        d = self.pred_pos - self.prey_pos
        dist = np.linalg.norm(d, axis=1)
        nearest = d[np.argmin(dist)]
        away = -nearest / (np.linalg.norm(nearest) + 1e-6)
        # soft wall repulsion
        wall = -np.clip(self.prey_pos, -1, 1) * (
            np.abs(self.prey_pos) > 0.9
        )
        u = away + 2.0 * wall
        nu = np.linalg.norm(u)
        return u / nu if nu > 1e-6 else u

    # ------------------------------------------------------------------
    def step(self, actions: np.ndarray):
        """actions: [n, 5] continuous. Returns (obs [n, obs_dim],
        rewards [n], done, info)."""
        self.t += 1
        a = np.asarray(actions, dtype=float).reshape(self.n, self.act_dim)
        u = np.stack([a[:, 1] - a[:, 2], a[:, 3] - a[:, 4]], axis=1)

        # integrate predators
        self.pred_vel = self.pred_vel * (1 - DAMPING) \
            + u * self.pred_accel * DT
        sp = np.linalg.norm(self.pred_vel, axis=1, keepdims=True)
        scale = np.where(sp > self.pred_max_speed,
                         self.pred_max_speed / (sp + 1e-9), 1.0)
        self.pred_vel = self.pred_vel * scale
        self.pred_pos = self.pred_pos + self.pred_vel * DT

        # integrate prey (heuristic)
        pu = self._prey_heuristic_action()
        self.prey_vel = self.prey_vel * (1 - DAMPING) \
            + pu * self.prey_accel * DT
        psp = np.linalg.norm(self.prey_vel)
        if psp > self.prey_max_speed:
            self.prey_vel *= self.prey_max_speed / psp
        self.prey_pos = self.prey_pos + self.prey_vel * DT

        # obstacle pushback (spring force approximation)
        for o in range(self.n_obst):
            for arr_pos, size in ((self.pred_pos, self.pred_size),
                                  (self.prey_pos.reshape(1, 2),
                                   self.prey_size)):
                delta = arr_pos - self.obst_pos[o]
                dist = np.linalg.norm(delta, axis=1, keepdims=True)
                min_d = self.obst_size + size
                pen = np.maximum(0.0, min_d - dist)
                push = delta / (dist + 1e-9) * pen * CONTACT_FORCE * \
                    DT * DT
                arr_pos += push

        # keep everyone in the box
        self.pred_pos = np.clip(self.pred_pos, -1.2, 1.2)
        self.prey_pos = np.clip(self.prey_pos, -1.2, 1.2)

        # rewards: +10 per predator touching the prey, shaped by
        # negative distance (standard shaped simple_tag adversary reward)
        d = np.linalg.norm(self.pred_pos - self.prey_pos, axis=1)
        catch = d < (self.pred_size + self.prey_size)
        rewards = 10.0 * catch.astype(float) - 0.1 * d
        done = self.t >= self.max_steps
        return self._observations(), rewards, done, {"caught": catch}
