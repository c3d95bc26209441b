"""Multi-agent particle environment: MPE ``simple_tag`` semantics.

The reference vendors a modified PettingZoo 1.10 MPE ``simple_tag``
(reference RL/pettingzoo/mpe/scenarios/simple_tag.py:1-151 and
RL/README.md:31-33 for the modification rationale); PettingZoo is
unavailable offline, so this is a native implementation of the SAME
game with MPE-faithful semantics, property-tested piece by piece
(tests/test_rl.py) against the reference's formulas:

* world: dt 0.1, damping 0.25, soft-margin contact forces
  ``penetration = logaddexp(0, -(dist - dist_min)/k) * k`` with
  k = 1e-3 and contact_force = 1e2 between every colliding entity
  pair, incl. predator-predator (reference _mpe_utils/core.py);
* entities: N adversaries/predators (size .075, accel 3.0, max speed
  1.0), one good agent/prey (.05, 4.0, 1.3), obstacles (size .2) at
  the reference's FIXED modified positions
  (simple_tag.py:51-53 pos_list);
* actions: MPE's 5-dim force parameterization
  (u_x = a[1]-a[2], u_y = a[3]-a[4]) scaled by accel;
* observation (adversary i) mirrors the MODIFIED scenario
  (simple_tag.py:135-151 — entity_pos is computed but NOT
  concatenated): [self_vel(2), self_pos(2), other_pred_rel
  (2*(N-1), world order), prey_rel(2), prey_vel(2)];
* rewards (adversary): the MPE team form — every adversary receives
  sum_adv -0.1 * min_prey dist(prey, adv) plus +10 per colliding
  (prey, adversary) pair (simple_tag.py:117-132: the loops run over
  ALL adversaries, so the reward is identical across predators);
* the prey runs the reference's heuristic evader
  (RL/dist_rl/dist_ppo.py:79-126): flee the nearest adversary with
  the force normalized by its max |component|, with per-axis cutoffs
  at +-1.2.
"""

from __future__ import annotations

import numpy as np

DT = 0.1
DAMPING = 0.25
CONTACT_MARGIN = 1e-3
CONTACT_FORCE = 1e2

# the reference's modified fixed obstacle layout (simple_tag.py:51-53)
OBSTACLE_POS = np.array(
    [[-1.2, -0.6], [0.1, -1.1], [-0.3, 0.4], [0.9, 0.75],
     [-0.9, 1.2], [-0.1, 1.3], [-1.2, 0.0], [1.3, 0.0]]
)


def mpe_collision_force(delta, dist, dist_min):
    """MPE soft-margin contact force magnitude vector for entity a at
    +delta from entity b (reference _mpe_utils/core.py
    get_collision_force)."""
    k = CONTACT_MARGIN
    penetration = np.logaddexp(0.0, -(dist - dist_min) / k) * k
    return CONTACT_FORCE * delta / np.maximum(dist, 1e-12) * penetration


class SimpleTagEnv:
    """Predator-prey tag with MPE dynamics. Predators are externally
    controlled (one per graph node); the prey runs the reference's
    heuristic evader."""

    def __init__(self, num_predators=3, num_obstacles=2, seed=0,
                 max_steps=100):
        self.n = num_predators
        self.n_obst = num_obstacles
        if num_obstacles > len(OBSTACLE_POS):
            raise ValueError("at most 8 obstacles (reference pos_list)")
        self.rng = np.random.default_rng(seed)
        self.max_steps = max_steps
        self.pred_size = 0.075
        self.prey_size = 0.05
        self.obst_size = 0.2
        self.pred_accel = 3.0
        self.prey_accel = 4.0
        self.pred_max_speed = 1.0
        self.prey_max_speed = 1.3
        # modified-scenario adversary observation: no obstacle entries
        self.obs_dim = 2 + 2 + 2 * (self.n - 1) + 2 + 2
        self.act_dim = 5
        self.reset()

    # ------------------------------------------------------------------
    def reset(self):
        self.t = 0
        self.pred_pos = self.rng.uniform(-1, 1, size=(self.n, 2))
        self.pred_vel = np.zeros((self.n, 2))
        self.prey_pos = self.rng.uniform(-1, 1, size=2)
        self.prey_vel = np.zeros(2)
        self.obst_pos = OBSTACLE_POS[: self.n_obst].copy()
        return self._observations()

    # ------------------------------------------------------------------
    def _observations(self):
        obs = []
        for i in range(self.n):
            parts = [self.pred_vel[i], self.pred_pos[i]]
            for j in range(self.n):  # world order, self skipped
                if j != i:
                    parts.append(self.pred_pos[j] - self.pred_pos[i])
            parts.append(self.prey_pos - self.pred_pos[i])
            parts.append(self.prey_vel)
            obs.append(np.concatenate(parts))
        return np.stack(obs)

    def _prey_observation(self):
        """Good-agent observation ([vel, pos, adv_rel...]) — what the
        reference's heuristic evader consumes."""
        parts = [self.prey_vel, self.prey_pos]
        for j in range(self.n):
            parts.append(self.pred_pos[j] - self.prey_pos)
        return np.concatenate(parts)

    # ------------------------------------------------------------------
    def _prey_heuristic_action(self):
        """Reference evader (RL/dist_rl/dist_ppo.py:79-126): move
        opposite the closest adversary, force normalized by its max
        |component|; zero the outward channel at the +-1.2 boundary."""
        obs = self._prey_observation()
        dists = obs[4:].reshape(-1, 2)
        near = dists[np.argmin(np.linalg.norm(dists, axis=1))]
        force = -near / max(np.max(np.abs(near)), 1e-12)
        action = np.zeros(5)
        if force[0] > 0:
            action[1] = force[0]
        else:
            action[2] = -force[0]
        if force[1] > 0:
            action[3] = force[1]
        else:
            action[4] = -force[1]
        if obs[2] <= -1.2:
            action[2] = 0.0
        elif obs[2] >= 1.2:
            action[1] = 0.0
        if obs[3] <= -1.2:
            action[4] = 0.0
        elif obs[3] >= 1.2:
            action[3] = 0.0
        return action

    # ------------------------------------------------------------------
    def _forces(self, u_pred, u_prey):
        """Action + pairwise contact forces for all movable entities
        (predators 0..n-1, prey = index n)."""
        pos = np.vstack([self.pred_pos, self.prey_pos[None]])
        sizes = np.array([self.pred_size] * self.n + [self.prey_size])
        f = np.vstack([u_pred * self.pred_accel,
                       (u_prey * self.prey_accel)[None]])
        # movable-movable pairs (action force +f on a, -f on b)
        for a in range(self.n + 1):
            for b in range(a + 1, self.n + 1):
                delta = pos[a] - pos[b]
                dist = np.linalg.norm(delta)
                fc = mpe_collision_force(delta, dist,
                                         sizes[a] + sizes[b])
                f[a] += fc
                f[b] -= fc
        # vs immovable obstacles
        for a in range(self.n + 1):
            for o in range(self.n_obst):
                delta = pos[a] - self.obst_pos[o]
                dist = np.linalg.norm(delta)
                f[a] += mpe_collision_force(
                    delta, dist, sizes[a] + self.obst_size
                )
        return f

    # ------------------------------------------------------------------
    def _rewards(self):
        """MPE adversary team reward (simple_tag.py:117-132): identical
        for every predator."""
        d = np.linalg.norm(self.pred_pos - self.prey_pos, axis=1)
        rew = -0.1 * d.sum()  # sum over advs of min over (1) prey
        ncoll = int(
            (d < (self.pred_size + self.prey_size)).sum()
        )  # colliding (prey, adv) pairs
        rew += 10.0 * ncoll
        return np.full(self.n, rew), d

    def prey_reward(self):
        """MPE good-agent reward (simple_tag.py:91-114): -10 per
        touching adversary, boundary penalty."""
        d = np.linalg.norm(self.pred_pos - self.prey_pos, axis=1)
        rew = -10.0 * float(
            (d < (self.pred_size + self.prey_size)).sum()
        )

        def bound(x):
            if x < 0.9:
                return 0.0
            if x < 1.0:
                return (x - 0.9) * 10
            return min(np.exp(2 * x - 2), 10)

        for p in range(2):
            rew -= bound(abs(self.prey_pos[p]))
        return rew

    # ------------------------------------------------------------------
    def step(self, actions: np.ndarray):
        """actions: [n, 5] continuous. Returns (obs [n, obs_dim],
        rewards [n], done, info)."""
        self.t += 1
        a = np.asarray(actions, dtype=float).reshape(
            self.n, self.act_dim
        )
        u_pred = np.stack([a[:, 1] - a[:, 2], a[:, 3] - a[:, 4]],
                          axis=1)
        pa = self._prey_heuristic_action()
        u_prey = np.array([pa[1] - pa[2], pa[3] - pa[4]])

        f = self._forces(u_pred, u_prey)

        # MPE integrator: vel = vel*(1-damping) + (f/m)*dt, clamp
        # speed, then pos += vel*dt (core.py integrate_state)
        vel = np.vstack([self.pred_vel, self.prey_vel[None]])
        pos = np.vstack([self.pred_pos, self.prey_pos[None]])
        vel = vel * (1 - DAMPING) + f * DT
        vmax = np.array(
            [self.pred_max_speed] * self.n + [self.prey_max_speed]
        )
        sp = np.linalg.norm(vel, axis=1)
        over = sp > vmax
        vel[over] = vel[over] / sp[over, None] * vmax[over, None]
        pos = pos + vel * DT

        self.pred_vel, self.prey_vel = vel[: self.n], vel[self.n]
        self.pred_pos, self.prey_pos = pos[: self.n], pos[self.n]

        rewards, d = self._rewards()
        catch = d < (self.pred_size + self.prey_size)
        done = self.t >= self.max_steps
        return self._observations(), rewards, done, {"caught": catch}
