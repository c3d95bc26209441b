"""Decentralized PPO optimizers (DiNNO / DSGD / DSGT over actor+critic).

Algorithm parity with the reference's RL tree
(``RL/dist_rl/{dinno,dsgd,dsgt}PPO.py``): per communication round,
collect an on-policy rollout with every node's current actor, exchange
the concatenated [actor | critic] parameter vectors with graph
neighbors, and apply the respective decentralized update. Differences by
design:

* the actor and critic blocks live in ONE flat vector per node; dual
  ascent / mixing decompose per coordinate, so this is mathematically
  identical to the reference's twin-block bookkeeping while halving the
  bookkeeping code. DSGT's separate actor/critic step sizes
  (dsgtPPO.py:47-48) are kept via a per-coordinate alpha vector;
* the reference DSGDPPO's aliasing bugs (critic mix lists built from
  actors and written into the actor lists, dsgdPPO.py:21-23,:66-73) are
  NOT reproduced;
* mixing is snapshot-synchronous (same deviation as optimizers/dsgd.py,
  documented there).

Checkpoint layout parity (reference dinnoPPO.py:231-265): every
``save_freq`` iterations write ``trained/ppo_actors_tag_<alg>_<id>.pth``
/ ``ppo_critics_...``, plus ``avg_ep_rews_<alg>_<id>.npy``,
``timesteps_<alg>_<id>.npy`` and ``agreements_<alg>_<id>.npz``.
"""

from __future__ import annotations

import math
import os

import numpy as np
import torch


class _PPOBase:
    alg = "base"

    def __init__(self, problem, conf: dict):
        self.pr = problem
        self.conf = conf
        self.device = problem.device
        self.run_id = conf.get("run_id", "0")
        self.save_freq = conf.get("save_freq", 0)
        self.out_dir = conf.get("output_dir", "./trained")
        self.avg_ep_rews = []
        self.timesteps = []
        self.agreements_actor = []
        self.agreements_critic = []

    # ------------------------------------------------------------------
    def _snapshot_vectors(self):
        return {
            i: self.pr.node_vector(i).detach().clone()
            for i in range(self.pr.N)
        }

    def _neighbors(self, i):
        return list(self.pr.graph.neighbors(i))

    # ------------------------------------------------------------------
    def _record(self, it):
        self.avg_ep_rews.append(self.pr.ep_rew_history[-1])
        self.timesteps.append(self.pr.total_timesteps)
        aa, cc = self.pr.agreement()
        self.agreements_actor.append(aa.numpy())
        self.agreements_critic.append(cc.numpy())
        if self.conf.get("verbose", True):
            print(
                f"[{self.alg}] iter {it} t={self.pr.total_timesteps} "
                f"avg_ep_rew={self.avg_ep_rews[-1]:.2f} "
                f"agree_max={aa.max().item():.4f}",
                flush=True,
            )
        if self.save_freq and (it + 1) % self.save_freq == 0:
            self.save(it)

    def save(self, it=None):
        os.makedirs(self.out_dir, exist_ok=True)
        tag = f"tag_{self.alg}_{self.run_id}"
        if it is not None:
            tag_k = f"{tag}_{it + 1}"
        else:
            tag_k = tag
        torch.save(
            {i: self.pr.actors[i].state_dict()
             for i in range(self.pr.N)},
            os.path.join(self.out_dir, f"ppo_actors_{tag_k}.pth"),
        )
        torch.save(
            {i: self.pr.critics[i].state_dict()
             for i in range(self.pr.N)},
            os.path.join(self.out_dir, f"ppo_critics_{tag_k}.pth"),
        )
        np.save(
            os.path.join(self.out_dir, f"avg_ep_rews_{tag}.npy"),
            np.asarray(self.avg_ep_rews),
        )
        np.save(
            os.path.join(self.out_dir, f"timesteps_{tag}.npy"),
            np.asarray(self.timesteps),
        )
        np.savez(
            os.path.join(self.out_dir, f"agreements_{tag}.npz"),
            actor=np.stack(self.agreements_actor),
            critic=np.stack(self.agreements_critic),
        )

    # ------------------------------------------------------------------
    def train(self):
        max_t = self.conf.get("max_rl_timesteps", 10_000)
        it = 0
        while self.pr.total_timesteps < max_t:
            self.pr.rollout()
            self.step_round(it)
            self._record(it)
            it += 1
        self.save()


class DiNNOPPO(_PPOBase):
    """Consensus-ADMM PPO (reference dinnoPPO.py:6-269)."""

    alg = "cadmm"

    def __init__(self, problem, conf):
        super().__init__(problem, conf)
        self.rho = conf.get("rho_init", 0.3)
        self.rho_scaling = conf.get("rho_scaling", 1.0)
        self.pits = conf.get("primal_iterations", 5)
        oits = conf.get("expected_iterations", 200)
        lr0 = conf.get("primal_lr_start", 1e-3)
        lr1 = conf.get("primal_lr_finish", 1e-4)
        decay = conf.get("lr_decay_type", "log")
        if decay == "constant":
            self.lr = lr0 * torch.ones(oits)
        elif decay == "linear":
            self.lr = torch.linspace(lr0, lr1, oits)
        else:
            self.lr = torch.logspace(
                math.log10(lr0), math.log10(lr1), oits
            )
        self.duals = {
            i: torch.zeros(problem.n, device=self.device)
            for i in range(problem.N)
        }

    # ------------------------------------------------------------------
    def _hip_available(self):
        if os.environ.get("NDTA_RL_HIP", "1") == "0":
            return False
        if self.device.type != "cuda":
            return False
        from ..ops import ext_available

        return ext_available()

    def _csr(self):
        """CSR neighbor tables over all nodes (single-process RL)."""
        if getattr(self, "_csr_cache", None) is None:
            offs, idx = [0], []
            for i in range(self.pr.N):
                ns = self._neighbors(i)
                idx.extend(ns)
                offs.append(len(idx))
            dev = self.device
            self._csr_cache = (
                torch.tensor(offs, dtype=torch.int32, device=dev),
                torch.tensor(idx, dtype=torch.int32, device=dev),
                torch.tensor(
                    [len(self._neighbors(i)) for i in range(self.pr.N)],
                    dtype=torch.int32, device=dev,
                ),
            )
        return self._csr_cache

    def _step_round_hip(self, it, lr):
        """DiNNO round math on the stacked CDNA4 kernels (VERDICT r1
        item 9; reference round math RL/dist_rl/dinnoPPO.py:87-131):
        the [actor|critic] vectors of all nodes form one [N, n] stack;
        dual ascent + s-reduction run in ONE dinno_dual_threg launch
        and each primal step is ONE fused penalty-Adam launch (the
        analytic consensus-penalty gradient folds into the update).
        Rollouts and pred-loss autograd stay per-node torch (the env
        stepping is host-side by construction)."""
        from ..ops import get_ext

        ext = get_ext()
        pr = self.pr
        dt = torch.get_default_dtype()
        ths = torch.stack(
            [pr.node_vector(i).detach() for i in range(pr.N)]
        ).contiguous()
        if getattr(self, "_duals_t", None) is None:
            self._duals_t = torch.zeros_like(ths)
            self._s_t = torch.empty_like(ths)
            self._m_t = torch.empty_like(ths)
            self._v_t = torch.empty_like(ths)
        offs, idx, deg = self._csr()
        ext.dinno_dual_threg(
            ths, None, offs, idx, self._duals_t, self._s_t, self.rho
        )
        # dict view kept in sync for checkpoints/inspection
        for i in range(pr.N):
            self.duals[i] = self._duals_t[i]
        theta = ths.clone()
        for pit in range(self.pits):
            grads = []
            for i in range(pr.N):
                for p in pr.node_parameters(i):
                    p.grad = None
                pred_loss = pr.local_batch_loss(i)
                pred_loss.backward()
                grads.append(
                    torch.cat(
                        [
                            (p.grad if p.grad is not None
                             else torch.zeros_like(p)).reshape(-1)
                            for p in pr.node_parameters(i)
                        ]
                    )
                )
            grad = torch.stack(grads).to(dt).contiguous()
            # fresh Adam per round (reference recreates the optimizer):
            # first_step resets the moments, step_t restarts at 1
            ext.fused_step(
                theta, grad, self._duals_t, self._s_t, deg,
                self._m_t, self._v_t,
                self.rho, lr, 0.9, 0.999, 1e-8, 0.0,
                pit + 1, 0, pit == 0, 1, False,
            )
            for i in range(pr.N):
                pr.set_node_vector(i, theta[i])

    def step_round(self, it):
        pr = self.pr
        self.rho *= self.rho_scaling
        lr = float(self.lr[min(it, len(self.lr) - 1)])
        if self._hip_available():
            return self._step_round_hip(it, lr)
        ths = self._snapshot_vectors()
        for i in range(pr.N):
            neighs = self._neighbors(i)
            thj = torch.stack([ths[j] for j in neighs])
            self.duals[i] += self.rho * torch.sum(ths[i] - thj, dim=0)
            th_reg = 0.5 * (thj + ths[i])
            opt = torch.optim.Adam(pr.node_parameters(i), lr)
            for _ in range(self.pits):
                opt.zero_grad()
                pred_loss = pr.local_batch_loss(i)
                th = pr.node_vector(i)
                reg = torch.sum(
                    torch.square(th.unsqueeze(0) - th_reg)
                )
                loss = (
                    pred_loss
                    + torch.dot(th, self.duals[i])
                    + self.rho * reg
                )
                loss.backward()
                opt.step()


class DSGDPPO(_PPOBase):
    """Decentralized SGD PPO (reference dsgdPPO.py:7-165, minus its
    actor/critic aliasing bugs)."""

    alg = "dsgd"

    def __init__(self, problem, conf):
        super().__init__(problem, conf)
        self.alph = conf.get("alpha0", 1e-3)
        self.mu = conf.get("mu", 1e-3)

    def step_round(self, it):
        from ..utils import graph_generation

        pr = self.pr
        W = graph_generation.get_metropolis(pr.graph).to(self.device)
        self.alph = self.alph * (1 - self.mu * self.alph)
        ths = self._snapshot_vectors()
        for i in range(pr.N):
            mixed = W[i, i] * ths[i]
            for j in self._neighbors(i):
                mixed = mixed + W[i, j] * ths[j]
            pr.set_node_vector(i, mixed)
        for i in range(pr.N):
            loss = pr.local_batch_loss(i)
            loss.backward()
            with torch.no_grad():
                for p in pr.node_parameters(i):
                    p.add_(p.grad, alpha=-self.alph)
                    p.grad.zero_()


class DSGTPPO(_PPOBase):
    """Gradient-tracking PPO (reference dsgtPPO.py:7-254) with separate
    actor/critic step sizes kept as a per-coordinate alpha vector."""

    alg = "dsgt"

    def __init__(self, problem, conf):
        super().__init__(problem, conf)
        a_act = conf.get("alpha_actor", 1e-3)
        a_cri = conf.get("alpha_critic", conf.get("alpha_actor", 1e-3))
        self.alpha_vec = torch.cat(
            [
                torch.full((problem.n_actor,), a_act),
                torch.full((problem.n_critic,), a_cri),
            ]
        ).to(self.device)
        self.y = {
            i: torch.zeros(problem.n, device=self.device)
            for i in range(problem.N)
        }
        self.g = {
            i: torch.zeros(problem.n, device=self.device)
            for i in range(problem.N)
        }
        self._bootstrapped = False

    def _grad_vector(self, i):
        pr = self.pr
        loss = pr.local_batch_loss(i)
        loss.backward()
        gs = []
        with torch.no_grad():
            for p in pr.node_parameters(i):
                gs.append(p.grad.reshape(-1).clone())
                p.grad.zero_()
        return torch.cat(gs)

    def step_round(self, it):
        from ..utils import graph_generation

        pr = self.pr
        if not self._bootstrapped:
            # reference dsgtPPO.py:58-85 bootstraps y, g from the
            # first rollout's gradients
            for i in range(pr.N):
                g = self._grad_vector(i)
                self.y[i] = g.clone()
                self.g[i] = g.clone()
            self._bootstrapped = True

        W = graph_generation.get_metropolis(pr.graph).to(self.device)
        ths = self._snapshot_vectors()
        ys = {i: self.y[i].clone() for i in range(pr.N)}
        y_new = {}
        for i in range(pr.N):
            p_mix = W[i, i] * (ths[i] - self.alpha_vec * ys[i])
            y_mix = W[i, i] * ys[i]
            for j in self._neighbors(i):
                p_mix = p_mix + W[i, j] * (
                    ths[j] - self.alpha_vec * ys[j]
                )
                y_mix = y_mix + W[i, j] * ys[j]
            pr.set_node_vector(i, p_mix)
            y_new[i] = y_mix
        for i in range(pr.N):
            g_next = self._grad_vector(i)
            self.y[i] = y_new[i] + g_next - self.g[i]
            self.g[i] = g_next


def build_ppo_optimizer(alg, problem, conf):
    cls = {"dinno": DiNNOPPO, "cadmm": DiNNOPPO, "dsgd": DSGDPPO,
           "dsgt": DSGTPPO}[alg]
    return cls(problem, conf)
