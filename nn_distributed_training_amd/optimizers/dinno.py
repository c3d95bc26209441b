"""DiNNO — Distributed Neural Network Optimization (consensus ADMM).

Algorithm parity with the reference's ``optimizers/dinno.py:5-130``
(arXiv:2109.08665): per communication round each node snapshots its flat
parameter vector, exchanges it with graph neighbors, performs dual ascent
``dual_i += rho * sum_j (th_i - th_j)``, forms regularization targets
``th_reg = (th_j + th_i) / 2`` and then runs ``primal_iterations`` steps of
a local optimizer on

    pred_loss + <th, dual_i> + rho * sum_j ||th - th_reg_j||^2 .

Differences from the reference (by design):
* nodes live on ranks; the neighbor snapshot gather
  (reference dinno.py:103-110,:120-122) is a batched RCCL P2P exchange of
  flat parameter buckets over xGMI (parallel/comm.py), not an in-memory
  read — DiNNO is snapshot-synchronous in the reference too, so the
  semantics are identical;
* the penalty is evaluated directly as ``((th - th_reg)**2).sum()``
  (the reference's ``cdist(th.reshape(1,-1), th_reg)`` squared-sum is the
  same value at O(deg · n) without the cdist machinery);
* when the problem carries a stacked HIP engine, the whole per-node loop
  (dual ascent, penalty-fused Adam, fwd/bwd) runs as batched CDNA4
  kernels over all local nodes (ops/).
"""

from __future__ import annotations

import math

import torch

from .neighbors import gather_neighbor_stacks


class DiNNO:
    def __init__(self, ddl_problem, device, conf):
        self.pr = ddl_problem
        self.conf = conf
        self.device = torch.device(device)

        self.duals = {
            i: torch.zeros(self.pr.n, device=self.device)
            for i in self.pr.local_nodes
        }
        self.rho = conf["rho_init"]
        self.rho_scaling = conf["rho_scaling"]
        self.primal_lr = self._lr_schedule(conf)
        self.pits = conf["primal_iterations"]

        self.opts = {}
        if conf["persistant_primal_opt"]:
            for i in self.pr.local_nodes:
                self.opts[i] = self._make_opt(i, float(self.primal_lr[0]))
        self.checkpoint_dir = conf.get("checkpoint_dir")

    # ------------------------------------------------------------------
    @staticmethod
    def _lr_schedule(conf) -> torch.Tensor:
        oits = conf["outer_iterations"]
        kind = conf["lr_decay_type"]
        if kind == "constant":
            return conf["primal_lr_start"] * torch.ones(oits)
        if kind == "linear":
            return torch.linspace(
                conf["primal_lr_start"], conf["primal_lr_finish"], oits
            )
        if kind == "log":
            return torch.logspace(
                math.log10(conf["primal_lr_start"]),
                math.log10(conf["primal_lr_finish"]),
                oits,
            )
        raise NameError("Unknown primal learning rate decay type.")

    def _make_opt(self, i, lr):
        params = self.pr.models[i].parameters()
        name = self.conf["primal_optimizer"]
        if name == "adam":
            return torch.optim.Adam(params, lr)
        if name == "sgd":
            return torch.optim.SGD(params, lr)
        if name == "adamw":
            return torch.optim.AdamW(params, lr)
        raise NameError("DiNNO primal optimizer is unknown.")

    # ------------------------------------------------------------------
    def primal_update(self, i, th_reg, k):
        if self.conf["persistant_primal_opt"]:
            opt = self.opts[i]
        else:
            opt = self._make_opt(i, float(self.primal_lr[k]))

        for _ in range(self.pits):
            opt.zero_grad()
            pred_loss = self.pr.local_batch_loss(i)
            th = torch.nn.utils.parameters_to_vector(
                self.pr.models[i].parameters()
            )
            reg = torch.sum(torch.square(th.unsqueeze(0) - th_reg))
            loss = pred_loss + torch.dot(th, self.duals[i]) + self.rho * reg
            loss.backward()
            opt.step()

    # ------------------------------------------------------------------
    # -- checkpoint/resume (optimizers/checkpointing.py) -----------------
    def _opt_state(self):
        st = {"duals": self.duals, "rho": self.rho}
        if self.conf["persistant_primal_opt"]:
            st["opts"] = {i: o.state_dict() for i, o in self.opts.items()}
        return st

    def _load_opt_state(self, st):
        self.duals = {
            i: v.to(self.device) for i, v in st["duals"].items()
        }
        self.rho = st["rho"]
        for i, sd in st.get("opts", {}).items():
            self.opts[i].load_state_dict(sd)

    def train(self, profiler=None):
        if self.pr.stacked is not None:
            return self._train_stacked(profiler)
        from .checkpointing import load_checkpoint, save_checkpoint

        pr = self.pr
        eval_every = pr.conf["metrics_config"]["evaluate_frequency"]
        oits = self.conf["outer_iterations"]
        ck_every = self.conf.get("checkpoint_every", 0)
        k0 = 0
        if self.conf.get("resume_from"):
            k0, st = load_checkpoint(self.conf["resume_from"], pr)
            self._load_opt_state(st)
        for k in range(k0, oits):
            if ck_every and k > k0 and k % ck_every == 0:
                save_checkpoint(
                    self.checkpoint_dir, pr, k - 1, self._opt_state()
                )
            if k % eval_every == 0 or k == oits - 1:
                pr.evaluate_metrics(at_end=(k == oits - 1))

            ths = pr.local_params_stack().clone()
            self.rho *= self.rho_scaling
            pr.update_graph()

            neigh = gather_neighbor_stacks(pr, ths)
            for li, i in enumerate(pr.local_nodes):
                thj = neigh[i]
                if thj.shape[0] == 0:
                    continue
                self.duals[i] += self.rho * torch.sum(
                    ths[li] - thj, dim=0
                )
                th_reg = 0.5 * (thj + ths[li])
                self.primal_update(i, th_reg, k)

            if profiler is not None:
                profiler.step()

    # ------------------------------------------------------------------
    def _train_stacked(self, profiler=None):
        """Fused HIP path: batched over this rank's nodes (ops/stacked.py)."""
        from ..ops.stacked import DiNNOStackedDriver

        driver = DiNNOStackedDriver(self, self.pr)
        driver.run(profiler)
