"""DSGT — decentralized SGD with gradient tracking.

Algorithm parity with the reference's ``optimizers/dsgt.py:6-115``: DSGD
plus a tracked gradient estimate y per node. Per round:

    p_i^{k+1} = sum_{j in N(i) ∪ i} W_ij (p_j^k - alpha * y_j^k)
    g_i^{k+1} = grad f_i(p_i^{k+1})
    y_i^{k+1} = sum_{j} W_ij y_j^k + g_i^{k+1} - g_i^k

which doubles the communication volume (params AND y per neighbor edge) —
one batched P2P exchange per round carries the concatenated [2n] bucket.

Deviation (same as DSGD, documented): the reference's mixing sweeps
sequentially in place over nodes; here both the p-mix and the y-mix use
the round-start snapshots (synchronous message passing).
"""

from __future__ import annotations

import torch

from ..utils import graph_generation
from .neighbors import gather_neighbor_stacks


class DSGT:
    def __init__(self, ddl_problem, device, conf):
        self.pr = ddl_problem
        self.conf = conf
        self.device = torch.device(device)
        self.alpha = conf["alpha"]

        n = self.pr.n
        self.y = {
            i: torch.zeros(n, device=self.device)
            for i in self.pr.local_nodes
        }
        self.g = {
            i: torch.zeros(n, device=self.device)
            for i in self.pr.local_nodes
        }
        self.checkpoint_dir = conf.get("checkpoint_dir")

    # ------------------------------------------------------------------
    def _local_grad_vector(self, i) -> torch.Tensor:
        """fwd/bwd on node i's next batch; returns the flat gradient."""
        pr = self.pr
        bloss = pr.local_batch_loss(i)
        bloss.backward()
        grads = []
        with torch.no_grad():
            for p in pr.models[i].parameters():
                grads.append(p.grad.reshape(-1).clone())
                p.grad.zero_()
        return torch.cat(grads)

    # ------------------------------------------------------------------
    def train(self, profiler=None):
        if self.pr.stacked is not None:
            return self._train_stacked(profiler)
        pr = self.pr
        eval_every = pr.conf["metrics_config"]["evaluate_frequency"]
        oits = self.conf["outer_iterations"]

        from .checkpointing import load_checkpoint, save_checkpoint

        ck_every = self.conf.get("checkpoint_every", 0)
        k0 = 0
        if self.conf.get("resume_from"):
            k0, st = load_checkpoint(self.conf["resume_from"], pr)
            self.y = {i: v.to(self.device) for i, v in st["y"].items()}
            self.g = {i: v.to(self.device) for i, v in st["g"].items()}
        elif self.conf["init_grads"]:
            for i in pr.local_nodes:
                g = self._local_grad_vector(i)
                self.y[i] = g.clone()
                self.g[i] = g.clone()

        for k in range(k0, oits):
            if ck_every and k > k0 and k % ck_every == 0:
                save_checkpoint(
                    self.checkpoint_dir, pr, k - 1,
                    {"y": self.y, "g": self.g},
                )
            if k % eval_every == 0 or k == oits - 1:
                pr.evaluate_metrics(at_end=(k == oits - 1))

            pr.update_graph()
            W = graph_generation.get_metropolis(pr.graph).to(self.device)

            # snapshot [L, 2n]: params and tracker ride one exchange
            ths = pr.local_params_stack()
            ys = torch.stack([self.y[i] for i in pr.local_nodes]) \
                if pr.local_nodes else ths.new_zeros(0, pr.n)
            bundle = torch.cat([ths, ys], dim=1)
            neigh = gather_neighbor_stacks(pr, bundle)

            n = pr.n
            y_new = {}
            for li, i in enumerate(pr.local_nodes):
                p_mix = W[i, i] * (ths[li] - self.alpha * self.y[i])
                y_mix = W[i, i] * self.y[i]
                for row, j in zip(neigh[i], pr.graph.neighbors(i)):
                    pj, yj = row[:n], row[n:]
                    p_mix = p_mix + W[i, j] * (pj - self.alpha * yj)
                    y_mix = y_mix + W[i, j] * yj
                torch.nn.utils.vector_to_parameters(
                    p_mix, pr.models[i].parameters()
                )
                y_new[i] = y_mix

            for i in pr.local_nodes:
                g_next = self._local_grad_vector(i)
                self.y[i] = y_new[i] + g_next - self.g[i]
                self.g[i] = g_next

            if profiler is not None:
                profiler.step()

    def _train_stacked(self, profiler=None):
        from ..ops.stacked import DSGTStackedDriver

        DSGTStackedDriver(self, self.pr).run(profiler)
