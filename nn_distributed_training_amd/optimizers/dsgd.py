"""DSGD — decentralized SGD with Metropolis mixing.

Algorithm parity with the reference's ``optimizers/dsgd.py:6-62``: per
round, rebuild the Metropolis weights W from the (possibly dynamic) graph,
decay the step size ``alpha <- alpha * (1 - mu * alpha)``, mix parameters
with neighbors, then take one local SGD step on a batch gradient.

Deviation (documented; SURVEY.md §7 "hard parts"): the reference mixes
sequentially IN-PLACE over nodes (dsgd.py:37-46), so node i>0 reads
already-mixed neighbor tensors (a Gauss–Seidel sweep whose result depends
on node order). A message-passing implementation is necessarily
synchronous: mixing here uses the round-start snapshot for every node
(``p_i <- W_ii p_i + sum_j W_ij p_j^(k)``), which is the textbook DSGD
update and is node-order independent. Convergence curves differ slightly
from the reference's sweep; tests validate against a synchronous golden
model.
"""

from __future__ import annotations

import torch

from ..utils import graph_generation
from .neighbors import gather_neighbor_stacks


class DSGD:
    def __init__(self, ddl_problem, device, conf):
        self.pr = ddl_problem
        self.conf = conf
        self.device = torch.device(device)
        self.alph0 = conf["alpha0"]
        self.mu = conf["mu"]
        self.checkpoint_dir = conf.get("checkpoint_dir")

    def train(self, profiler=None):
        if self.pr.stacked is not None:
            return self._train_stacked(profiler)
        from .checkpointing import load_checkpoint, save_checkpoint

        pr = self.pr
        eval_every = pr.conf["metrics_config"]["evaluate_frequency"]
        oits = self.conf["outer_iterations"]
        ck_every = self.conf.get("checkpoint_every", 0)

        alph = self.alph0
        k0 = 0
        if self.conf.get("resume_from"):
            k0, st = load_checkpoint(self.conf["resume_from"], pr)
            alph = st["alph"]
        for k in range(k0, oits):
            if ck_every and k > k0 and k % ck_every == 0:
                save_checkpoint(
                    self.checkpoint_dir, pr, k - 1, {"alph": alph}
                )
            if k % eval_every == 0 or k == oits - 1:
                pr.evaluate_metrics(at_end=(k == oits - 1))

            pr.update_graph()
            W = graph_generation.get_metropolis(pr.graph).to(self.device)
            alph = alph * (1 - self.mu * alph)

            # synchronous mixing on the round-start snapshot
            ths = pr.local_params_stack()
            neigh = gather_neighbor_stacks(pr, ths)
            for li, i in enumerate(pr.local_nodes):
                mixed = W[i, i] * ths[li]
                for row, j in zip(neigh[i], pr.graph.neighbors(i)):
                    mixed = mixed + W[i, j] * row
                torch.nn.utils.vector_to_parameters(
                    mixed, pr.models[i].parameters()
                )

            # local gradient step
            for i in pr.local_nodes:
                bloss = pr.local_batch_loss(i)
                bloss.backward()
                with torch.no_grad():
                    for p in pr.models[i].parameters():
                        p.add_(p.grad, alpha=-alph)
                        p.grad.zero_()

            if profiler is not None:
                profiler.step()

    def _train_stacked(self, profiler=None):
        from ..ops.stacked import DSGDStackedDriver

        DSGDStackedDriver(self, self.pr).run(profiler)
