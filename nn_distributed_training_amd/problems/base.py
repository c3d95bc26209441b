"""Shared problem-layer machinery.

A *problem* owns the communication graph, the node replicas THIS RANK
hosts, per-node data iterators and the metric registry. The L3 (optimizer)
<-> L2 (problem) contract follows the reference:
``pr.N``, ``pr.n``, ``pr.graph``, ``pr.models[i]``, ``pr.local_batch_loss(i)``,
``pr.update_graph()``, ``pr.evaluate_metrics(at_end=...)``,
``pr.save_metrics(dir)`` — with two distributed-era extensions:

* ``pr.models`` holds only the nodes packed onto this rank
  (``pr.local_nodes``); optimizers iterate local nodes and fetch remote
  neighbor vectors through ``pr.comm`` (parallel/comm.py);
* metric evaluation is collective: per-node values are computed where the
  node lives and all-gathered; printing/saving happens on rank 0.

In single-process mode (world_size 1) this degenerates exactly to the
reference's behavior: all N nodes local, no communication.
"""

from __future__ import annotations

import copy
import os

import torch

from ..parallel.comm import Communicator, NodeLayout


class ProblemBase:
    def __init__(self, graph, base_model, base_loss, train_sets, val_set,
                 device, conf):
        self.graph = graph
        self.base_loss = base_loss
        self.train_sets = train_sets
        self.val_set = val_set
        self.conf = conf
        self.device = torch.device(device)

        self.N = graph.number_of_nodes()
        self.n = torch.nn.utils.parameters_to_vector(
            base_model.parameters()
        ).shape[0]

        self.comm = Communicator(self.device)
        self.layout = NodeLayout(self.N, self.comm.world)
        self.local_nodes = list(self.layout.nodes_of(self.comm.rank))
        self.is_root = self.comm.rank == 0

        # replicate the base model for the nodes this rank hosts
        self.models = {
            i: copy.deepcopy(base_model).to(self.device)
            for i in self.local_nodes
        }

        # Per-node shuffle generators seeded by (data_seed, node id): node
        # i's batch stream is identical no matter which rank hosts it, so
        # a multi-rank run reproduces the single-process run exactly
        # (tested in tests/test_distributed.py).
        data_seed = int(self.conf.get("data_seed", 0))
        self.train_loaders = {}
        self.train_iters = {}
        for i in self.local_nodes:
            g = torch.Generator()
            g.manual_seed(data_seed * 100003 + i)
            self.train_loaders[i] = torch.utils.data.DataLoader(
                self.train_sets[i],
                batch_size=self.conf["train_batch_size"],
                shuffle=True,
                generator=g,
            )
            self.train_iters[i] = iter(self.train_loaders[i])

        self.val_loader = torch.utils.data.DataLoader(
            self.val_set, batch_size=self.conf["val_batch_size"]
        )

        self.metrics = {m: [] for m in self.conf["metrics"]}
        self.epoch_tracker = torch.zeros(self.N)
        self.forward_cnt = 0

        # populated by optimizers that run the stacked HIP engine
        self.stacked = None

    # ------------------------------------------------------------------
    def next_batch(self, i):
        """Next local batch for node i, with epoch-wrap tracking."""
        try:
            batch = next(self.train_iters[i])
        except StopIteration:
            self.epoch_tracker[i] += 1
            self.train_iters[i] = iter(self.train_loaders[i])
            batch = next(self.train_iters[i])
        if i == 0:
            # node 0 counts forward passes; symmetric across nodes
            # (reference problems/dist_mnist_problem.py:90-94)
            self.forward_cnt += self.conf["train_batch_size"]
        return batch

    # ------------------------------------------------------------------
    def local_params_stack(self) -> torch.Tensor:
        """[L, n] detached snapshot of this rank's node parameters."""
        rows = [
            torch.nn.utils.parameters_to_vector(
                self.models[i].parameters()
            ).detach()
            for i in self.local_nodes
        ]
        return torch.stack(rows) if rows else torch.zeros(
            0, self.n, device=self.device
        )

    def update_graph(self):
        """Static problems keep their graph; dynamic ones override."""
        pass

    # ------------------------------------------------------------------
    def consensus_error(self):
        """Normalized pairwise parameter distances across ALL nodes.

        The one true collective in the framework: all-gather of the
        [L, n] stacks, then cdist on the normalized [N, n] stack (parity
        with reference problems/dist_mnist_problem.py:152-175).
        """
        with torch.no_grad():
            if self.stacked is not None:
                local = self.stacked.theta
            else:
                local = self.local_params_stack()
            th_stack = self.comm.all_gather_stack(self.layout, local)
            th_stack = th_stack.to(torch.float64)
            if self.stacked is not None and th_stack.is_cuda:
                # first-party pairwise-distance kernels (the last
                # eager-torch op in the system; tested vs torch.cdist
                # in tests/test_ops_gpu.py)
                d_all, d_mean = self.stacked.ext.consensus_cdist(
                    th_stack.contiguous()
                )
                return d_all.cpu(), d_mean.cpu()
            th_stack = torch.nn.functional.normalize(th_stack, dim=1)
            distances_all = torch.cdist(th_stack, th_stack)
            th_mean = th_stack.mean(dim=0, keepdim=True)
            distances_mean = torch.cdist(th_stack, th_mean)
        return distances_all.cpu(), distances_mean.cpu()

    # ------------------------------------------------------------------
    def gather_per_node(self, local_vals: torch.Tensor) -> torch.Tensor:
        """All-gather a [L] per-node metric into an [N] tensor."""
        col = local_vals.reshape(-1, 1).to(self.device)
        full = self.comm.all_gather_rows(self.layout, col)
        return full.reshape(-1).cpu()

    def gather_per_node_rows(self, local_rows: torch.Tensor) -> torch.Tensor:
        """All-gather an [L, ...] per-node payload into [N, ...].

        Used for metrics with a per-node vector/image payload
        (validation_as_vector, mesh_grid_density): rank 0 is the only
        rank that saves metrics, so payloads must cross ranks or a
        multi-rank run would silently persist rank 0's nodes only
        (ADVICE r1 item 1).
        """
        L = local_rows.shape[0]
        tail = local_rows.shape[1:]
        flat = local_rows.reshape(L, -1)
        # collectives don't carry bool; round-trip via uint8
        cast = flat.dtype == torch.bool
        if cast:
            flat = flat.to(torch.uint8)
        full = self.comm.all_gather_rows(self.layout, flat.to(self.device))
        if cast:
            full = full.to(torch.bool)
        return full.reshape(self.N, *tail).cpu()

    # ------------------------------------------------------------------
    def save_metrics(self, output_dir):
        if not self.is_root:
            return
        file_name = self.conf["problem_name"] + "_results.pt"
        torch.save(self.metrics, os.path.join(output_dir, file_name))

    def _print(self, msg):
        # honors the documented `verbose_evals` switch (reference
        # README.md:176 documents it but problems/*.py:210 prints
        # unconditionally — we follow the documentation)
        if self.is_root and self.conf.get("verbose_evals", True):
            print(msg, flush=True)
