"""Static implicit-density problem (lidar occupancy field).

Capability parity with the reference's ``problems/dist_dense_problem.py``.
The reference version's ``evaluate_metrics`` lacked the ``at_end`` kwarg
the optimizers pass (dist_dense_problem.py:154 vs optimizers/dinno.py:100)
— fixed here so the class works with all three optimizers.
"""

from __future__ import annotations

import copy

import numpy as np
import torch

from .base import ProblemBase


class DistDensityProblem(ProblemBase):
    def __init__(self, graph, base_model, base_loss, train_sets, val_set,
                 device, conf):
        super().__init__(
            graph, base_model, base_loss, train_sets, val_set, device, conf
        )
        if "mesh_grid_density" in self.metrics:
            X, Y = np.meshgrid(val_set.lidar.xs, val_set.lidar.ys)
            mesh = np.stack(
                [X[::8, ::8].reshape(-1), Y[::8, ::8].reshape(-1)], axis=1
            )
            self.mesh_inputs = torch.as_tensor(
                mesh, dtype=torch.get_default_dtype()
            ).to(self.device)
            self.metrics["mesh_inputs"] = self.mesh_inputs.cpu()

    def local_batch_loss(self, i):
        locs, dens = self.next_batch(i)
        yh = self.models[i].forward(locs.to(self.device))
        return self.base_loss(torch.squeeze(yh), dens.to(self.device))

    def validate(self, i):
        val_loss = 0.0
        with torch.no_grad():
            for locs, dens in self.val_loader:
                locs = locs.to(self.device)
                dens = dens.to(self.device)
                yh = self.models[i].forward(locs)
                val_loss += self.base_loss(torch.squeeze(yh), dens).item()
        return val_loss

    def mesh_grid_density(self, i):
        with torch.no_grad():
            return self.models[i].forward(self.mesh_inputs)

    def evaluate_metrics(self, at_end=False):
        if self.stacked is not None:
            self.stacked.flush_to_models()
        evalprint = "| "
        for met_name in self.conf["metrics"]:
            if met_name == "consensus_error":
                distances_all, distances_mean = self.consensus_error()
                self.metrics[met_name].append(
                    (distances_all, distances_mean)
                )
                evalprint += "Consensus: {:.4f} - {:.4f} | ".format(
                    distances_mean.amin().item(),
                    distances_mean.amax().item(),
                )
            elif met_name == "validation_loss":
                if self.stacked is not None:
                    vl = self.stacked.validate_all()
                else:
                    vl = torch.tensor(
                        [self.validate(i) for i in self.local_nodes]
                    )
                val_losses = self.gather_per_node(vl)
                self.metrics[met_name].append(val_losses)
                evalprint += "Val Loss: {:.4f} - {:.4f} | ".format(
                    val_losses.amin().item(), val_losses.amax().item()
                )
            elif met_name == "mesh_grid_density":
                dens = [
                    self.mesh_grid_density(i) for i in self.local_nodes
                ]
                self.metrics[met_name].append(
                    torch.stack(dens).cpu() if dens else torch.zeros(0)
                )
            elif met_name == "forward_pass_count":
                self.metrics[met_name].append(self.forward_cnt)
                evalprint += "Num Forward: {} | ".format(self.forward_cnt)
            elif met_name == "current_epoch":
                ep = self.gather_per_node(
                    self.epoch_tracker[self.local_nodes]
                )
                self.metrics[met_name].append(copy.deepcopy(ep))
                evalprint += "Ep Range: {} - {} | ".format(
                    int(ep.amin().item()), int(ep.amax().item())
                )
            else:
                raise NameError("Unknown metric.")
        self._print(evalprint)
