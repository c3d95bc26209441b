"""Distributed MNIST classification problem.

Capability parity with the reference's ``problems/dist_mnist_problem.py``
(same metric names including the reference's ``valdiation_as_vector``
gate typo being fixed to also accept the correctly spelled name), built on
the rank-aware ProblemBase.
"""

from __future__ import annotations

import copy

import torch

from .base import ProblemBase


class DistMNISTProblem(ProblemBase):
    def __init__(self, graph, base_model, base_loss, train_sets, val_set,
                 device, conf):
        super().__init__(
            graph, base_model, base_loss, train_sets, val_set, device, conf
        )

    def local_batch_loss(self, i):
        """One forward pass on node i's next local batch -> scalar loss
        with autograd graph (reference dist_mnist_problem.py:65-98)."""
        x, y = self.next_batch(i)
        yh = self.models[i].forward(x.to(self.device))
        return self.base_loss(yh, y.to(self.device))

    def validate(self, i):
        """Validation loss / top-1 accuracy / correctness vector of node
        i's model (reference dist_mnist_problem.py:111-132)."""
        model = self.models[i]
        with torch.no_grad():
            loss, correct, correct_list = 0.0, 0, []
            for x, y in self.val_loader:
                x, y = x.to(self.device), y.to(self.device)
                yh = model.forward(x)
                loss += self.base_loss(yh, y).item()
                pred = yh.argmax(dim=1, keepdim=True)
                cv = pred.eq(y.view_as(pred))
                correct += cv.sum().item()
                correct_list.append(cv)
        nval = len(self.val_loader.dataset)
        return loss / nval, correct / nval, torch.vstack(correct_list)

    def evaluate_metrics(self, at_end=False):
        want_val = any(
            m in self.metrics
            for m in (
                "validation_loss",
                "top1_accuracy",
                "validation_as_vector",
                "valdiation_as_vector",
            )
        )
        if want_val:
            if self.stacked is not None:
                # batched stacked-kernel validation: all local nodes in
                # one forward per val chunk
                vl, va, vc = self.stacked.validate_all()
                loc_losses = vl.tolist()
                loc_accs = va.tolist()
                valid_vecs = {
                    i: vc[li].reshape(-1, 1)
                    for li, i in enumerate(self.local_nodes)
                }
                self.stacked.flush_to_models()
            else:
                loc_losses, loc_accs, valid_vecs = [], [], {}
                for i in self.local_nodes:
                    l, a, v = self.validate(i)
                    loc_losses.append(l)
                    loc_accs.append(a)
                    valid_vecs[i] = v
            avg_losses = self.gather_per_node(torch.tensor(loc_losses))
            accs = self.gather_per_node(torch.tensor(loc_accs))

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
                self.metrics[met_name].append(avg_losses)
                evalprint += "Val Loss: {:.4f} - {:.4f} | ".format(
                    avg_losses.amin().item(), avg_losses.amax().item()
                )
            elif met_name == "top1_accuracy":
                self.metrics[met_name].append(accs)
                evalprint += "Top1: {:.2f} - {:.2f} |".format(
                    accs.amin().item(), accs.amax().item()
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
            elif met_name in ("validation_as_vector",
                              "valdiation_as_vector"):
                # per-sample correctness vectors, gathered across ranks
                # so rank 0 (the saver) holds every node's vector
                if valid_vecs:
                    loc = torch.stack(
                        [valid_vecs[i].reshape(-1)
                         for i in self.local_nodes]
                    )
                else:
                    loc = torch.zeros(0, len(self.val_set),
                                      dtype=torch.bool)
                full = self.gather_per_node_rows(loc)
                self.metrics[met_name].append(
                    {i: full[i].reshape(-1, 1) for i in range(self.N)}
                )
            else:
                raise NameError("Unknown metric.")

        self._print(evalprint)
