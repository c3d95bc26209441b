"""Checkpoint / resume for the decentralized optimizers.

The reference only ever WRITES artifacts (SURVEY.md §5.4 — no resume
path anywhere). Here every optimizer can periodically write a round
checkpoint and resume from it:

* ``optimizer_config.checkpoint_every: K`` — write
  ``<output_dir>/<problem_name>_ckpt_rank<r>.pt`` every K rounds
  (per-rank shards: each rank owns its local nodes);
* ``optimizer_config.resume_from: <dir-or-file>`` — restore node
  parameters, optimizer state (duals/rho, trackers, Adam moments) and
  the round counter, then continue.

Semantics: model/optimizer state restore exactly; the shuffled data
streams restart from their seeds (documented — convergence is
unaffected for these stochastic methods, bitwise replay of the
remaining rounds is not guaranteed).
"""

from __future__ import annotations

import os

import torch


def ckpt_path(directory: str, problem_name: str, rank: int) -> str:
    return os.path.join(
        directory, f"{problem_name}_ckpt_rank{rank}.pt"
    )


def resolve_resume_path(resume_from: str, problem_name: str,
                        rank: int) -> str:
    if os.path.isdir(resume_from):
        return ckpt_path(resume_from, problem_name, rank)
    return resume_from


def save_checkpoint(directory, pr, k, opt_state: dict):
    if not directory:
        raise ValueError(
            "checkpoint_every is set but no checkpoint_dir/output_dir "
            "is available"
        )
    os.makedirs(directory, exist_ok=True)
    if pr.stacked is not None:
        pr.stacked.flush_to_models()
    payload = {
        "round": k,
        "local_nodes": list(pr.local_nodes),
        "models": {
            i: pr.models[i].state_dict() for i in pr.local_nodes
        },
        "epoch_tracker": pr.epoch_tracker,
        "forward_cnt": pr.forward_cnt,
        "metrics": pr.metrics,
        "opt_state": opt_state,
    }
    path = ckpt_path(directory, pr.conf["problem_name"], pr.comm.rank)
    tmp = path + ".tmp"
    torch.save(payload, tmp)
    os.replace(tmp, path)  # atomic: never leave a torn checkpoint


def load_checkpoint(resume_from, pr) -> tuple:
    """Restore problem-side state; returns (start_round, opt_state)."""
    path = resolve_resume_path(
        resume_from, pr.conf["problem_name"], pr.comm.rank
    )
    payload = torch.load(path, map_location=pr.device,
                         weights_only=False)
    if payload["local_nodes"] != list(pr.local_nodes):
        # not an assert: must survive `python -O`, or a mismatched
        # world size would load a wrong packing
        raise RuntimeError(
            f"checkpoint {path} was written with node->rank packing "
            f"{payload['local_nodes']} but this rank hosts "
            f"{list(pr.local_nodes)} — resume with the same world size "
            "as the run that wrote the checkpoint"
        )
    for i in pr.local_nodes:
        pr.models[i].load_state_dict(payload["models"][i])
    pr.epoch_tracker = payload["epoch_tracker"]
    pr.forward_cnt = payload["forward_cnt"]
    pr.metrics = payload["metrics"]
    if pr.stacked is not None:
        # re-pack the stack from the restored modules
        rows = [
            torch.nn.utils.parameters_to_vector(
                pr.models[i].parameters()
            ).detach().to(pr.stacked.device, pr.stacked.dtype)
            for i in pr.local_nodes
        ]
        pr.stacked.theta.copy_(torch.stack(rows))
    return payload["round"] + 1, payload["opt_state"]
