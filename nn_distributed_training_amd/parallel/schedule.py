"""P2P schedule construction for the stacked engine.

Given the communication graph, the node→rank layout and this rank's
local nodes, build the per-round transfer plan the fused kernels
consume:

* packed remote receive buffer + {node -> destination row} views
  (irecv writes remote neighbor vectors straight into kernel-indexable
  rows — no staging copies);
* CSR neighbor lists over table rows (local stack rows first, then
  packed remote rows in sorted node order — both endpoints derive the
  identical plan deterministically);
* per-node degrees and optional Metropolis weights per CSR entry.

Separated from ops/stacked.py so the multi-rank plan+exchange path is
unit-testable on CPU with gloo (tests/test_schedule_multirank.py) —
the GPU boxes available during development host a single MI355X, so
this code must be correct by construction before an 8-GPU run.
"""

from __future__ import annotations

import torch

from .comm import Communicator, NodeLayout


def remote_plan(comm: Communicator, layout: NodeLayout, graph,
                n: int, device, dtype, width_factor: int = 1):
    """(remote_nodes sorted, remote_buf [R, w*n] or None,
    dests {node: buffer row view})."""
    if comm.world == 1:
        return [], None, {}
    _, recv_nodes = comm.edge_transfers(layout, list(graph.edges()))
    remote_nodes = sorted(recv_nodes)
    if not remote_nodes:
        return [], None, {}
    buf = torch.empty(
        len(remote_nodes), width_factor * n, device=device, dtype=dtype
    )
    dests = {j: buf[r] for r, j in enumerate(remote_nodes)}
    return remote_nodes, buf, dests


def row_map(local_nodes, remote_nodes):
    """Table-row index for every node this rank touches."""
    row_of = {i: li for li, i in enumerate(local_nodes)}
    L = len(local_nodes)
    for r, j in enumerate(remote_nodes):
        row_of[j] = L + r
    return row_of


def build_csr(graph, local_nodes, row_of, device, dtype,
              include_self=False, W=None):
    """(offsets int32 [L+1], indices int32, weights or None)."""
    offs = [0]
    idx = []
    wts = []
    for i in local_nodes:
        if include_self:
            idx.append(row_of[i])
            if W is not None:
                wts.append(float(W[i, i]))
        for j in graph.neighbors(i):
            idx.append(row_of[j])
            if W is not None:
                wts.append(float(W[i, j]))
        offs.append(len(idx))
    # pinned staging + non_blocking H2D: a pageable tensor-to-device
    # copy SYNCS the stream, and dynamic graphs rebuild the CSR often —
    # the sync was serializing host and GPU on the density bench
    def _to_dev(lst, dt):
        t = torch.tensor(lst, dtype=dt)
        if device is not None and torch.device(device).type == "cuda":
            return t.pin_memory().to(device, non_blocking=True)
        return t.to(device)

    offs_t = _to_dev(offs, torch.int32)
    idx_t = _to_dev(idx, torch.int32)
    w_t = _to_dev(wts, dtype) if W is not None else None
    return offs_t, idx_t, w_t


def degrees(graph, local_nodes, device):
    t = torch.tensor(
        [graph.degree(i) for i in local_nodes], dtype=torch.int32
    )
    if torch.device(device).type == "cuda":
        return t.pin_memory().to(device, non_blocking=True)
    return t.to(device)
