"""Neighbor-vector gathering shared by the three optimizers.

Turns the communication graph + a local [L, n] snapshot (optionally with
per-node auxiliary vectors concatenated, e.g. DSGT's gradient tracker)
into per-node neighbor stacks, fetching remote rows through the
Communicator's batched P2P exchange. This is the multi-rank equivalent
of the reference's in-process neighbor reads (`torch.stack([ths[j] for
j in graph.neighbors(i)])`, reference optimizers/dinno.py:120-122 and
the plist walks in dsgd.py:37-46 / dsgt.py:58-75).
"""

from __future__ import annotations

from typing import Dict

import torch


def gather_neighbor_stacks(
    pr, local_stack: torch.Tensor
) -> Dict[int, torch.Tensor]:
    """{global node i -> [deg(i), d] stack of its neighbors' vectors}.

    ``local_stack`` is this rank's [L, d] snapshot (d = n, or 2n when a
    tracker is riding along). Remote rows arrive via one batched
    isend/irecv group; local rows are plain reads.
    """
    edges = list(pr.graph.edges())
    remote = pr.comm.exchange_node_vectors(pr.layout, edges, local_stack)

    out = {}
    rank = pr.comm.rank
    for i in pr.local_nodes:
        rows = []
        for j in pr.graph.neighbors(i):
            if pr.layout.rank_of(j) == rank:
                rows.append(local_stack[pr.layout.local_index(j, rank)])
            else:
                rows.append(remote[j])
        out[i] = (
            torch.stack(rows)
            if rows
            else torch.zeros(
                0,
                local_stack.shape[1],
                dtype=local_stack.dtype,
                device=local_stack.device,
            )
        )
    return out
