"""Rank layout and neighbor exchange over RCCL (or gloo on CPU).

This replaces the reference's in-memory neighbor reads (a single process
holding all node replicas, e.g. optimizers/dsgd.py:43-46) with a true
multi-rank design:

* logical graph nodes are packed onto ranks in contiguous blocks
  (:class:`NodeLayout`) — 8 nodes on 1 GPU, 1 node per GPU on 8, or 32
  nodes across 8 for the scaling study;
* per communication round, every cross-rank graph edge becomes a
  point-to-point send/recv of that node's flat parameter bucket.  All of a
  round's transfers are issued in ONE ``dist.batch_isend_irecv`` group
  (ncclGroupStart/End under RCCL), so RCCL can drive each GPU's 7 xGMI
  links concurrently instead of serializing per edge — and deliberately
  NOT as a global all-reduce (the algorithms are neighbor-local).
  Explicit host-side edge-coloring (SURVEY.md §7 step 4 floated it) is
  deliberately NOT done: within a grouped launch RCCL already places
  each peer pair on its own channel/link, every pair here maps to a
  distinct xGMI link on a single node anyway (≤7 neighbors per GPU),
  and a hand-built coloring would serialize rounds of the schedule that
  the grouped launch runs concurrently;
* same-rank edges never touch the network: the consumer reads the local
  stack row directly;
* the only collectives are an all-gather of the parameter stacks at
  consensus-error evaluation time and an all-gather of robot positions for
  the dynamic graph (problems/dist_online_dense_problem).

Works without ``torch.distributed`` initialization (world_size == 1): every
exchange degenerates to local reads, which is the single-GPU packed mode
(BASELINE configs 1 and 2).
"""

from __future__ import annotations

import os
import sys
import time
from datetime import timedelta
from typing import Dict, List, Sequence

import torch
import torch.distributed as dist

#: deadlock watchdog: P2P waits that exceed this raise instead of
#: hanging the job (a disconnected dynamic graph or a desynchronized
#: schedule shows up as a stuck recv). Override: NDTA_COMM_TIMEOUT_S.
COMM_TIMEOUT_S = float(os.environ.get("NDTA_COMM_TIMEOUT_S", "300"))

#: NDTA_COMM_DEBUG=1 logs every exchange's schedule (peers/ops/bytes)
#: per rank to stderr — first-line diagnostic for a desynchronized or
#: hung multi-GPU round.
COMM_DEBUG = os.environ.get("NDTA_COMM_DEBUG", "0") not in ("0", "")


def _comm_log(rank: int, msg: str):
    if COMM_DEBUG:
        print(f"[ndta-comm rank {rank} t={time.monotonic():.3f}] {msg}",
              file=sys.stderr, flush=True)


def _wait_all(works, what: str, rank: int):
    """Wait for a batch of P2P works with the watchdog timeout.

    Backend split: under gloo, ``wait(timedelta)``
    honors the timeout and returns False on expiry — we raise.  Under
    NCCL/RCCL the per-work timeout is NOT honored by ``wait`` (it only
    stream-orders); the guard there is the process group's own watchdog
    thread, armed by ``init_from_env`` via the PG timeout +
    TORCH_NCCL_ASYNC_ERROR_HANDLING, which tears the job down with a
    rank-attributed error instead of hanging.  A host-side poll of
    ``is_completed`` is additionally run when NDTA_COMM_BLOCKING=1 (the
    debug mode) so a stuck transfer raises in *this* stack with `what`
    attached.
    """
    blocking = os.environ.get("NDTA_COMM_BLOCKING", "0") not in ("0", "")
    if blocking:
        deadline = time.monotonic() + COMM_TIMEOUT_S
        pending = list(works)
        while pending:
            pending = [w for w in pending if not w.is_completed()]
            if not pending:
                break
            if time.monotonic() > deadline:
                raise RuntimeError(
                    f"[rank {rank}] neighbor-exchange watchdog: {what} "
                    f"did not complete within {COMM_TIMEOUT_S}s — check "
                    "graph connectivity / schedule symmetry "
                    "(NDTA_COMM_DEBUG=1 logs the per-rank schedule)"
                )
            time.sleep(0.001)
        return
    for w in works:
        try:
            ok = w.wait(timedelta(seconds=COMM_TIMEOUT_S))
        except TypeError:  # backend without timeout support
            w.wait()
            ok = True
        if ok is False:
            raise RuntimeError(
                f"[rank {rank}] neighbor-exchange watchdog: {what} did "
                f"not complete within {COMM_TIMEOUT_S}s — check graph "
                "connectivity / schedule symmetry"
            )


class NodeLayout:
    """Contiguous block partition of N logical nodes over `world` ranks.

    Ranks r < N % world get ceil(N/world) nodes; the rest get floor.
    """

    def __init__(self, N: int, world: int):
        self.N = N
        self.world = world
        base, extra = divmod(N, world)
        counts = [base + (1 if r < extra else 0) for r in range(world)]
        starts = [0]
        for c in counts[:-1]:
            starts.append(starts[-1] + c)
        self.counts = counts
        self.starts = starts

    def rank_of(self, node: int) -> int:
        for r in range(self.world):
            if node < self.starts[r] + self.counts[r]:
                return r
        raise IndexError(node)

    def nodes_of(self, rank: int) -> range:
        return range(self.starts[rank], self.starts[rank] + self.counts[rank])

    def local_index(self, node: int, rank: int) -> int:
        return node - self.starts[rank]


class Communicator:
    """Neighbor exchange + eval collectives for one rank.

    If torch.distributed is not initialized this is a trivial single-rank
    communicator (all nodes local).
    """

    def __init__(self, device: torch.device):
        self.device = torch.device(device)
        if dist.is_available() and dist.is_initialized():
            self.rank = dist.get_rank()
            self.world = dist.get_world_size()
        else:
            self.rank = 0
            self.world = 1

    # ------------------------------------------------------------------
    def edge_transfers(self, layout: NodeLayout, edges: Sequence[tuple]):
        """(send_pairs, recv_nodes) for this rank given the edge list.

        send_pairs: {(local_node, peer_rank)}; recv_nodes: {remote_node}.
        Both endpoints of an edge derive the identical sorted transfer
        list, so batched P2P ops pair up deterministically.
        """
        r = self.rank
        recv_nodes = set()
        send_pairs = set()
        for a, b in edges:
            ra, rb = layout.rank_of(a), layout.rank_of(b)
            if ra == rb:
                continue
            if ra == r:
                recv_nodes.add(b)
                send_pairs.add((a, rb))
            elif rb == r:
                recv_nodes.add(a)
                send_pairs.add((b, ra))
        return send_pairs, recv_nodes

    def exchange_rows(
        self,
        layout: NodeLayout,
        edges: Sequence[tuple],
        stacks: Sequence[torch.Tensor],
        dests: Sequence[Dict[int, torch.Tensor]],
    ) -> None:
        """Batched P2P exchange of per-node rows for several stacks.

        ``stacks[k]`` is this rank's [L, d_k] tensor; ``dests[k]`` maps
        each remote neighbor node id to the (contiguous) destination row
        irecv writes into.  All sends/recvs of the round go in ONE
        batch_isend_irecv group so RCCL can spread them across xGMI
        links. No-op when world == 1.
        """
        if self.world == 1:
            return
        send_pairs, recv_nodes = self.edge_transfers(layout, edges)
        r = self.rank
        ops: List[dist.P2POp] = []
        for node, peer in sorted(send_pairs, key=lambda t: (t[1], t[0])):
            li = layout.local_index(node, r)
            for st in stacks:
                ops.append(
                    dist.P2POp(dist.isend, st[li].contiguous(), peer)
                )
        for j in sorted(recv_nodes, key=lambda j: (layout.rank_of(j), j)):
            peer = layout.rank_of(j)
            for d in dests:
                ops.append(dist.P2POp(dist.irecv, d[j], peer))
        if ops:
            _comm_log(
                self.rank,
                f"exchange_rows: {len(ops)} ops, "
                f"send={sorted(send_pairs, key=lambda t: (t[1], t[0]))}, "
                f"recv={sorted(recv_nodes)}, "
                f"stacks={[tuple(s.shape) for s in stacks]}",
            )
            _wait_all(dist.batch_isend_irecv(ops),
                      f"{len(ops)} P2P ops", self.rank)
            _comm_log(self.rank, "exchange_rows: done")

    # ------------------------------------------------------------------
    def exchange_node_vectors(
        self,
        layout: NodeLayout,
        edges: Sequence[tuple],
        local_stack: torch.Tensor,
    ) -> Dict[int, torch.Tensor]:
        """Per-round neighbor exchange.

        ``edges`` is the communication graph's edge list over global node
        ids; ``local_stack`` is this rank's [L, n] stack (row l = node
        layout.starts[rank]+l).  Returns {global_node_id: [n] vector} for
        every REMOTE node that is a neighbor of one of this rank's nodes.
        Local neighbors are not returned — the caller reads the stack.

        Deterministic schedule: both endpoints of an edge enumerate the
        same (node, peer) transfer list sorted by (peer_rank, node_id), so
        the batched P2P ops match up without any negotiation.
        """
        if self.world == 1:
            return {}

        r = self.rank
        send_pairs, recv_nodes = self.edge_transfers(layout, edges)
        n = local_stack.shape[1]
        recv_bufs: Dict[int, torch.Tensor] = {
            j: torch.empty(n, dtype=local_stack.dtype, device=self.device)
            for j in recv_nodes
        }

        ops: List[dist.P2POp] = []
        # one canonical order on both sides: sends sorted by
        # (peer, node), then recvs sorted by (peer, node)
        for node, peer in sorted(send_pairs, key=lambda t: (t[1], t[0])):
            row = local_stack[layout.local_index(node, r)].contiguous()
            ops.append(dist.P2POp(dist.isend, row, peer))
        for j in sorted(recv_nodes, key=lambda j: (layout.rank_of(j), j)):
            ops.append(dist.P2POp(dist.irecv, recv_bufs[j], layout.rank_of(j)))
        if ops:
            _comm_log(
                self.rank,
                f"exchange_node_vectors: {len(ops)} ops, "
                f"send={sorted(send_pairs, key=lambda t: (t[1], t[0]))}, "
                f"recv={sorted(recv_nodes)}, n={n}",
            )
            _wait_all(dist.batch_isend_irecv(ops),
                      f"{len(ops)} P2P ops", self.rank)
            _comm_log(self.rank, "exchange_node_vectors: done")
        return recv_bufs

    # ------------------------------------------------------------------
    def all_gather_stack(
        self, layout: NodeLayout, local_stack: torch.Tensor
    ) -> torch.Tensor:
        """Gather every node's vector into an [N, n] stack on all ranks.

        Used only at metric-evaluation time (consensus error needs all
        nodes) — never in the training hot loop.
        """
        if self.world == 1:
            return local_stack
        n = local_stack.shape[1]
        maxL = max(layout.counts)
        padded = torch.zeros(
            maxL, n, dtype=local_stack.dtype, device=self.device
        )
        padded[: local_stack.shape[0]] = local_stack
        out = [torch.empty_like(padded) for _ in range(self.world)]
        dist.all_gather(out, padded)
        rows = [out[rk][: layout.counts[rk]] for rk in range(self.world)]
        return torch.cat(rows, dim=0)

    # ------------------------------------------------------------------
    def all_gather_rows(self, layout: NodeLayout, local_rows: torch.Tensor):
        """All-gather small per-node payloads (e.g. [L, 2] robot positions)."""
        return self.all_gather_stack(layout, local_rows)

    # ------------------------------------------------------------------
    def barrier(self):
        if self.world > 1:
            dist.barrier()

    def broadcast_scalar(self, value: float, src: int = 0) -> float:
        if self.world == 1:
            return value
        t = torch.tensor([value], device=self.device, dtype=torch.float64)
        dist.broadcast(t, src)
        return t.item()


def init_from_env(backend: str | None = None) -> tuple:
    """Initialize torch.distributed from torchrun env vars if present.

    Returns (rank, world, local_rank). Safe to call when not launched via
    torchrun (returns (0, 1, 0) without initializing).
    """
    import os

    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return 0, 1, 0
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            # arm the NCCL watchdog: on a P2P/collective exceeding the
            # PG timeout the watchdog tears the job down with a
            # rank-attributed error instead of hanging forever.  Must be
            # set BEFORE init_process_group (read at PG construction).
            os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
        dist.init_process_group(
            backend=backend,
            timeout=timedelta(seconds=max(COMM_TIMEOUT_S, 60.0)),
        )
    rank = dist.get_rank()
    world = dist.get_world_size()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    return rank, world, local_rank
