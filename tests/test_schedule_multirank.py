"""Multi-rank plan + exchange tests for the STACKED engine's comm path.

The stacked drivers exchange packed remote buffers built by
parallel/schedule.py through Communicator.exchange_rows. The GPU boxes
available in development host a single MI355X, so this path must be
proven on CPU: 2 gloo ranks run the exact plan + exchange code
(including the DSGT two-stack form) and verify every received row.
"""

import os
import pickle

import networkx as nx
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from nn_distributed_training_amd.parallel import schedule
from nn_distributed_training_amd.parallel.comm import (
    Communicator,
    NodeLayout,
)

N, NDIM = 4, 7


def test_row_map_and_csr_single_rank():
    g = nx.cycle_graph(N)
    layout = NodeLayout(N, 1)
    local = list(layout.nodes_of(0))
    row_of = schedule.row_map(local, [])
    offs, idx, w = schedule.build_csr(
        g, local, row_of, torch.device("cpu"), torch.float64,
        include_self=True,
        W=torch.eye(N) * 0.5 + 0.25 * torch.as_tensor(
            nx.to_numpy_array(g)
        ),
    )
    assert offs.tolist() == [0, 3, 6, 9, 12]  # self + 2 neighbors each
    assert w.shape[0] == idx.shape[0] == 12


def _worker(rank, world, port, out_dir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        torch.set_default_dtype(torch.float64)
        dev = torch.device("cpu")
        g = nx.cycle_graph(N)  # edges (0,1),(1,2),(2,3),(0,3)
        layout = NodeLayout(N, world)
        comm = Communicator(dev)
        local = list(layout.nodes_of(rank))
        L = len(local)

        # node i's vector = i + linspace: globally reconstructible
        def vec(i, salt=0.0):
            return torch.arange(NDIM, dtype=torch.float64) + 10.0 * i \
                + salt

        p_stack = torch.stack([vec(i) for i in local])
        y_stack = torch.stack([vec(i, salt=0.5) for i in local])

        # --- single-stack plan + exchange (DiNNO/DSGD form)
        remote_nodes, rbuf, dests = schedule.remote_plan(
            comm, layout, g, NDIM, dev, torch.float64
        )
        comm.exchange_rows(layout, list(g.edges()), [p_stack], [dests])
        got = {j: dests[j].clone() for j in remote_nodes}

        # --- two-stack plan (DSGT bundles [p | y])
        rn2, rbuf2, _ = schedule.remote_plan(
            comm, layout, g, NDIM, dev, torch.float64, width_factor=2
        )
        dp = {j: rbuf2[r, :NDIM] for r, j in enumerate(rn2)}
        dy = {j: rbuf2[r, NDIM:] for r, j in enumerate(rn2)}
        comm.exchange_rows(
            layout, list(g.edges()), [p_stack, y_stack], [dp, dy]
        )
        got_p = {j: dp[j].clone() for j in rn2}
        got_y = {j: dy[j].clone() for j in rn2}

        # row map + CSR consistency over local + remote rows
        row_of = schedule.row_map(local, remote_nodes)
        offs, idx, _ = schedule.build_csr(
            g, local, row_of, dev, torch.float64
        )

        with open(os.path.join(out_dir, f"r{rank}.pkl"), "wb") as f:
            pickle.dump(
                {
                    "local": local,
                    "remote_nodes": remote_nodes,
                    "got": {k: v.numpy() for k, v in got.items()},
                    "got_p": {k: v.numpy() for k, v in got_p.items()},
                    "got_y": {k: v.numpy() for k, v in got_y.items()},
                    "offs": offs.numpy(),
                    "idx": idx.numpy(),
                },
                f,
            )
    finally:
        dist.destroy_process_group()


def test_two_rank_stacked_exchange(tmp_path):
    mp.start_processes(
        _worker, args=(2, 29601, str(tmp_path)), nprocs=2, join=True,
        start_method="spawn",
    )

    def vec(i, salt=0.0):
        return (
            torch.arange(NDIM, dtype=torch.float64) + 10.0 * i + salt
        ).numpy()

    for rank in range(2):
        with open(tmp_path / f"r{rank}.pkl", "rb") as f:
            d = pickle.load(f)
        local = d["local"]
        # ring 0-1-2-3-0, layout [0,1] | [2,3]: rank 0 needs {2, 3}
        # (edges (1,2) and (0,3)), rank 1 needs {0, 1}
        expect_remote = [2, 3] if rank == 0 else [0, 1]
        assert d["remote_nodes"] == expect_remote
        for j in expect_remote:
            assert (d["got"][j] == vec(j)).all()
            assert (d["got_p"][j] == vec(j)).all()
            assert (d["got_y"][j] == vec(j, 0.5)).all()
        # CSR rows: local rows 0..L-1, remotes packed after
        L = len(local)
        assert d["offs"][-1] == sum(2 for _ in local)  # ring: deg 2
        assert set(d["idx"]) <= set(range(L + len(expect_remote)))
