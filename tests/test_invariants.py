"""Property-based invariants for the math-critical helpers.

Uses hypothesis to sweep shapes the example-based tests fix: Metropolis
mixing matrices must be symmetric doubly-stochastic for ANY graph,
NodeLayout must partition nodes for ANY (N, world), and the model specs'
flat-vector offsets must tile the parameter vector exactly.
"""

import networkx as nx
import numpy as np
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from nn_distributed_training_amd.models import (
    FFReLUNet,
    FourierNet,
    MNISTConvNet,
)
from nn_distributed_training_amd.models.spec import model_spec, param_layout
from nn_distributed_training_amd.parallel.comm import NodeLayout
from nn_distributed_training_amd.utils import graph_generation


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(2, 24),
    p=st.floats(0.2, 1.0),
    seed=st.integers(0, 10**6),
)
def test_metropolis_is_symmetric_doubly_stochastic(n, p, seed):
    g = nx.erdos_renyi_graph(n, p, seed=seed)
    W = graph_generation.get_metropolis(g)
    assert W.shape == (n, n)
    assert torch.all(W >= 0)
    torch.testing.assert_close(W, W.T, rtol=0, atol=0)
    torch.testing.assert_close(
        W.sum(dim=1), torch.ones(n, dtype=W.dtype), rtol=0, atol=1e-12
    )
    # off-diagonal support exactly matches the edge set
    for i, j in g.edges():
        assert W[i, j] > 0
    comp = nx.complement(g)
    for i, j in comp.edges():
        assert W[i, j] == 0


@settings(max_examples=60, deadline=None)
@given(N=st.integers(1, 64), world=st.integers(1, 16))
def test_node_layout_partitions(N, world):
    lay = NodeLayout(N, world)
    seen = []
    for r in range(world):
        nodes = list(lay.nodes_of(r))
        assert len(nodes) == lay.counts[r]
        for node in nodes:
            assert lay.rank_of(node) == r
            assert lay.local_index(node, r) == node - lay.starts[r]
        seen.extend(nodes)
    assert seen == list(range(N))
    # contiguous blocks, sizes differ by at most one
    assert max(lay.counts) - min(lay.counts) <= 1


@settings(max_examples=20, deadline=None)
@given(
    widths=st.lists(st.integers(1, 9), min_size=2, max_size=5),
)
def test_param_layout_tiles_the_vector(widths):
    model = FFReLUNet(widths)
    layout, n = param_layout(model)
    vec = torch.nn.utils.parameters_to_vector(model.parameters())
    assert n == vec.numel()
    off = 0
    for name, shape, o in layout:
        assert o == off, "offsets must be contiguous in vector order"
        off += int(torch.tensor(shape).prod()) if shape else 1
    assert off == n


def test_model_spec_offsets_match_named_parameters():
    for model in (
        MNISTConvNet(3, 5, 64),
        FourierNet([2, 16, 8, 1], scale=0.05),
        FFReLUNet([4, 8, 2]),
    ):
        spec = model_spec(model)
        vec = torch.nn.utils.parameters_to_vector(model.parameters())
        assert spec.n == vec.numel()
        params = dict(model.named_parameters())
        layout, _ = param_layout(model)
        offs = {name: off for name, _, off in layout}
        for layer in spec.layers:
            # the weight slice at w_off must BE the layer's weight
            w = [p for nm, p in params.items()
                 if offs[nm] == layer.w_off][0]
            torch.testing.assert_close(
                vec[layer.w_off : layer.w_off + w.numel()],
                w.reshape(-1),
                rtol=0,
                atol=0,
            )


@settings(max_examples=40, deadline=None)
@given(
    N=st.integers(2, 20),
    world=st.integers(1, 6),
    rank=st.integers(0, 5),
    p=st.floats(0.2, 0.9),
    seed=st.integers(0, 10**6),
)
def test_csr_plan_matches_graph(N, world, rank, p, seed):
    """The kernel-facing CSR must enumerate exactly each local node's
    neighborhood, with local stack rows first and remote rows mapped
    behind them — for any graph and any node->rank packing."""
    from nn_distributed_training_amd.parallel import schedule

    rank = rank % world
    g = nx.erdos_renyi_graph(N, p, seed=seed)
    lay = NodeLayout(N, world)
    local = list(lay.nodes_of(rank))

    # remote nodes: what edge_transfers would deliver for this rank
    remote = sorted(
        {j for i in local for j in g.neighbors(i)
         if lay.rank_of(j) != rank}
    )
    row_of = schedule.row_map(local, remote)
    offs, idx, w = schedule.build_csr(
        g, local, row_of, torch.device("cpu"), torch.float64,
        include_self=True, W=graph_generation.get_metropolis(g),
    )
    offs = offs.tolist()
    idx = idx.tolist()
    assert offs[0] == 0 and offs[-1] == len(idx)
    L = len(local)
    for li, i in enumerate(local):
        rows = idx[offs[li] : offs[li + 1]]
        assert rows[0] == li, "self row first under include_self"
        expect = {row_of[j] for j in g.neighbors(i)}
        assert set(rows[1:]) == expect
        for r in rows:
            assert 0 <= r < L + len(remote)
    # weights: each row's entries sum to 1 (self + Metropolis neighbors)
    wl = w.tolist()
    for li in range(L):
        assert abs(sum(wl[offs[li] : offs[li + 1]]) - 1.0) < 1e-9


@settings(max_examples=20, deadline=None)
@given(
    lengths=st.lists(st.integers(3, 40), min_size=1, max_size=4),
    batch=st.integers(1, 16),
    draws=st.integers(1, 60),
)
def test_stream_sampler_epoch_accounting(lengths, batch, draws):
    """Every batch column is a valid index; epochs advance exactly at
    consumed/len boundaries; each full permutation covers the dataset."""
    from nn_distributed_training_amd.ops.stacked import _StreamSampler

    epochs = [0] * len(lengths)

    def cb(li):
        epochs[li] += 1

    s = _StreamSampler(lengths, batch, torch.device("cpu"), seed=5,
                       epoch_cb=cb, stream_batches=8)
    for _ in range(draws):
        stream, stride, off = s.next_ref()
        for li, n in enumerate(lengths):
            col = stream[li, off : off + batch]
            assert int(col.max()) < n and int(col.min()) >= 0
    for li, n in enumerate(lengths):
        assert epochs[li] == (draws * batch) // n


def test_golden_dsgd_round_matches_matrix_form():
    """One DSGD round (full-batch, world=1) must equal the closed form
    theta' = W @ theta - alpha * grad(W @ theta) — the mixing is
    snapshot-synchronous by design (documented deviation from the
    reference's in-place Gauss-Seidel walk, optimizers/dsgd.py)."""
    from nn_distributed_training_amd.data.mnist import (
        SyntheticMNIST,
        split_train_set,
    )
    from nn_distributed_training_amd.optimizers.dsgd import DSGD
    from nn_distributed_training_amd.problems.dist_mnist_problem import (
        DistMNISTProblem,
    )

    torch.set_default_dtype(torch.float64)
    torch.manual_seed(4)
    N, B = 4, 32
    g = nx.cycle_graph(N)
    train = SyntheticMNIST(N * B, seed=0)
    val = SyntheticMNIST(32, seed=1)
    subsets = split_train_set(train, N, "random")
    conf = {
        "problem_name": "closedform",
        "train_batch_size": B,  # == per-node size: full batch
        "val_batch_size": 32,
        "data_seed": 2,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 10**9},
        "optimizer_config": {
            "alg_name": "dsgd", "outer_iterations": 1,
            "alpha0": 0.01, "mu": 0.001, "profile": False,
        },
    }
    pr = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    theta0 = pr.local_params_stack().clone()

    opt = DSGD(pr, pr.device, conf["optimizer_config"])
    opt.train()
    got = pr.local_params_stack()

    # closed form on a fresh problem (same seeds -> same batches)
    pr2 = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    for li, i in enumerate(pr2.local_nodes):
        torch.nn.utils.vector_to_parameters(
            theta0[li], pr2.models[i].parameters()
        )
    W = graph_generation.get_metropolis(g)
    mixed = W @ theta0
    alpha = 0.01 * (1 - 0.001 * 0.01)
    want = torch.empty_like(mixed)
    for li, i in enumerate(pr2.local_nodes):
        torch.nn.utils.vector_to_parameters(
            mixed[li], pr2.models[i].parameters()
        )
        loss = pr2.local_batch_loss(i)
        loss.backward()
        gvec = torch.cat(
            [p.grad.reshape(-1) for p in pr2.models[i].parameters()]
        )
        want[li] = mixed[li] - alpha * gvec
    torch.testing.assert_close(got, want, rtol=1e-12, atol=1e-12)


def test_lidar_scan_geometry():
    """Scan points stay within beam length of the pose; output is
    [num_beams * beam_samps, 3]; rounded densities are {0, 1}."""
    from nn_distributed_training_amd.data.floorplan import (
        synthetic_floorplan,
    )
    from nn_distributed_training_amd.data.lidar import Lidar2D

    img = synthetic_floorplan(nx=96, ny=96, num_walls=3,
                              border_width=10, seed=1)
    lidar = Lidar2D(img, 8, 0.25, 10, 1.0, 30, 3)
    # find a free pose
    pos = None
    for x in lidar.xs[::5]:
        for y in lidar.ys[::5]:
            if lidar.density.ev(x, y) < 0.4:
                pos = np.array([x, y])
                break
        if pos is not None:
            break
    scan = lidar.scan(pos)
    assert scan.shape == (8 * 10, 3)
    d = np.linalg.norm(scan[:, :2] - pos, axis=1)
    assert d.max() <= lidar.beam_len * 1.0001
    # density channel within spline overshoot tolerance of [0, 1]
    assert scan[:, 2].min() > -0.5 and scan[:, 2].max() < 1.5


def test_golden_dsgt_round_matches_matrix_form():
    """One DSGT round must equal the closed form
    p' = W @ (p - alpha*y);  y' = W @ y + g(p') - g_old
    with the init_grads bootstrap (y0 = g0 from the first batch)."""
    from nn_distributed_training_amd.data.mnist import (
        SyntheticMNIST,
        split_train_set,
    )
    from nn_distributed_training_amd.optimizers.dsgt import DSGT
    from nn_distributed_training_amd.problems.dist_mnist_problem import (
        DistMNISTProblem,
    )

    torch.set_default_dtype(torch.float64)
    torch.manual_seed(5)
    N, B = 4, 32
    g = nx.cycle_graph(N)
    train = SyntheticMNIST(N * B, seed=0)
    val = SyntheticMNIST(32, seed=1)
    subsets = split_train_set(train, N, "random")
    conf = {
        "problem_name": "closedform_dsgt",
        "train_batch_size": B,  # full batch: gradients are state-only
        "val_batch_size": 32,
        "data_seed": 6,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 10**9},
        "optimizer_config": {
            "alg_name": "dsgt", "outer_iterations": 1, "alpha": 0.01,
            "init_grads": True, "profile": False,
        },
    }
    pr = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    theta0 = pr.local_params_stack().clone()
    opt = DSGT(pr, pr.device, conf["optimizer_config"])
    opt.train()
    got = pr.local_params_stack()

    # closed form on a fresh problem with identical state
    pr2 = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    for li, i in enumerate(pr2.local_nodes):
        torch.nn.utils.vector_to_parameters(
            theta0[li], pr2.models[i].parameters()
        )

    def grads_at(stack):
        out = torch.empty_like(stack)
        for li, i in enumerate(pr2.local_nodes):
            torch.nn.utils.vector_to_parameters(
                stack[li], pr2.models[i].parameters()
            )
            pr2.local_batch_loss(i).backward()
            gs = []
            for p in pr2.models[i].parameters():
                gs.append(p.grad.reshape(-1).clone())
                p.grad.zero_()
            out[li] = torch.cat(gs)
        return out

    W = graph_generation.get_metropolis(g)
    y0 = grads_at(theta0)  # init_grads bootstrap consumes batch 1
    want = W @ (theta0 - 0.01 * y0)
    torch.testing.assert_close(got, want, rtol=1e-12, atol=1e-12)


def test_golden_dinno_round_matches_closed_form():
    """One DiNNO round, one primal Adam step, must equal the closed
    form: after dual ascent, g_total = pred_grad + dual' +
    2*rho*(deg*theta - sum_j th_reg_j), and torch Adam's first step is
    exactly theta - lr * g/(|g| + eps) (bias correction cancels at
    t=1). This pins the analytic penalty gradient the fused kernels
    also implement."""
    from nn_distributed_training_amd.data.mnist import (
        SyntheticMNIST,
        split_train_set,
    )
    from nn_distributed_training_amd.optimizers.dinno import DiNNO
    from nn_distributed_training_amd.problems.dist_mnist_problem import (
        DistMNISTProblem,
    )

    torch.set_default_dtype(torch.float64)
    torch.manual_seed(6)
    N, B = 4, 32
    lr, rho0, rho_scale = 0.004, 0.5, 1.001
    g = nx.cycle_graph(N)
    train = SyntheticMNIST(N * B, seed=0)
    val = SyntheticMNIST(32, seed=1)
    subsets = split_train_set(train, N, "random")
    conf = {
        "problem_name": "closedform_dinno",
        "train_batch_size": B,
        "val_batch_size": 32,
        "data_seed": 8,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 10**9},
        "optimizer_config": {
            "alg_name": "dinno", "rho_init": rho0,
            "rho_scaling": rho_scale, "outer_iterations": 1,
            "primal_iterations": 1, "primal_optimizer": "adam",
            "persistant_primal_opt": False, "primal_lr_start": lr,
            "primal_lr_finish": lr, "lr_decay_type": "constant",
            "profile": False,
        },
    }
    pr = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    theta0 = pr.local_params_stack().clone()
    opt = DiNNO(pr, pr.device, conf["optimizer_config"])
    opt.train()
    got = pr.local_params_stack()

    # closed form
    pr2 = DistMNISTProblem(
        g, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    for li, i in enumerate(pr2.local_nodes):
        torch.nn.utils.vector_to_parameters(
            theta0[li], pr2.models[i].parameters()
        )
    pred_grad = torch.empty_like(theta0)
    for li, i in enumerate(pr2.local_nodes):
        pr2.local_batch_loss(i).backward()
        gs = []
        for p in pr2.models[i].parameters():
            gs.append(p.grad.reshape(-1).clone())
            p.grad.zero_()
        pred_grad[li] = torch.cat(gs)

    rho = rho0 * rho_scale  # scaled before the round
    A = graph_generation.adjacency(g).astype(float)
    deg = torch.as_tensor(A.sum(1))
    S = torch.as_tensor(A) @ theta0  # sum of neighbor snapshots
    dual = rho * (deg[:, None] * theta0 - S)  # duals start at zero
    s_reg = 0.5 * (deg[:, None] * theta0 + S)  # sum_j (th_j + th_i)/2
    g_tot = pred_grad + dual + 2 * rho * (deg[:, None] * theta0 - s_reg)
    want = theta0 - lr * g_tot / (g_tot.abs() + 1e-8)
    torch.testing.assert_close(got, want, rtol=1e-10, atol=1e-10)
