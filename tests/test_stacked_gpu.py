"""Stacked-HIP-engine vs golden-torch parity on a real GPU.

Full-batch configuration (train_batch_size == per-node dataset size), so
both engines see identical batch content every iteration and the final
parameter stacks must agree to fp64 GEMM-reduction-order tolerance.
"""

import copy

import networkx as nx
import pytest
import torch

from nn_distributed_training_amd.data.mnist import (
    SyntheticMNIST,
    split_train_set,
)
from nn_distributed_training_amd.models import MNISTConvNet, FourierNet
from nn_distributed_training_amd.optimizers import build_optimizer
from nn_distributed_training_amd.problems.dist_mnist_problem import (
    DistMNISTProblem,
)
from nn_distributed_training_amd.ops.stacked import StackedEngine

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)

N_NODES = 4
PER_NODE = 64  # samples per node; == batch size for determinism


def _build_problem(conf):
    torch.manual_seed(11)
    graph = nx.cycle_graph(N_NODES)
    train = SyntheticMNIST(PER_NODE * N_NODES, seed=0)
    val = SyntheticMNIST(64, seed=1)
    subsets = split_train_set(train, N_NODES, "random")
    base_model = MNISTConvNet(3, 5, 64)
    return DistMNISTProblem(
        graph, base_model, torch.nn.NLLLoss(), subsets, val,
        torch.device("cuda"), conf,
    )


def _conf(alg_conf):
    return {
        "problem_name": "parity",
        "train_batch_size": PER_NODE,
        "val_batch_size": 64,
        "data_seed": 3,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 1000},
        "optimizer_config": alg_conf,
    }


ALG_CONFS = {
    "dinno": {
        "alg_name": "dinno",
        "rho_init": 0.4,
        "rho_scaling": 1.0005,
        "outer_iterations": 3,
        "primal_iterations": 2,
        "primal_optimizer": "adam",
        "persistant_primal_opt": False,
        "primal_lr_start": 0.004,
        "primal_lr_finish": 0.001,
        "lr_decay_type": "log",
        "profile": False,
    },
    "dsgd": {
        "alg_name": "dsgd",
        "outer_iterations": 3,
        "alpha0": 0.004,
        "mu": 0.001,
        "profile": False,
    },
    "dsgt": {
        "alg_name": "dsgt",
        "outer_iterations": 3,
        "alpha": 0.004,
        "init_grads": True,
        "profile": False,
    },
}


@requires_gpu
def test_stacked_dinno_hipgraph_matches_golden(monkeypatch):
    """hipGraph capture+replay path == golden torch (4 rounds: two
    warmups, one capture+replay, one replay)."""
    monkeypatch.setenv("NDTA_GRAPHS", "1")
    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS["dinno"]))
    conf["optimizer_config"]["outer_iterations"] = 4

    pr_g = _build_problem(conf)
    opt_g = build_optimizer(pr_g, pr_g.device, conf["optimizer_config"])
    opt_g.train()
    golden = pr_g.local_params_stack()

    pr_s = _build_problem(conf)
    pr_s.stacked = StackedEngine(pr_s)
    opt_s = build_optimizer(pr_s, pr_s.device, conf["optimizer_config"])
    opt_s.train()
    # the graph path must actually have been used (not a silent fallback)
    torch.testing.assert_close(
        pr_s.stacked.theta, golden, rtol=1e-8, atol=1e-8
    )


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float64, torch.float32])
@pytest.mark.parametrize("alg", ["dinno", "dsgd", "dsgt"])
def test_stacked_matches_golden(alg, dtype):
    torch.set_default_dtype(dtype)
    if dtype == torch.float64:
        tol = dict(rtol=1e-8, atol=1e-8)  # reduction-order only
    elif alg == "dinno":
        # Adam's sqrt/eps nonlinearity amplifies fp32 rounding across
        # the 6 primal steps + dual ascent; fp64 is the exactness check
        tol = dict(rtol=5e-2, atol=5e-3)
    else:
        tol = dict(rtol=5e-3, atol=5e-4)
    conf = _conf(copy.deepcopy(ALG_CONFS[alg]))

    # golden eager torch engine
    pr_g = _build_problem(conf)
    opt_g = build_optimizer(pr_g, pr_g.device, conf["optimizer_config"])
    opt_g.train()
    golden = pr_g.local_params_stack()

    # stacked HIP engine
    pr_s = _build_problem(conf)
    pr_s.stacked = StackedEngine(pr_s)
    opt_s = build_optimizer(pr_s, pr_s.device, conf["optimizer_config"])
    opt_s.train()
    stacked = pr_s.stacked.theta

    torch.testing.assert_close(stacked, golden, **tol)


@requires_gpu
def test_stacked_dinno_isolated_node_matches_golden():
    """A degree-0 node must stay FROZEN in both engines.

    The golden engine skips the whole primal update for a node with no
    neighbors (the reference crashes there: optimizers/dinno.py:121
    torch.stack of an empty list); the fused-step kernel freezes
    deg==0 rows to match. Dynamic disk graphs can produce isolation,
    so this is a real runtime case, not just an API corner.
    """
    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS["dinno"]))

    def _build_isolated(conf):
        torch.manual_seed(11)
        graph = nx.path_graph(N_NODES - 1)  # 0-1-2 chain
        graph.add_node(N_NODES - 1)         # node 3 isolated
        train = SyntheticMNIST(PER_NODE * N_NODES, seed=0)
        val = SyntheticMNIST(64, seed=1)
        subsets = split_train_set(train, N_NODES, "random")
        base_model = MNISTConvNet(3, 5, 64)
        return DistMNISTProblem(
            graph, base_model, torch.nn.NLLLoss(), subsets, val,
            torch.device("cuda"), conf,
        )

    pr_g = _build_isolated(conf)
    init = pr_g.local_params_stack().clone()
    opt_g = build_optimizer(pr_g, pr_g.device, conf["optimizer_config"])
    opt_g.train()
    golden = pr_g.local_params_stack()
    # golden really froze the isolated node (guards the premise)
    torch.testing.assert_close(golden[-1], init[-1], rtol=0, atol=0)
    assert not torch.allclose(golden[0], init[0])

    pr_s = _build_isolated(conf)
    pr_s.stacked = StackedEngine(pr_s)
    opt_s = build_optimizer(pr_s, pr_s.device, conf["optimizer_config"])
    opt_s.train()
    torch.testing.assert_close(
        pr_s.stacked.theta, golden, rtol=1e-8, atol=1e-8
    )


@requires_gpu
@pytest.mark.parametrize("alg", ["dinno", "dsgt"])
def test_stacked_density_matches_golden(alg):
    """FourierNet + BCE density problem: stacked engine vs golden torch
    over 3 rounds (equal-size random-pose node datasets, full batch)."""
    import numpy as np

    from nn_distributed_training_amd.data.floorplan import (
        synthetic_floorplan,
    )
    from nn_distributed_training_amd.data.lidar import (
        Lidar2D,
        RandomPoseLidarDataset,
    )
    from nn_distributed_training_amd.problems.dist_dense_problem import (
        DistDensityProblem,
    )

    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS[alg]))

    def build():
        torch.manual_seed(7)
        np.random.seed(7)
        img = synthetic_floorplan(nx=96, ny=96, num_walls=3,
                                  border_width=10, seed=0)
        lidar = Lidar2D(img, 6, 0.25, 8, 1.0, 20, 3)
        sets = [
            RandomPoseLidarDataset(lidar, 4) for _ in range(N_NODES)
        ]
        val = RandomPoseLidarDataset(lidar, 4)
        per = len(sets[0])
        c = dict(conf, train_batch_size=per)
        graph = nx.cycle_graph(N_NODES)
        model = FourierNet([2, 16, 8, 1], scale=0.05)
        return DistDensityProblem(
            graph, model, torch.nn.BCELoss(), sets, val,
            torch.device("cuda"), c,
        ), c

    pr_g, c = build()
    opt_g = build_optimizer(pr_g, pr_g.device, c["optimizer_config"])
    opt_g.train()
    golden = pr_g.local_params_stack()

    pr_s, c = build()
    pr_s.stacked = StackedEngine(pr_s)
    opt_s = build_optimizer(pr_s, pr_s.device, c["optimizer_config"])
    opt_s.train()
    torch.testing.assert_close(
        pr_s.stacked.theta, golden, rtol=1e-8, atol=1e-8
    )


@requires_gpu
@pytest.mark.parametrize("alg", ["dinno", "dsgt"])
def test_stacked_checkpoint_resume(alg, tmp_path):
    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS[alg]))
    conf["optimizer_config"]["outer_iterations"] = 6
    conf["optimizer_config"]["checkpoint_every"] = 4
    conf["optimizer_config"]["checkpoint_dir"] = str(tmp_path)

    pr = _build_problem(conf)
    pr.stacked = StackedEngine(pr)
    opt = build_optimizer(pr, pr.device, conf["optimizer_config"])
    opt.train()
    ckpt = tmp_path / "parity_ckpt_rank0.pt"
    assert ckpt.exists()

    conf2 = _conf(copy.deepcopy(ALG_CONFS[alg]))
    conf2["optimizer_config"]["outer_iterations"] = 6
    conf2["optimizer_config"]["resume_from"] = str(tmp_path)
    pr2 = _build_problem(conf2)
    pr2.stacked = StackedEngine(pr2)
    opt2 = build_optimizer(pr2, pr2.device, conf2["optimizer_config"])
    init_theta = pr2.stacked.theta.clone()  # pre-train (fresh init)
    opt2.train()
    assert torch.isfinite(pr2.stacked.theta).all()
    payload = torch.load(ckpt, weights_only=False)
    assert payload["round"] == 3
    # the resumed run trained (moved past the restored checkpoint) and
    # did not restart from the fresh initialization
    assert not torch.allclose(pr2.stacked.theta, init_theta)


@requires_gpu
@pytest.mark.parametrize("dtype", [torch.float64, torch.float32])
def test_fused_mnist_step_matches_layered(dtype, monkeypatch):
    """The one-launch fused train-step's gradient == the layered
    kernel chain's gradient for the same batch."""
    torch.set_default_dtype(dtype)
    conf = _conf(copy.deepcopy(ALG_CONFS["dsgd"]))

    def grads(fused):
        monkeypatch.setenv("NDTA_FUSED", "1" if fused else "0")
        pr = _build_problem(conf)
        pr.stacked = StackedEngine(pr)
        eng = pr.stacked
        assert eng.fused_step_available() == fused
        if fused:
            off = eng.fused_advance()
            eng.run_fused_mnist(off=off)
            eng.reduce_fused_grad()
        else:
            # force the LAYERED chain (next_batch skips the gather on
            # the fused-fc path, whose grads are covered elsewhere)
            monkeypatch.setenv("NDTA_FC_BLOCK", "0")
            xb, yb = eng.next_batch()
            eng.forward(xb, train_skip_logp=True)
            eng.backward(xb, yb)
            monkeypatch.delenv("NDTA_FC_BLOCK")
        return eng.grad.clone()

    g_layered = grads(False)
    g_fused = grads(True)
    tol = (
        dict(rtol=1e-10, atol=1e-12)
        if dtype == torch.float64
        else dict(rtol=2e-3, atol=1e-5)
    )
    torch.testing.assert_close(g_fused, g_layered, **tol)


@requires_gpu
def test_stacked_validation_matches_eager():
    """Stacked batched validation == eager per-node torch validation."""
    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS["dsgd"]))
    pr = _build_problem(conf)
    # desynchronize the replicas so per-node metrics differ
    for i in pr.local_nodes:
        with torch.no_grad():
            for p in pr.models[i].parameters():
                p.add_(0.01 * (i + 1) * torch.randn_like(p))
    eager = [pr.validate(i) for i in pr.local_nodes]

    pr.stacked = StackedEngine(pr)
    vl, va, vc = pr.stacked.validate_all()
    for li, (l, a, v) in enumerate(eager):
        assert abs(vl[li].item() - l) < 1e-9
        assert abs(va[li].item() - a) < 1e-12
        assert vc[li].sum().item() == v.sum().item()


@requires_gpu
def test_stacked_fourier_forward_matches_module():
    """Stacked FourierNet forward == eager module forward (fp64)."""
    torch.set_default_dtype(torch.float64)
    torch.manual_seed(5)
    dev = torch.device("cuda")
    model = FourierNet([2, 32, 16, 1], scale=0.05).to(dev)

    class _FakeProblem:
        pass

    import types

    from nn_distributed_training_amd.models.spec import model_spec
    from nn_distributed_training_amd.ops import get_ext

    ext = get_ext()
    spec = model_spec(model)
    th = torch.nn.utils.parameters_to_vector(model.parameters()) \
        .detach().reshape(1, -1).contiguous()
    B = 64
    x = torch.randn(B, 2, device=dev)
    cur = x
    for layer in spec.layers:
        out = torch.empty(B, layer.out_dim, device=dev)
        z = torch.empty_like(out) if layer.activation == "sin_relu" else None
        from nn_distributed_training_amd.ops.stacked import ACT_IDS

        ext.linear_fwd(
            cur.contiguous(), th, out, z, layer.w_off, layer.b_off,
            B, layer.in_dim, layer.out_dim, ACT_IDS[layer.activation],
            layer.scale,
        )
        cur = out
    ref = model(x)
    torch.testing.assert_close(cur, ref, rtol=1e-10, atol=1e-10)


@requires_gpu
@pytest.mark.parametrize("alg", ["dinno", "dsgt"])
def test_chain_dispatch_matches_python_loops(alg, monkeypatch):
    """The C++ fwd/bwd chain dispatchers launch exactly the kernels
    the python per-op loops launch: bitwise-identical parameters after
    3 rounds (NDTA_PY_CHAIN A/B)."""
    torch.set_default_dtype(torch.float64)
    conf = _conf(copy.deepcopy(ALG_CONFS[alg]))

    def run(py_chain):
        monkeypatch.setenv("NDTA_PY_CHAIN", "1" if py_chain else "0")
        monkeypatch.setenv("NDTA_FC_BLOCK", "0")  # exercise the chains
        pr = _build_problem(conf)
        pr.stacked = StackedEngine(pr)
        assert pr.stacked._use_chain == (not py_chain)
        opt = build_optimizer(pr, pr.device, conf["optimizer_config"])
        opt.train()
        return pr.stacked.theta.clone()

    a = run(False)
    b = run(True)
    # conv dW accumulates atomically -> run-to-run reassociation at
    # ~1e-13; anything above 1e-10 would be a real dispatch difference
    torch.testing.assert_close(a, b, rtol=1e-10, atol=1e-10)
