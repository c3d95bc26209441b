"""The torch.profiler wrap (reference dist_mnist_ex.py:207-220 parity):
profile: true must produce a trace directory and step the profiler
per communication round."""

import networkx as nx
import torch

from nn_distributed_training_amd.data.mnist import (
    SyntheticMNIST,
    split_train_set,
)
from nn_distributed_training_amd.experiments import common
from nn_distributed_training_amd.models import MNISTConvNet
from nn_distributed_training_amd.optimizers import build_optimizer
from nn_distributed_training_amd.problems.dist_mnist_problem import (
    DistMNISTProblem,
)


def test_profiler_wrap_produces_trace(tmp_path):
    torch.manual_seed(0)
    graph = nx.cycle_graph(3)
    train = SyntheticMNIST(300, seed=0)
    val = SyntheticMNIST(60, seed=1)
    subsets = split_train_set(train, 3, "random")
    prob_conf = {
        "problem_name": "prof",
        "train_batch_size": 16,
        "val_batch_size": 60,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 1000},
        "optimizer_config": {
            "alg_name": "dsgd",
            "outer_iterations": 12,  # wait 1 + warmup 1 + 3x active 3
            "alpha0": 0.004,
            "mu": 0.001,
            "profile": True,
        },
    }
    pr = DistMNISTProblem(
        graph, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), prob_conf,
    )
    dopt = build_optimizer(pr, pr.device, prob_conf["optimizer_config"])
    common.run_problem(
        pr, dopt, prob_conf, {"writeout": True}, str(tmp_path)
    )
    trace_dir = tmp_path / "profopt_profile"
    assert trace_dir.exists()
    assert any(trace_dir.iterdir())  # tensorboard trace files written
    assert (tmp_path / "prof_results.pt").exists()
