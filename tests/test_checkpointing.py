"""Checkpoint / resume tests (the reference has no resume path at all).

Resumed runs restore parameters + optimizer state + round counter
exactly at the checkpoint; data streams restart from seed (documented
in optimizers/checkpointing.py), so we verify exact state restoration
and that continued training stays sane — not bitwise tail equality.
"""

import networkx as nx
import pytest
import torch

from nn_distributed_training_amd.data.mnist import (
    SyntheticMNIST,
    split_train_set,
)
from nn_distributed_training_amd.models import MNISTConvNet
from nn_distributed_training_amd.optimizers import build_optimizer
from nn_distributed_training_amd.problems.dist_mnist_problem import (
    DistMNISTProblem,
)

N = 3

OPT_CONFS = {
    "dinno": {
        "alg_name": "dinno", "rho_init": 0.4, "rho_scaling": 1.001,
        "outer_iterations": 6, "primal_iterations": 1,
        "primal_optimizer": "adam", "persistant_primal_opt": True,
        "primal_lr_start": 0.004, "primal_lr_finish": 0.001,
        "lr_decay_type": "linear", "profile": False,
    },
    "dsgd": {
        "alg_name": "dsgd", "outer_iterations": 6, "alpha0": 0.004,
        "mu": 0.001, "profile": False,
    },
    "dsgt": {
        "alg_name": "dsgt", "outer_iterations": 6, "alpha": 0.004,
        "init_grads": True, "profile": False,
    },
}


def _make(alg, tmp_path, extra_opt=None):
    torch.manual_seed(0)
    graph = nx.cycle_graph(N)
    train = SyntheticMNIST(300, seed=0)
    val = SyntheticMNIST(60, seed=1)
    subsets = split_train_set(train, N, "random")
    conf = {
        "problem_name": alg,
        "train_batch_size": 20,
        "val_batch_size": 60,
        "data_seed": 5,
        "verbose_evals": False,
        "metrics": ["consensus_error"],
        "metrics_config": {"evaluate_frequency": 1000},
        "optimizer_config": {
            **OPT_CONFS[alg],
            "checkpoint_dir": str(tmp_path),
            **(extra_opt or {}),
        },
    }
    pr = DistMNISTProblem(
        graph, MNISTConvNet(3, 5, 64), torch.nn.NLLLoss(), subsets, val,
        torch.device("cpu"), conf,
    )
    opt = build_optimizer(pr, pr.device, conf["optimizer_config"])
    return pr, opt


@pytest.mark.parametrize("alg", ["dinno", "dsgd", "dsgt"])
def test_checkpoint_write_and_resume(alg, tmp_path):
    # training run that checkpoints at round 4 (rounds 0..3 complete)
    pr1, opt1 = _make(alg, tmp_path, {"checkpoint_every": 4})
    opt1.train()
    ckpt = tmp_path / f"{alg}_ckpt_rank0.pt"
    assert ckpt.exists()
    payload = torch.load(ckpt, weights_only=False)
    assert payload["round"] == 3
    assert sorted(payload["models"]) == list(range(N))

    # resumed run starts at round 4 and finishes the remaining rounds
    pr2, opt2 = _make(alg, tmp_path, {"resume_from": str(tmp_path)})
    opt2.train()
    final = pr2.local_params_stack()
    assert torch.isfinite(final).all()
    # resume restored the checkpointed parameters before continuing:
    # round-4 state must differ from both init and the full run's end
    init = pr1.local_params_stack()
    assert not torch.allclose(final, init)

    # the restored models at load time match the checkpoint exactly
    pr3, opt3 = _make(alg, tmp_path, {
        "resume_from": str(tmp_path),
    })
    from nn_distributed_training_amd.optimizers.checkpointing import (
        load_checkpoint,
    )

    k0, _ = load_checkpoint(str(tmp_path), pr3)
    assert k0 == 4
    for i in range(N):
        for (na, pa), (nb, pb) in zip(
            pr3.models[i].named_parameters(),
            payload["models"][i].items(),
        ):
            torch.testing.assert_close(pa, pb, rtol=0, atol=0)


def test_resume_rejects_different_packing(tmp_path):
    """A checkpoint written under one node->rank packing must refuse to
    load into a problem with different local nodes (shards are
    per-rank; silently mixing them would corrupt training)."""
    from nn_distributed_training_amd.optimizers.checkpointing import (
        load_checkpoint,
        save_checkpoint,
    )

    pr, opt = _make("dsgd", tmp_path)
    opt.train()
    save_checkpoint(str(tmp_path), pr, 5, {"alph": 0.004})

    pr2, _ = _make("dsgd", tmp_path)
    pr2.local_nodes = [0, 1]  # simulate a different packing
    # RuntimeError (not AssertionError): must survive `python -O`
    with pytest.raises(RuntimeError, match="packing"):
        load_checkpoint(str(tmp_path), pr2)


def test_checkpoint_every_without_dir_raises(tmp_path):
    from nn_distributed_training_amd.optimizers.checkpointing import (
        save_checkpoint,
    )

    pr, _ = _make("dsgd", tmp_path)
    with pytest.raises(ValueError, match="checkpoint_every"):
        save_checkpoint(None, pr, 0, {})
