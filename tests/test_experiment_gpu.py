"""GPU end-to-end: the YAML experiment driver on engine=auto must run
the stacked HIP engine (not a silent eager fallback) and produce the
reference checkpoint layout."""

import os

import pytest
import torch
import yaml

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@requires_gpu
def test_mnist_experiment_hip_engine(tmp_path, monkeypatch):
    from nn_distributed_training_amd.experiments import dist_mnist_ex

    conf = {
        "experiment": {
            "name": "gpu_e2e",
            "data_dir": str(tmp_path / "data"),
            "output_metadir": str(tmp_path / "out"),
            "use_cuda": True,
            "writeout": True,
            "data_split_type": "hetero",
            "data_source": "synthetic",
            "train_samples": 1024,
            "val_samples": 256,
            "seed": 0,
            "loss": "NLL",
            "precision": "fp64",
            "engine": "hip",  # REQUIRE the stacked engine
            "graph": {"num_nodes": 4, "type": "cycle", "p": 0.5,
                      "gen_attempts": 20},
            "model": {"num_filters": 3, "kernel_size": 5,
                      "linear_width": 64},
            "individual_training": {
                "train_solo": False, "optimizer": "adam", "lr": 0.005,
                "epochs": 1, "train_batch_size": 64,
                "val_batch_size": 64, "verbose": False,
            },
        },
        "problem_configs": {
            "p1": {
                "problem_name": "dinno",
                "train_batch_size": 32,
                "val_batch_size": 128,
                "verbose_evals": False,
                "metrics": [
                    "forward_pass_count", "validation_loss",
                    "consensus_error", "top1_accuracy",
                    "current_epoch",
                ],
                "metrics_config": {"evaluate_frequency": 10},
                "optimizer_config": {
                    "alg_name": "dinno", "rho_init": 0.5,
                    "rho_scaling": 1.0003, "outer_iterations": 40,
                    "primal_iterations": 2,
                    "primal_optimizer": "adam",
                    "persistant_primal_opt": False,
                    "primal_lr_start": 0.005,
                    "primal_lr_finish": 0.001,
                    "lr_decay_type": "log", "profile": False,
                },
            },
            "p2": {
                "problem_name": "dsgd",
                "train_batch_size": 32,
                "val_batch_size": 128,
                "verbose_evals": False,
                "metrics": ["top1_accuracy", "consensus_error"],
                "metrics_config": {"evaluate_frequency": 20},
                "optimizer_config": {
                    "alg_name": "dsgd", "outer_iterations": 20,
                    "alpha0": 0.005, "mu": 0.001, "profile": False,
                },
            },
        },
    }
    pth = tmp_path / "conf.yaml"
    with open(pth, "w") as f:
        yaml.safe_dump(conf, f)
    dist_mnist_ex.experiment(str(pth))

    runs = list((tmp_path / "out").iterdir())
    assert len(runs) == 1
    files = {p.name for p in runs[0].iterdir()}
    assert "graph.gpickle" in files
    assert "dinno_results.pt" in files and "dsgd_results.pt" in files
    res = torch.load(
        os.path.join(runs[0], "dinno_results.pt"), weights_only=False
    )
    accs = res["top1_accuracy"][-1]
    # 40 DiNNO rounds on the easy synthetic task: clearly above chance
    assert accs.amin().item() > 0.3
    assert len(res["consensus_error"]) == 5  # k=0,10,20,30,39
    assert res["forward_pass_count"][-1] > 0
