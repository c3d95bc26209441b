"""End-to-end CPU tests for the density drivers (static + online)."""

import os

import torch
import yaml

from nn_distributed_training_amd.experiments import (
    dist_dense_ex,
    dist_online_dense_ex,
)


def _data_conf():
    return {
        "floorplan_size": 96,
        "num_walls": 4,
        "floorplan_seed": 0,
        "border_width": 10,
        "num_nodes": 3,
        "split_type": "trajectory",
        "num_scans_in_window": 3,
        "num_beams": 6,
        "beam_samps": 8,
        "beam_length": 0.25,
        "collision_samps": 20,
        "fine_samps": 3,
        "samp_distribution_factor": 1.0,
        "round_density": True,
        "spline_res": 2,
        "num_validation_scans": 6,
    }


def _exp_conf(tmp_path, extra_data=None):
    data = _data_conf()
    if extra_data:
        data.update(extra_data)
    return {
        "name": "tiny_dense",
        "output_metadir": str(tmp_path / "out"),
        "writeout": True,
        "use_cuda": False,
        "precision": "fp64",
        "engine": "torch",
        "seed": 0,
        "data": data,
        "loss": "BCE",
        "model": {"shape": [2, 16, 8, 1], "scale": 0.05},
        "individual_training": {
            "train_solo": False,
            "optimizer": "adam",
            "lr": 0.001,
            "epochs": 1,
            "train_batch_size": 64,
            "val_batch_size": 64,
            "verbose": False,
        },
    }


def _metrics_conf(extra_metrics=(), **mc):
    base = {
        "problem_name": None,
        "train_batch_size": 32,
        "val_batch_size": 64,
        "verbose_evals": False,
        "metrics": [
            "forward_pass_count",
            "validation_loss",
            "consensus_error",
            "current_epoch",
            *extra_metrics,
        ],
        "metrics_config": {"evaluate_frequency": 4, **mc},
    }
    return base


def test_static_density_dinno_and_dsgt(tmp_path):
    conf = {
        "experiment": dict(_exp_conf(tmp_path), graph={
            "num_nodes": 3, "type": "cycle", "p": 0.5,
            "gen_attempts": 20,
        }),
        "problem_configs": {
            "p1": dict(
                _metrics_conf(("mesh_grid_density",)),
                problem_name="dinno",
                optimizer_config={
                    "alg_name": "dinno", "rho_init": 0.3,
                    "rho_scaling": 1.0004, "outer_iterations": 5,
                    "primal_iterations": 2,
                    "primal_optimizer": "adam",
                    "persistant_primal_opt": False,
                    "primal_lr_start": 0.001,
                    "primal_lr_finish": 0.0005,
                    "lr_decay_type": "log", "profile": False,
                },
            ),
            "p2": dict(
                _metrics_conf(),
                problem_name="dsgt",
                optimizer_config={
                    "alg_name": "dsgt", "alpha": 0.001,
                    "outer_iterations": 5, "init_grads": True,
                    "profile": False,
                },
            ),
        },
    }
    pth = tmp_path / "conf.yaml"
    with open(pth, "w") as f:
        yaml.safe_dump(conf, f)
    dist_dense_ex.experiment(str(pth))

    runs = list((tmp_path / "out").iterdir())
    files = {p.name for p in runs[0].iterdir()}
    assert "dinno_results.pt" in files and "dsgt_results.pt" in files
    res = torch.load(
        os.path.join(runs[0], "dinno_results.pt"), weights_only=False
    )
    assert len(res["validation_loss"]) == 2  # evals at k=0 and final
    assert "mesh_inputs" in res
    assert res["mesh_grid_density"][0].shape[0] == 3  # per node


def test_online_density_dynamic_graph(tmp_path):
    conf = {
        "experiment": _exp_conf(tmp_path),
        "problem_configs": {
            "p1": dict(
                _metrics_conf(
                    ("train_loss_moving_average", "current_position",
                     "current_graph"),
                    tloss_decay=0.2,
                    mesh_only_at_end=True,
                ),
                problem_name="dsgd",
                comm_radius=500.0,
                dynamic_graph=True,
                save_models=True,
                optimizer_config={
                    "alg_name": "dsgd", "alpha0": 0.001, "mu": 0.001,
                    "outer_iterations": 6, "profile": False,
                },
            ),
        },
    }
    pth = tmp_path / "conf.yaml"
    with open(pth, "w") as f:
        yaml.safe_dump(conf, f)
    dist_online_dense_ex.experiment(str(pth))

    runs = list((tmp_path / "out").iterdir())
    files = {p.name for p in runs[0].iterdir()}
    assert "dsgd_results.pt" in files
    assert "dsgd_models.pt" in files  # save_models
    res = torch.load(
        os.path.join(runs[0], "dsgd_results.pt"), weights_only=False
    )
    # dynamic graph metrics recorded
    assert len(res["current_position"]) == 3  # evals at k=0,4,5
    assert res["current_position"][0].shape == (3, 2)
    assert len(res["current_graph"]) == 3
    assert res["current_graph"][0].number_of_nodes() == 3
    # sliding windows advanced -> positions changed between evals
    assert not (res["current_position"][0]
                == res["current_position"][-1]).all()
    assert len(res["train_loss_moving_average"]) == 3
