"""Every shipped YAML in configs/ must stay runnable end-to-end.

Each config is loaded, shrunk (tiny data, 2 rounds, CPU) and executed
through its real experiment driver — the same code path as
``python -m nn_distributed_training_amd.experiments.<driver> cfg.yaml``.
This guards the two-part schema contract against
driver drift without needing a GPU or paper-scale runtimes.
"""

import copy
import glob
import os

import pytest
import torch
import yaml

CONFIGS = sorted(glob.glob(
    os.path.join(os.path.dirname(__file__), "..", "configs", "*.yaml")
))


def _shrink(conf):
    exp = conf["experiment"]
    exp["use_cuda"] = False
    if "graph" in exp:
        exp["graph"]["num_nodes"] = 4
    if "sweep" in exp:
        exp["sweep"]["num_nodes"] = [6]
    if "data" in exp:
        d = exp["data"]
        d["floorplan_size"] = 96
        d["num_walls"] = 3
        d["border_width"] = 10
        d["num_beams"] = 6
        d["beam_samps"] = 8
        d["collision_samps"] = 20
        d["fine_samps"] = 3
        d["num_validation_scans"] = 4
        if "num_nodes" in d:
            d["num_nodes"] = 3
        if "num_scans_in_window" in d:
            d["num_scans_in_window"] = 4
    it = exp.get("individual_training")
    if it:
        it["train_solo"] = False
    for p in conf["problem_configs"].values():
        oc = p["optimizer_config"]
        oc["outer_iterations"] = 2
        if "primal_iterations" in oc:
            oc["primal_iterations"] = 1
        p["train_batch_size"] = min(p["train_batch_size"], 16)
        p["val_batch_size"] = min(p["val_batch_size"], 32)
        p["metrics_config"]["evaluate_frequency"] = 2
        p["verbose_evals"] = False
    return conf


def _runner_for(name):
    from nn_distributed_training_amd.experiments import (
        dist_dense_ex,
        dist_mnist_ex,
        dist_mnist_scaling,
        dist_online_dense_ex,
    )

    if "scaling" in name:
        return dist_mnist_scaling.experiment
    if "online" in name:
        return dist_online_dense_ex.experiment
    if "mnist" in name:
        return dist_mnist_ex.experiment
    return dist_dense_ex.experiment


@pytest.mark.parametrize(
    "cfg_path", CONFIGS, ids=[os.path.basename(c) for c in CONFIGS]
)
def test_config_runs_end_to_end(cfg_path, tmp_path):
    torch.set_default_dtype(torch.float64)
    conf = _shrink(copy.deepcopy(yaml.safe_load(open(cfg_path))))
    conf["experiment"]["output_metadir"] = str(tmp_path)
    if "data_dir" in conf["experiment"]:
        conf["experiment"]["data_dir"] = str(tmp_path)
    shrunk = tmp_path / "cfg.yaml"
    yaml.safe_dump(conf, open(shrunk, "w"))

    _runner_for(os.path.basename(cfg_path))(str(shrunk))

    results = glob.glob(str(tmp_path / "*" / "*results*.pt")) + \
        glob.glob(str(tmp_path / "*" / "*summary*.pt"))
    assert results, "experiment wrote no metric artifacts"
