"""Static implicit-density experiment driver.

Capability parity with the reference's ``experiments/dist_dense_ex.py``:
lidar datasets (random-pose or per-node trajectory splits), FourierNet
base model, BCE/MSE/L1 losses, solo baseline, per-problem loop. The
floorplan can be a PNG (``data.floorplan_img``) or, by default here, a
procedurally generated synthetic floorplan + waypoint set (no data files
in this environment).
"""

from __future__ import annotations

import copy
import os
import sys

import numpy as np
import torch
import yaml

from ..data.floorplan import synthetic_floorplan, synthetic_waypoints
from ..data.lidar import (
    Lidar2D,
    RandomPoseLidarDataset,
    TrajectoryLidarDataset,
)
from ..models.fourier import FourierNet
from ..optimizers import build_optimizer
from ..problems.dist_dense_problem import DistDensityProblem
from ..utils import graph_generation
from . import common


def build_lidar(data_conf: dict) -> Lidar2D:
    if "floorplan_img" in data_conf:
        img = data_conf["floorplan_img"]
    else:
        img = synthetic_floorplan(
            nx=data_conf.get("floorplan_size", 256),
            ny=data_conf.get("floorplan_size", 256),
            num_walls=data_conf.get("num_walls", 10),
            border_width=data_conf.get("border_width", 16),
            seed=data_conf.get("floorplan_seed", 0),
        )
        data_conf = dict(data_conf, border_width=0)  # already bordered
    return Lidar2D(
        img,
        data_conf["num_beams"],
        data_conf["beam_length"],
        data_conf["beam_samps"],
        data_conf.get("samp_distribution_factor", 1.0),
        data_conf.get("collision_samps", 50),
        data_conf.get("fine_samps", 3),
        border_width=data_conf.get("border_width", 0),
    )


def build_train_sets(lidar, data_conf, N):
    """Per-node train sets: trajectory split (one synthetic/recorded
    path per node) or random-pose split."""
    split = data_conf.get("split_type", "trajectory")
    if split == "trajectory":
        paths = synthetic_waypoints(
            lidar.img, N, seed=data_conf.get("floorplan_seed", 0)
        )
        return [
            TrajectoryLidarDataset(
                lidar, wp, data_conf["spline_res"],
                round_density=data_conf.get("round_density", True),
            )
            for wp in paths
        ]
    if split == "random":
        return [
            RandomPoseLidarDataset(
                lidar, data_conf["num_scans_per_node"],
                round_density=data_conf.get("round_density", True),
            )
            for _ in range(N)
        ]
    raise NameError(f"Unknown density split type: {split}")


def experiment(yaml_pth: str):
    with open(yaml_pth) as f:
        conf_dict = yaml.safe_load(f)
    exp_conf = conf_dict["experiment"]

    rank, world, local_rank = common.init_distributed()
    common.set_precision(exp_conf)
    torch.manual_seed(exp_conf.get("seed", 0))
    np.random.seed(exp_conf.get("seed", 0))

    output_dir = common.setup_run(yaml_pth, exp_conf, rank)

    graph_conf = dict(exp_conf["graph"])
    graph_conf.setdefault("seed", exp_conf.get("seed", 0))
    N, graph = graph_generation.generate_from_conf(graph_conf)
    if exp_conf["writeout"] and rank == 0:
        common.save_graph(graph, os.path.join(output_dir, "graph.gpickle"))

    data_conf = exp_conf["data"]
    if rank == 0:
        print("Generating lidar data ...")
    lidar = build_lidar(data_conf)
    train_subsets = build_train_sets(lidar, data_conf, N)
    val_set = RandomPoseLidarDataset(
        lidar,
        data_conf["num_validation_scans"],
        round_density=data_conf.get("round_density", True),
    )

    model_conf = exp_conf["model"]
    base_model = FourierNet(model_conf["shape"], scale=model_conf["scale"])
    base_loss = common.make_loss(exp_conf["loss"])
    device = common.select_device(exp_conf, local_rank)
    if rank == 0:
        print(f"Device is set to {device} (world size {world})")

    solo_confs = exp_conf["individual_training"]
    if solo_confs["train_solo"] and rank == 0:
        _train_solo_all(
            N, base_model, base_loss, train_subsets, val_set, device,
            solo_confs, exp_conf, output_dir,
        )

    cent_conf = exp_conf.get("centralized_training",
                             {"train_centralized": False})
    if cent_conf.get("train_centralized", False) and rank == 0:
        _train_centralized(
            base_model, base_loss, train_subsets, val_set, device,
            cent_conf, exp_conf, output_dir,
        )

    for prob_key, prob_conf in conf_dict["problem_configs"].items():
        opt_conf = prob_conf["optimizer_config"]
        prob = DistDensityProblem(
            graph, base_model, base_loss, train_subsets, val_set, device,
            prob_conf,
        )
        common.maybe_attach_stacked(prob, exp_conf, opt_conf)
        dopt = build_optimizer(prob, device, opt_conf)
        if rank == 0:
            print("-" * 55)
            print("Running problem: " + prob_conf["problem_name"])
        common.run_problem(prob, dopt, prob_conf, exp_conf, output_dir)


def _train_solo_all(N, base_model, base_loss, train_subsets, val_set,
                    device, solo_confs, exp_conf, output_dir):
    solo_results = {}
    print("Performing individual training ...")
    for i in range(N):
        model = copy.deepcopy(base_model).to(device)
        loader = torch.utils.data.DataLoader(
            train_subsets[i], solo_confs["train_batch_size"], shuffle=True
        )
        opt = torch.optim.Adam(model.parameters(), lr=solo_confs["lr"])
        for _ in range(solo_confs["epochs"]):
            for locs, dens in loader:
                opt.zero_grad()
                yh = model.forward(locs.to(device))
                base_loss(torch.squeeze(yh), dens.to(device)).backward()
                opt.step()
        with torch.no_grad():
            vloss = 0.0
            vloader = torch.utils.data.DataLoader(
                val_set, solo_confs["val_batch_size"]
            )
            for locs, dens in vloader:
                yh = model.forward(locs.to(device))
                vloss += base_loss(
                    torch.squeeze(yh), dens.to(device)
                ).item()
        solo_results[i] = {"validation_loss": vloss}
        if solo_confs["verbose"]:
            print(f"Node {i} - Validation Loss = {vloss:.4f}")
    if exp_conf["writeout"]:
        torch.save(
            solo_results, os.path.join(output_dir, "solo_results.pt")
        )


def _train_centralized(base_model, base_loss, train_subsets, val_set,
                       device, cent_conf, exp_conf, output_dir):
    """Pooled-data upper-bound baseline (reference
    centralized/online_density.ipynb cell 4: the same FourierNet
    trained centrally on the union of all nodes' scans), with a
    per-epoch validation-loss curve."""
    print("Performing centralized (pooled-data) training ...")
    model = copy.deepcopy(base_model).to(device)
    pooled = torch.utils.data.ConcatDataset(list(train_subsets))
    loader = torch.utils.data.DataLoader(
        pooled, cent_conf["train_batch_size"], shuffle=True
    )
    opt = torch.optim.Adam(model.parameters(), lr=cent_conf["lr"])
    curves = {"validation_loss": [], "epoch": []}
    for ep in range(cent_conf["epochs"]):
        for locs, dens in loader:
            opt.zero_grad()
            yh = model.forward(locs.to(device))
            base_loss(torch.squeeze(yh), dens.to(device)).backward()
            opt.step()
        with torch.no_grad():
            vloss = 0.0
            vloader = torch.utils.data.DataLoader(
                val_set, cent_conf["val_batch_size"]
            )
            for locs, dens in vloader:
                yh = model.forward(locs.to(device))
                vloss += base_loss(
                    torch.squeeze(yh), dens.to(device)
                ).item()
        curves["epoch"].append(ep)
        curves["validation_loss"].append(vloss)
        if cent_conf.get("verbose", False):
            print(f"Centralized epoch {ep} - Val Loss = {vloss:.4f}")
    if exp_conf["writeout"]:
        torch.save(
            curves,
            os.path.join(output_dir, "centralized_results.pt"),
        )


if __name__ == "__main__":
    yaml_pth = sys.argv[1]
    if not os.path.exists(yaml_pth):
        raise NameError("YAML configuration file does not exist, exiting!")
    experiment(yaml_pth)
