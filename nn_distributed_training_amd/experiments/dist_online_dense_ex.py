"""Online (streaming) density-mapping experiment driver.

Capability parity with the reference's
``experiments/dist_online_dense_ex.py`` (the paper's headline robotics
demo): per-node OnlineTrajectoryLidarDataset sliding windows, dynamic
position-based communication graph rebuilt every round, all three
optimizers, solo baseline, seeded reproducibility.
"""

from __future__ import annotations

import os
import sys

import numpy as np
import torch
import yaml

from ..data.floorplan import synthetic_waypoints
from ..data.lidar import OnlineTrajectoryLidarDataset, RandomPoseLidarDataset
from ..models.fourier import FourierNet
from ..optimizers import build_optimizer
from ..problems.dist_online_dense_problem import DistOnlineDensityProblem
from . import common
from .dist_dense_ex import build_lidar


def experiment(yaml_pth: str):
    with open(yaml_pth) as f:
        conf_dict = yaml.safe_load(f)
    exp_conf = conf_dict["experiment"]

    rank, world, local_rank = common.init_distributed()
    common.set_precision(exp_conf)
    torch.manual_seed(exp_conf["seed"])
    np.random.seed(exp_conf["seed"])

    output_dir = common.setup_run(yaml_pth, exp_conf, rank)

    data_conf = exp_conf["data"]
    if rank == 0:
        print("Loading the data ...")
    lidar = build_lidar(data_conf)

    # one trajectory per node: waypoint .npy files when given a data_dir
    # with a waypoint_subdir, else synthetic paths
    if "data_dir" in data_conf and "waypoint_subdir" in data_conf:
        import glob

        paths = sorted(
            glob.glob(
                os.path.join(
                    data_conf["data_dir"], data_conf["waypoint_subdir"],
                    "*.npy",
                )
            )
        )
        waypoint_sets = [np.load(p) for p in paths]
    else:
        waypoint_sets = synthetic_waypoints(
            lidar.img, data_conf["num_nodes"],
            seed=data_conf.get("floorplan_seed", 0),
        )
    N = len(waypoint_sets)

    train_subsets = [
        OnlineTrajectoryLidarDataset(
            lidar, wp, data_conf["spline_res"],
            data_conf["num_scans_in_window"],
            round_density=data_conf.get("round_density", True),
        )
        for wp in waypoint_sets
    ]
    if rank == 0:
        for i in range(N):
            hd = (
                torch.sum(train_subsets[i].scans[:, 2] == 1.0)
                / train_subsets[i].scans.shape[0]
            ).item()
            print(f"Node {i} train set size: {len(train_subsets[i])} "
                  f"(hd ratio {hd:.4f})")

    val_set = RandomPoseLidarDataset(
        lidar, data_conf["num_validation_scans"],
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
        from .dist_dense_ex import _train_solo_all

        _train_solo_all(
            N, base_model, base_loss, train_subsets, val_set, device,
            solo_confs, exp_conf, output_dir,
        )
        # solo training consumed sliding windows; rebuild the datasets
        train_subsets = [
            OnlineTrajectoryLidarDataset(
                lidar, wp, data_conf["spline_res"],
                data_conf["num_scans_in_window"],
                round_density=data_conf.get("round_density", True),
            )
            for wp in waypoint_sets
        ]

    cent_conf = exp_conf.get("centralized_training",
                             {"train_centralized": False})
    if cent_conf.get("train_centralized", False) and rank == 0:
        from .dist_dense_ex import _train_centralized

        # pooled = every node's FULL trajectory (the reference's
        # centralized/online_density.ipynb cell 4 trains on all data)
        _train_centralized(
            base_model, base_loss, train_subsets, val_set, device,
            cent_conf, exp_conf, output_dir,
        )

    for prob_key, prob_conf in conf_dict["problem_configs"].items():
        opt_conf = prob_conf["optimizer_config"]
        # fresh sliding windows per problem run
        subsets = [
            OnlineTrajectoryLidarDataset(
                lidar, wp, data_conf["spline_res"],
                data_conf["num_scans_in_window"],
                round_density=data_conf.get("round_density", True),
            )
            for wp in waypoint_sets
        ]
        prob = DistOnlineDensityProblem(
            base_model, base_loss, subsets, val_set, device, prob_conf
        )
        common.maybe_attach_stacked(prob, exp_conf, opt_conf)
        dopt = build_optimizer(prob, device, opt_conf)
        if rank == 0:
            print("-" * 55)
            print("Running problem: " + prob_conf["problem_name"])
        common.run_problem(prob, dopt, prob_conf, exp_conf, output_dir)


if __name__ == "__main__":
    yaml_pth = sys.argv[1]
    if not os.path.exists(yaml_pth):
        raise NameError("YAML configuration file does not exist, exiting!")
    experiment(yaml_pth)
