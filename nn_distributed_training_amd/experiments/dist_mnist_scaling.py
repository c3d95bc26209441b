"""MNIST scaling-study driver.

Capability parity with the reference's
``experiments/dist_mnist_scaling.py``: sweep either node count N at a
fixed Fiedler value (algebraic connectivity) via Fiedler-targeted disk
graphs, or Fiedler value at fixed N — recording per-trial graphs and
metric streams. Node data uses the label-sorted chunk split so arbitrary
N is supported (reference :122-129).

This is BASELINE config 5's path: 2→32 logical nodes packed across the
available ranks (1/2/4/8 GPUs) with comm rounds/sec + convergence
reported per trial.
"""

from __future__ import annotations

import os
import sys
import time

import torch
import yaml

from ..data.mnist import load_mnist, split_train_set
from ..models.mnist_conv import MNISTConvNet
from ..optimizers import build_optimizer
from ..problems.dist_mnist_problem import DistMNISTProblem
from ..utils import graph_generation
from . import common


def experiment(yaml_pth: str):
    with open(yaml_pth) as f:
        conf_dict = yaml.safe_load(f)
    exp_conf = conf_dict["experiment"]

    rank, world, local_rank = common.init_distributed()
    common.set_precision(exp_conf)
    torch.manual_seed(exp_conf.get("seed", 0))

    output_dir = common.setup_run(yaml_pth, exp_conf, rank)

    train_set, val_set = load_mnist(
        exp_conf.get("data_dir", "./data"),
        source=exp_conf.get("data_source", "synthetic"),
        train_samples=exp_conf.get("train_samples", 60000),
        val_samples=exp_conf.get("val_samples", 10000),
        seed=exp_conf.get("seed", 0),
    )

    model_conf = exp_conf["model"]
    base_model = MNISTConvNet(
        model_conf["num_filters"],
        model_conf["kernel_size"],
        model_conf["linear_width"],
    )
    base_loss = common.make_loss(exp_conf["loss"])
    device = common.select_device(exp_conf, local_rank)

    sweep = exp_conf["sweep"]
    if sweep["type"] == "nodes":
        trials = [
            (int(n), float(sweep["fiedler"])) for n in sweep["num_nodes"]
        ]
    elif sweep["type"] == "fiedler":
        trials = [
            (int(sweep["num_nodes"]), float(f))
            for f in sweep["fiedler_values"]
        ]
    else:
        raise NameError("Unknown sweep type.")

    summary = {}
    for t, (N, fied) in enumerate(trials):
        if rank == 0:
            print(f"--- trial {t}: N={N}, fiedler={fied} ---")
        graph = graph_generation.disk_with_fied(N, fied)
        if exp_conf["writeout"] and rank == 0:
            common.save_graph(
                graph, os.path.join(output_dir, f"{t}.gpickle")
            )

        train_subsets = split_train_set(train_set, N, "hetero_sorted")

        for prob_key, prob_conf in conf_dict["problem_configs"].items():
            opt_conf = prob_conf["optimizer_config"]
            prob = DistMNISTProblem(
                graph, base_model, base_loss, train_subsets, val_set,
                device, prob_conf,
            )
            common.maybe_attach_stacked(prob, exp_conf, opt_conf)
            dopt = build_optimizer(prob, device, opt_conf)
            t0 = time.perf_counter()
            common.run_problem(prob, dopt, prob_conf, exp_conf,
                               output_dir)
            elapsed = time.perf_counter() - t0
            rounds = opt_conf["outer_iterations"]
            if rank == 0:
                rps = rounds / elapsed
                print(
                    f"trial {t} {prob_conf['problem_name']}: "
                    f"{rps:.1f} rounds/s"
                )
                summary[(t, prob_conf["problem_name"])] = {
                    "N": N, "fiedler": fied, "rounds_per_sec": rps,
                }
            if exp_conf["writeout"] and rank == 0:
                os.rename(
                    os.path.join(
                        output_dir,
                        prob_conf["problem_name"] + "_results.pt",
                    ),
                    os.path.join(
                        output_dir,
                        f"{t}_{prob_conf['problem_name']}_results.pt",
                    ),
                )
    if exp_conf["writeout"] and rank == 0:
        torch.save(
            summary, os.path.join(output_dir, "scaling_summary.pt")
        )


if __name__ == "__main__":
    yaml_pth = sys.argv[1]
    if not os.path.exists(yaml_pth):
        raise NameError("YAML configuration file does not exist, exiting!")
    experiment(yaml_pth)
