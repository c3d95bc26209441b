"""MNIST experiment driver: YAML in -> results directory out.

Capability parity with the reference's ``experiments/dist_mnist_ex.py``
(same YAML schema, data splits, solo baseline, per-problem loop, profiler
hook, checkpoint layout). Differences: runs multi-rank under torchrun
(one process per GPU, logical nodes packed onto ranks), data defaults to
the offline synthetic MNIST (no network in this environment), and
precision is a config knob (default fp64 = reference parity).
"""

from __future__ import annotations

import copy
import os
import sys

import torch
import yaml

from ..data.mnist import load_mnist, split_train_set
from ..models.mnist_conv import MNISTConvNet
from ..optimizers import build_optimizer
from ..problems.dist_mnist_problem import DistMNISTProblem
from ..utils import graph_generation
from . import common


def train_solo(model, loss, train_set, val_set, device, conf):
    """Individual (no-communication) baseline for one node.

    Parity with reference experiments/dist_mnist_ex.py:22-62.
    """
    trainloader = torch.utils.data.DataLoader(
        train_set, conf["train_batch_size"], shuffle=True
    )
    valloader = torch.utils.data.DataLoader(
        val_set, conf["val_batch_size"], shuffle=True
    )
    model = model.to(device)
    opts = {
        "adam": torch.optim.Adam,
        "sgd": torch.optim.SGD,
        "adamw": torch.optim.AdamW,
    }
    if conf["optimizer"] not in opts:
        raise NameError("Unknown individual optimizer.")
    opt = opts[conf["optimizer"]](model.parameters(), lr=conf["lr"])

    for _ in range(conf["epochs"]):
        for x, y in trainloader:
            opt.zero_grad()
            out = model.forward(x.to(device))
            loss(out, y.to(device)).backward()
            opt.step()

    with torch.no_grad():
        val_loss, correct = 0.0, 0
        for x, y in valloader:
            x, y = x.to(device), y.to(device)
            out = model.forward(x)
            val_loss += loss(out, y).item()
            pred = out.argmax(dim=1, keepdim=True)
            correct += pred.eq(y.view_as(pred)).sum().item()
        nval = len(valloader.dataset)
    return {
        "validation_loss": val_loss / nval,
        "validation_accuracy": correct / nval,
    }


def train_centralized(model, loss, train_subsets, val_set, device,
                      conf):
    """Pooled-data upper-bound baseline: ONE model trained on the
    union of every node's shard, validation tracked per epoch.

    This reproduces the reference's centralized notebooks (the 0.985
    top-1 'centralized' line every figure in
    visualization/mnist_four.ipynb cells 1-5 is anchored to) — the
    reference has no script for it, only notebook cells; here it is a
    config-driven part of the experiment (`centralized_training:`).
    """
    pooled = torch.utils.data.ConcatDataset(list(train_subsets))
    trainloader = torch.utils.data.DataLoader(
        pooled, conf["train_batch_size"], shuffle=True
    )
    valloader = torch.utils.data.DataLoader(
        val_set, conf["val_batch_size"]
    )
    model = model.to(device)
    opts = {
        "adam": torch.optim.Adam,
        "sgd": torch.optim.SGD,
        "adamw": torch.optim.AdamW,
    }
    if conf["optimizer"] not in opts:
        raise NameError("Unknown centralized optimizer.")
    opt = opts[conf["optimizer"]](model.parameters(), lr=conf["lr"])

    curves = {"validation_loss": [], "validation_accuracy": [],
              "epoch": []}
    for ep in range(conf["epochs"]):
        for x, y in trainloader:
            opt.zero_grad()
            out = model.forward(x.to(device))
            loss(out, y.to(device)).backward()
            opt.step()
        with torch.no_grad():
            val_loss, correct = 0.0, 0
            for x, y in valloader:
                x, y = x.to(device), y.to(device)
                out = model.forward(x)
                val_loss += loss(out, y).item()
                pred = out.argmax(dim=1, keepdim=True)
                correct += pred.eq(y.view_as(pred)).sum().item()
            nval = len(valloader.dataset)
        curves["epoch"].append(ep)
        curves["validation_loss"].append(val_loss / nval)
        curves["validation_accuracy"].append(correct / nval)
        if conf.get("verbose", False):
            print(
                "Centralized epoch {} - Validation Acc = {:.4f}".format(
                    ep, curves["validation_accuracy"][-1]
                )
            )
    return curves


def experiment(yaml_pth: str):
    with open(yaml_pth) as f:
        conf_dict = yaml.safe_load(f)
    exp_conf = conf_dict["experiment"]

    rank, world, local_rank = common.init_distributed()
    common.set_precision(exp_conf)
    if "seed" in exp_conf:
        torch.manual_seed(exp_conf["seed"])

    output_dir = common.setup_run(yaml_pth, exp_conf, rank)

    # communication graph — identical on every rank (seeded generation or
    # rank-0 broadcast via the YAML-seeded generator)
    graph_conf = dict(exp_conf["graph"])
    graph_conf.setdefault("seed", exp_conf.get("seed", 0))
    N, graph = graph_generation.generate_from_conf(graph_conf)
    if exp_conf["writeout"] and rank == 0:
        common.save_graph(graph, os.path.join(output_dir, "graph.gpickle"))

    # data
    train_set, val_set = load_mnist(
        exp_conf.get("data_dir", "./data"),
        source=exp_conf.get("data_source", "synthetic"),
        train_samples=exp_conf.get("train_samples", 60000),
        val_samples=exp_conf.get("val_samples", 10000),
        seed=exp_conf.get("seed", 0),
    )
    train_subsets = split_train_set(train_set, N, exp_conf["data_split_type"])

    model_conf = exp_conf["model"]
    base_model = MNISTConvNet(
        model_conf["num_filters"],
        model_conf["kernel_size"],
        model_conf["linear_width"],
    )
    base_loss = common.make_loss(exp_conf["loss"])
    device = common.select_device(exp_conf, local_rank)
    if rank == 0:
        print(f"Device is set to {device} (world size {world})")

    # individual-training baseline (single-rank concern; rank 0 runs it)
    solo_confs = exp_conf["individual_training"]
    if solo_confs["train_solo"] and rank == 0:
        solo_results = {}
        print("Performing individual training ...")
        for i in range(N):
            solo_results[i] = train_solo(
                copy.deepcopy(base_model), base_loss, train_subsets[i],
                val_set, device, solo_confs,
            )
            if solo_confs["verbose"]:
                print(
                    "Node {} - Validation Acc = {:.4f}".format(
                        i, solo_results[i]["validation_accuracy"]
                    )
                )
        if exp_conf["writeout"]:
            torch.save(
                solo_results, os.path.join(output_dir, "solo_results.pt")
            )

    # pooled-data centralized baseline (upper-bound curve; reference
    # produced it in centralized/*.ipynb notebooks only)
    cent_conf = exp_conf.get("centralized_training",
                             {"train_centralized": False})
    if cent_conf.get("train_centralized", False) and rank == 0:
        print("Performing centralized (pooled-data) training ...")
        cent_results = train_centralized(
            copy.deepcopy(base_model), base_loss, train_subsets,
            val_set, device, cent_conf,
        )
        if exp_conf["writeout"]:
            torch.save(
                cent_results,
                os.path.join(output_dir, "centralized_results.pt"),
            )

    # per-(problem, optimizer) loop
    for prob_key, prob_conf in conf_dict["problem_configs"].items():
        opt_conf = prob_conf["optimizer_config"]
        prob = DistMNISTProblem(
            graph, base_model, base_loss, train_subsets, val_set, device,
            prob_conf,
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
