"""Shared driver machinery: output layout, losses, devices, engine attach.

The output/checkpoint layout matches the reference
(experiments/dist_mnist_ex.py:74-95): per run a directory
``<output_metadir>/<YYYY-MM-DD_HH-MM>_<name>/`` containing the YAML
snapshot ``<ts>.yaml``, ``graph.gpickle``, per-problem
``<problem_name>_results.pt`` and optional ``solo_results.pt`` /
``<problem_name>_models.pt``.
"""

from __future__ import annotations

import os
import pickle
from datetime import datetime
from shutil import copyfile

import torch

from ..parallel.comm import init_from_env


def setup_run(yaml_pth: str, exp_conf: dict, rank: int) -> str:
    """Create the run output dir (rank 0 only) and snapshot the config."""
    import torch.distributed as dist

    output_metadir = exp_conf["output_metadir"]
    time_now = datetime.now().strftime("%Y-%m-%d_%H-%M")
    if dist.is_available() and dist.is_initialized():
        # all ranks must agree on the run dir (checkpoint shards live
        # there); a rank clocking in across a minute boundary must not
        # invent its own timestamp
        box = [time_now]
        dist.broadcast_object_list(box, src=0)
        time_now = box[0]
    output_dir = os.path.join(output_metadir, time_now + "_" + exp_conf["name"])
    if exp_conf["writeout"] and rank == 0:
        os.makedirs(output_metadir, exist_ok=True)
        os.makedirs(output_dir, exist_ok=True)
        copyfile(yaml_pth, os.path.join(output_dir, time_now + ".yaml"))
    exp_conf["output_dir"] = output_dir
    return output_dir


def save_graph(graph, path):
    """gpickle-compatible graph snapshot (networkx >= 3 removed
    write_gpickle; plain pickle produces the same file contents)."""
    with open(path, "wb") as f:
        pickle.dump(graph, f, pickle.HIGHEST_PROTOCOL)


def select_device(exp_conf: dict, local_rank: int) -> torch.device:
    if torch.cuda.is_available() and exp_conf.get("use_cuda", True):
        dev = torch.device("cuda", local_rank % torch.cuda.device_count())
        torch.cuda.set_device(dev)
        return dev
    return torch.device("cpu")


def make_loss(name: str):
    if name == "NLL":
        return torch.nn.NLLLoss()
    if name == "BCE":
        return torch.nn.BCELoss()
    if name == "MSE":
        return torch.nn.MSELoss()
    if name == "L1":
        return torch.nn.L1Loss()
    raise NameError("Unknown loss function.")


def set_precision(exp_conf: dict):
    """Default dtype knob: fp64 (reference parity) | fp32 | bf16."""
    prec = exp_conf.get("precision", "fp64")
    torch.set_default_dtype(
        {"fp64": torch.float64, "fp32": torch.float32,
         "bf16": torch.bfloat16}[prec]
    )
    return prec


def init_distributed():
    """torchrun-aware init; no-op single rank otherwise."""
    return init_from_env()


def maybe_attach_stacked(problem, exp_conf, opt_conf):
    """Attach the HIP stacked engine when requested and available.

    engine: auto — use HIP kernels when running on a GPU (fails loudly if
            the extension is missing there: silent eager fallback on a GPU
            box would be a lie);
            torch — force the golden eager engine;
            hip — require the HIP engine (error anywhere it can't load).
    """
    engine = exp_conf.get("engine", "auto")
    if engine == "torch":
        return
    on_gpu = problem.device.type == "cuda"
    if engine == "auto" and not on_gpu:
        return
    from ..ops.stacked import StackedEngine

    problem.stacked = StackedEngine(problem)


def run_problem(problem, dopt, prob_conf, exp_conf, output_dir):
    """Optionally profile (parity with reference dist_mnist_ex.py:207-220),
    train, save metrics."""
    opt_conf = prob_conf["optimizer_config"]
    # checkpoints land in the run dir unless the config says otherwise
    opt_conf.setdefault("checkpoint_dir", output_dir)
    if getattr(dopt, "checkpoint_dir", None) is None:
        dopt.checkpoint_dir = output_dir
    if opt_conf.get("profile", False):
        with torch.profiler.profile(
            schedule=torch.profiler.schedule(
                wait=1, warmup=1, active=3, repeat=3
            ),
            on_trace_ready=torch.profiler.tensorboard_trace_handler(
                os.path.join(
                    output_dir, prob_conf["problem_name"] + "opt_profile"
                )
            ),
            record_shapes=True,
            with_stack=True,
        ) as prof:
            dopt.train(profiler=prof)
    else:
        dopt.train()

    if exp_conf["writeout"]:
        problem.save_metrics(output_dir)
