#!/usr/bin/env python3
"""Flagship benchmark: DiNNO MNIST on an 8-node communication graph.

Measures communication rounds/sec (the BASELINE.json headline metric) for
the MI355X-native stacked engine: 8 logical graph nodes packed across
--gpus ranks (strong scaling — the job is fixed, more GPUs split it),
neighbor exchange over RCCL P2P, per-node compute as fused CDNA4 kernels.
One round = snapshot + neighbor exchange + dual ascent + primal_iterations
x (fwd + bwd + penalty-fused Adam) on every node — the full algorithm,
nothing skipped; metric evaluation runs AFTER the timed window (the
reference evaluates every 20 rounds outside the round path too).

Launch (multi-GPU, by the driver):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Data: synthetic MNIST (no network in this environment), hetero split over
nodes, random-init MNISTConvNet(3,5,64) — the reference paper config.
dtype fp64 by default = the reference's torch.DoubleTensor default
(experiments/dist_mnist_ex.py:19); --dtype fp32 selects the fp32 path.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import networkx as nx
import torch
import torch.distributed as dist


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=50)
    p.add_argument("--nodes", type=int, default=8,
                   help="logical graph nodes (the 8-node headline config)")
    p.add_argument("--graph", default="ring",
                   choices=["ring", "complete", "random"])
    p.add_argument("--dtype", default="fp64", choices=["fp64", "fp32"])
    p.add_argument("--batch", type=int, default=0,
                   help="0 = workload default (64 mnist / 20000 density)")
    p.add_argument("--engine", default="auto",
                   choices=["auto", "torch", "hip"])
    p.add_argument("--samples-per-node", type=int, default=2048)
    p.add_argument(
        "--workload", default="mnist", choices=["mnist", "density"],
        help="mnist = DiNNO MNIST 8-node (headline, BASELINE cfg 3/5); "
             "density = DSGT online implicit density (BASELINE cfg 4)",
    )
    p.add_argument(
        "--alg", default=None, choices=["dinno", "dsgd", "dsgt"],
        help="mnist workload algorithm (default dinno; dsgd with "
             "--nodes 4 = BASELINE cfg 2)",
    )
    return p.parse_args()


def main():
    args = parse_args()
    torch.set_default_dtype(
        torch.float64 if args.dtype == "fp64" else torch.float32
    )

    # distributed init (torchrun provides RANK/WORLD_SIZE)
    rank, world = 0, 1
    if "RANK" in os.environ and "WORLD_SIZE" in os.environ:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if backend == "nccl":
            # arm the NCCL watchdog: a desynchronized/hung P2P tears
            # the job down with a rank-attributed error, not a hang
            os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
        from datetime import timedelta

        dist.init_process_group(backend=backend,
                                timeout=timedelta(seconds=300))
        rank = dist.get_rank()
        world = dist.get_world_size()
        # per-rank hang diagnostic: if the process is still alive but
        # stuck 600s from now, dump all thread stacks to stderr (renew
        # is cheap; cancelled after the timed region)
        import faulthandler

        faulthandler.dump_traceback_later(600, repeat=True)
    local_rank = int(os.environ.get("LOCAL_RANK", 0))
    if torch.cuda.is_available():
        # one rank per GPU normally; the mod keeps oversubscribed
        # smoke runs (2 ranks on a 1-GPU box) working like comm.py
        device = torch.device(
            "cuda", local_rank % torch.cuda.device_count()
        )
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")

    torch.manual_seed(0)
    use_hip = args.engine == "hip" or (
        args.engine == "auto" and device.type == "cuda"
    )
    if args.workload == "mnist":
        pr, step_fn, wl_cfg = _build_mnist(args, device, use_hip)
    else:
        pr, step_fn, wl_cfg = _build_density(args, device, use_hip)

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize()
        if world > 1:
            dist.barrier()
            if device.type == "cuda":
                torch.cuda.synchronize()

    for k in range(args.warmup):
        step_fn(k)
    sync()
    t0 = time.perf_counter()
    for k in range(args.warmup, args.warmup + args.steps):
        step_fn(k)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines the job)
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64, device=device
                         if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    # ---- sustained window (supplementary evidence; VERDICT r1 weak
    # #4/#7): the headline value above times EXACTLY --steps rounds per
    # the bench contract, but with driver-chosen small K that region can
    # be milliseconds — too thin for GPU-busy sampling and variance.  So
    # we keep stepping for >= NDTA_BENCH_SUSTAIN_S wall seconds (default
    # 12 s, ~3 SMI samples), with per-step host timestamps, and report a
    # corroborating sustained rounds/s + variance + the section-timing
    # breakdown.  Every rank derives the same extra-step count from the
    # all-reduced `elapsed`, so multi-rank collectives stay in lockstep.
    sustain_s = float(os.environ.get("NDTA_BENCH_SUSTAIN_S", "12"))
    per_step = elapsed / args.steps
    extra = int(min(max(sustain_s / max(per_step, 1e-9), 1), 200_000))
    try:
        from nn_distributed_training_amd.ops.stacked import (
            set_timing,
            timing_dict,
            timing_reset,
        )

        timing_reset()
        set_timing(True)
    except ImportError:
        timing_dict = None
    stamps = [time.perf_counter()]
    k0 = args.warmup + args.steps
    for k in range(k0, k0 + extra):
        # schedules (DiNNO primal lr) are sized to warmup+steps rounds;
        # sustained steps hold the final-round hyperparameters
        step_fn(min(k, k0 - 1))
        stamps.append(time.perf_counter())
    sync()
    sustained_s = time.perf_counter() - stamps[0]
    if timing_dict is not None:
        set_timing(False)
    if world > 1:
        t = torch.tensor([sustained_s], dtype=torch.float64,
                         device=device
                         if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        sustained_s = t.item()
    import statistics

    diffs = [
        (b - a) * 1e3 for a, b in zip(stamps[:-1], stamps[1:])
    ]
    diffs.sort()
    sustain_stats = {
        "sustained_rounds_per_sec": extra / sustained_s,
        "sustained_steps": extra,
        "sustained_s": round(sustained_s, 3),
        "per_step_ms_p50": round(diffs[len(diffs) // 2], 4),
        "per_step_ms_p90": round(diffs[int(len(diffs) * 0.9)], 4),
        "per_step_ms_std": round(
            statistics.pstdev(diffs) if len(diffs) > 1 else 0.0, 4
        ),
    }

    if world > 1:
        import faulthandler

        faulthandler.cancel_dump_traceback_later()

    # post-timing quality metrics (val acc/loss + consensus error)
    pr.evaluate_metrics(at_end=True)
    cons = pr.metrics["consensus_error"][-1][1]
    quality = {"consensus_err_max": float(cons.amax())}
    if "top1_accuracy" in pr.metrics:
        accs = pr.metrics["top1_accuracy"][-1]
        quality["val_acc_min"] = round(float(accs.amin()), 4)
        quality["val_acc_max"] = round(float(accs.amax()), 4)
    if "validation_loss" in pr.metrics and pr.metrics["validation_loss"]:
        vl = pr.metrics["validation_loss"][-1]
        quality["val_loss_max"] = round(float(vl.amax()), 4)

    if rank == 0:
        rounds_per_sec = args.steps / elapsed
        out = {
            "metric": "comm_rounds_per_sec",
            "value": rounds_per_sec,
            "unit": "rounds/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "timed_region_s": round(elapsed, 6),
            **sustain_stats,
            "config": {
                **wl_cfg,
                "parallelism": f"graph-decentralized dp, "
                               f"{wl_cfg['nodes']} nodes on "
                               f"{world} rank(s)",
                "engine": "hip-stacked" if use_hip else "torch-golden",
                **quality,
            },
        }
        if timing_dict is not None:
            td = timing_dict()
            if td:
                out["timing_breakdown_ms"] = td
        print(json.dumps(out), flush=True)

    if dist.is_initialized():
        dist.destroy_process_group()


def _make_graph(args, N):
    if args.graph == "ring":
        return nx.cycle_graph(N)
    if args.graph == "complete":
        return nx.complete_graph(N)
    return nx.erdos_renyi_graph(N, 0.5, seed=1)


def _build_mnist(args, device, use_hip):
    from nn_distributed_training_amd.data.mnist import (
        SyntheticMNIST,
        split_train_set,
    )
    from nn_distributed_training_amd.models import MNISTConvNet
    from nn_distributed_training_amd.optimizers import build_optimizer
    from nn_distributed_training_amd.problems.dist_mnist_problem import (
        DistMNISTProblem,
    )

    N = args.nodes
    B = args.batch or 64
    alg = args.alg or "dinno"
    graph = _make_graph(args, N)
    train = SyntheticMNIST(args.samples_per_node * N, seed=0)
    val = SyntheticMNIST(1024, seed=1)
    subsets = split_train_set(train, N, "hetero" if N <= 10 else
                              "hetero_sorted")
    base_model = MNISTConvNet(3, 5, 64)

    oits = args.warmup + args.steps
    opt_confs = {
        "dinno": {
            "alg_name": "dinno", "rho_init": 0.5,
            "rho_scaling": 1.0003, "outer_iterations": oits,
            "primal_iterations": 2, "primal_optimizer": "adam",
            "persistant_primal_opt": False, "primal_lr_start": 0.005,
            "primal_lr_finish": 0.0005, "lr_decay_type": "log",
            "profile": False,
        },
        "dsgd": {
            "alg_name": "dsgd", "outer_iterations": oits,
            "alpha0": 0.005, "mu": 0.001, "profile": False,
        },
        "dsgt": {
            "alg_name": "dsgt", "outer_iterations": oits,
            "alpha": 0.005, "init_grads": True, "profile": False,
        },
    }
    opt_conf = opt_confs[alg]
    prob_conf = {
        "problem_name": f"bench_{alg}",
        "train_batch_size": B,
        "val_batch_size": 256,
        "data_seed": 0,
        "verbose_evals": False,
        "metrics": ["consensus_error", "validation_loss",
                    "top1_accuracy"],
        "metrics_config": {"evaluate_frequency": 10**9},
        "optimizer_config": opt_conf,
    }
    pr = DistMNISTProblem(
        graph, base_model, torch.nn.NLLLoss(), subsets, val, device,
        prob_conf,
    )
    if use_hip:
        from nn_distributed_training_amd.ops.stacked import (
            DiNNOStackedDriver,
            DSGDStackedDriver,
            DSGTStackedDriver,
            StackedEngine,
        )

        pr.stacked = StackedEngine(pr)
        dopt = build_optimizer(pr, device, opt_conf)
        drv_cls = {
            "dinno": DiNNOStackedDriver,
            "dsgd": DSGDStackedDriver,
            "dsgt": DSGTStackedDriver,
        }[alg]
        driver = drv_cls(dopt, pr)
        driver.prepare()
        step_fn = driver.step_round
    else:
        dopt = build_optimizer(pr, device, opt_conf)
        if alg == "dinno":
            step_fn = _golden_round_fn(dopt, pr)
        elif alg == "dsgt":
            state = {"ready": False}

            def step_fn(k, _d=dopt):
                if not state["ready"]:
                    if _d.conf["init_grads"]:
                        for i in pr.local_nodes:
                            g = _d._local_grad_vector(i)
                            _d.y[i] = g.clone()
                            _d.g[i] = g.clone()
                    state["ready"] = True
                _golden_dsgt_round(_d, pr)
        else:
            state = {"alph": dopt.alph0}

            def step_fn(k, _d=dopt):
                _golden_dsgd_round(_d, pr, state)
    cfg = {
        "model": "MNISTConvNet(3,5,64) n=28440",
        "alg": alg,
        "nodes": N,
        "graph": args.graph,
        "global_batch": B * N,
        "primal_iterations": 2 if alg == "dinno" else None,
    }
    return pr, step_fn, cfg


def _golden_dsgd_round(opt, pr, state):
    import torch as _t

    from nn_distributed_training_amd.optimizers.neighbors import (
        gather_neighbor_stacks,
    )
    from nn_distributed_training_amd.utils import graph_generation

    pr.update_graph()
    W = graph_generation.get_metropolis(pr.graph).to(pr.device)
    state["alph"] = state["alph"] * (1 - opt.mu * state["alph"])
    ths = pr.local_params_stack()
    neigh = gather_neighbor_stacks(pr, ths)
    for li, i in enumerate(pr.local_nodes):
        mixed = W[i, i] * ths[li]
        for row, j in zip(neigh[i], pr.graph.neighbors(i)):
            mixed = mixed + W[i, j] * row
        _t.nn.utils.vector_to_parameters(
            mixed, pr.models[i].parameters()
        )
    for i in pr.local_nodes:
        bloss = pr.local_batch_loss(i)
        bloss.backward()
        with _t.no_grad():
            for p in pr.models[i].parameters():
                p.add_(p.grad, alpha=-state["alph"])
                p.grad.zero_()


def _build_density(args, device, use_hip):
    """BASELINE config 4: DSGT online implicit density (FourierNet,
    synthetic lidar sliding windows, dynamic disk graph)."""
    import numpy as np

    from nn_distributed_training_amd.data.floorplan import (
        synthetic_floorplan,
        synthetic_waypoints,
    )
    from nn_distributed_training_amd.data.lidar import (
        Lidar2D,
        OnlineTrajectoryLidarDataset,
        RandomPoseLidarDataset,
    )
    from nn_distributed_training_amd.optimizers.dsgt import DSGT
    from nn_distributed_training_amd.problems.dist_online_dense_problem \
        import DistOnlineDensityProblem

    np.random.seed(0)
    N = args.nodes
    B = args.batch or 20000  # paper DSGT batch (dist_online_dense_PAPER)
    img = synthetic_floorplan(nx=256, ny=256, num_walls=10,
                              border_width=24, seed=0)
    lidar = Lidar2D(img, 20, 0.2, 25, 1.0, 50, 3)
    waypoints = synthetic_waypoints(img, N, seed=0)
    subsets = [
        OnlineTrajectoryLidarDataset(lidar, wp, 30, 100)
        for wp in waypoints
    ]
    val = RandomPoseLidarDataset(lidar, 100)

    from nn_distributed_training_amd.models import FourierNet

    base_model = FourierNet([2, 256, 64, 64, 64, 1], scale=0.05)
    opt_conf = {
        "alg_name": "dsgt",
        "alpha": 0.001,
        "outer_iterations": args.warmup + args.steps,
        "init_grads": True,
        "profile": False,
    }
    prob_conf = {
        "problem_name": "bench_dsgt_density",
        "train_batch_size": B,
        "val_batch_size": 10000,
        "comm_radius": 120.0,
        "dynamic_graph": True,
        "save_models": False,
        "data_seed": 0,
        "verbose_evals": False,
        "metrics": ["consensus_error", "validation_loss"],
        "metrics_config": {"evaluate_frequency": 10**9,
                           "tloss_decay": 0.2,
                           "mesh_only_at_end": True},
        "optimizer_config": opt_conf,
    }
    pr = DistOnlineDensityProblem(
        base_model, torch.nn.BCELoss(), subsets, val, device, prob_conf
    )
    if use_hip:
        from nn_distributed_training_amd.ops.stacked import (
            DSGTStackedDriver,
            StackedEngine,
        )

        pr.stacked = StackedEngine(pr)
        driver = DSGTStackedDriver(DSGT(pr, device, opt_conf), pr)
        driver.prepare()
        step_fn = driver.step_round
    else:
        dopt = DSGT(pr, device, opt_conf)
        # golden per-round closure (init_grads bootstrap first)
        state = {"ready": False}

        def step_fn(k, _dopt=dopt):
            if not state["ready"]:
                if _dopt.conf["init_grads"]:
                    for i in pr.local_nodes:
                        g = _dopt._local_grad_vector(i)
                        _dopt.y[i] = g.clone()
                        _dopt.g[i] = g.clone()
                state["ready"] = True
            _golden_dsgt_round(_dopt, pr)

    cfg = {
        "model": "FourierNet[2,256,64,64,64,1] n=25601",
        "alg": "dsgt",
        "nodes": N,
        "graph": "dynamic-disk(r=120)",
        "global_batch": B * N,
        "window": 100,
    }
    return pr, step_fn, cfg


def _golden_dsgt_round(opt, pr):
    import torch as _t

    from nn_distributed_training_amd.optimizers.neighbors import (
        gather_neighbor_stacks,
    )
    from nn_distributed_training_amd.utils import graph_generation

    pr.update_graph()
    W = graph_generation.get_metropolis(pr.graph).to(pr.device)
    ths = pr.local_params_stack()
    ys = _t.stack([opt.y[i] for i in pr.local_nodes])
    bundle = _t.cat([ths, ys], dim=1)
    neigh = gather_neighbor_stacks(pr, bundle)
    n = pr.n
    y_new = {}
    for li, i in enumerate(pr.local_nodes):
        p_mix = W[i, i] * (ths[li] - opt.alpha * opt.y[i])
        y_mix = W[i, i] * opt.y[i]
        for row, j in zip(neigh[i], pr.graph.neighbors(i)):
            p_mix = p_mix + W[i, j] * (row[:n] - opt.alpha * row[n:])
            y_mix = y_mix + W[i, j] * row[n:]
        _t.nn.utils.vector_to_parameters(
            p_mix, pr.models[i].parameters()
        )
        y_new[i] = y_mix
    for i in pr.local_nodes:
        g_next = opt._local_grad_vector(i)
        opt.y[i] = y_new[i] + g_next - opt.g[i]
        opt.g[i] = g_next


def _golden_round_fn(opt, pr):
    """Single DiNNO round on the golden engine (CPU fallback path)."""
    from nn_distributed_training_amd.optimizers.neighbors import (
        gather_neighbor_stacks,
    )

    def step(k):
        ths = pr.local_params_stack().clone()
        opt.rho *= opt.rho_scaling
        pr.update_graph()
        neigh = gather_neighbor_stacks(pr, ths)
        for li, i in enumerate(pr.local_nodes):
            thj = neigh[i]
            if thj.shape[0] == 0:
                continue
            opt.duals[i] += opt.rho * torch.sum(ths[li] - thj, dim=0)
            th_reg = 0.5 * (thj + ths[li])
            opt.primal_update(i, th_reg, k)

    return step


if __name__ == "__main__":
    main()
