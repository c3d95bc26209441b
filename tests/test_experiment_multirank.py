"""2-rank run of a full YAML experiment driver over gloo.

Covers the driver-level multi-rank path that the algorithm parity tests
bypass: torchrun-style env init, the rank-agreed run directory
(setup_run broadcasts rank 0's timestamp), rank-0-only artifact writes.
"""

import glob
import os

import torch.multiprocessing as mp
import yaml


def _worker(rank, td, port):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        RANK=str(rank),
        WORLD_SIZE="2",
        LOCAL_RANK=str(rank),
    )
    import torch.distributed as dist

    from nn_distributed_training_amd.experiments.dist_mnist_ex import (
        experiment,
    )

    try:
        experiment(os.path.join(td, "cfg.yaml"))
    finally:
        if dist.is_initialized():
            dist.destroy_process_group()


def test_mnist_driver_two_ranks(tmp_path):
    cfg = os.path.join(
        os.path.dirname(__file__), "..", "configs", "dist_mnist_anim.yaml"
    )
    conf = yaml.safe_load(open(cfg))
    conf["experiment"]["use_cuda"] = False
    conf["experiment"]["graph"]["num_nodes"] = 4
    conf["experiment"]["output_metadir"] = str(tmp_path)
    conf["experiment"]["data_dir"] = str(tmp_path)
    for p in conf["problem_configs"].values():
        p["optimizer_config"]["outer_iterations"] = 2
        p["train_batch_size"] = 16
        p["val_batch_size"] = 32
        p["metrics_config"]["evaluate_frequency"] = 2
        p["verbose_evals"] = False
    yaml.safe_dump(conf, open(tmp_path / "cfg.yaml", "w"))

    mp.start_processes(
        _worker, args=(str(tmp_path), 29781), nprocs=2, join=True,
        start_method="spawn",
    )

    runs = glob.glob(str(tmp_path / "*_dist_mnist_anim"))
    assert len(runs) == 1, "ranks must agree on ONE run directory"
    names = {os.path.basename(f) for f in glob.glob(runs[0] + "/*")}
    assert "dinno_results.pt" in names
    assert "graph.gpickle" in names
