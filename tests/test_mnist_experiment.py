"""End-to-end driver test: tiny DiNNO/DSGD/DSGT MNIST run on CPU
(BASELINE config 1: plumbing parity, 4-node random graph)."""

import os

import torch
import yaml

from nn_distributed_training_amd.experiments import dist_mnist_ex

TINY_CONF = {
    "experiment": {
        "name": "tiny_mnist",
        "data_dir": "./data",
        "output_metadir": None,  # filled by fixture
        "use_cuda": False,
        "writeout": True,
        "data_split_type": "hetero",
        "data_source": "synthetic",
        "train_samples": 800,
        "val_samples": 200,
        "seed": 0,
        "loss": "NLL",
        "precision": "fp64",
        "engine": "torch",
        "graph": {"num_nodes": 4, "type": "random", "p": 0.6,
                  "gen_attempts": 50},
        "model": {"num_filters": 3, "kernel_size": 5, "linear_width": 64},
        "individual_training": {
            "train_solo": False,
            "optimizer": "adam",
            "lr": 0.005,
            "epochs": 1,
            "train_batch_size": 64,
            "val_batch_size": 64,
            "verbose": False,
        },
    },
    "problem_configs": {},
}


def _prob_conf(name, opt_conf):
    return {
        "problem_name": name,
        "train_batch_size": 32,
        "val_batch_size": 64,
        "verbose_evals": False,
        "metrics": [
            "forward_pass_count",
            "validation_loss",
            "consensus_error",
            "top1_accuracy",
            "current_epoch",
        ],
        "metrics_config": {"evaluate_frequency": 5},
        "optimizer_config": opt_conf,
    }


def _write_conf(tmp_path, problems):
    conf = dict(TINY_CONF)
    conf["experiment"] = dict(TINY_CONF["experiment"])
    conf["experiment"]["output_metadir"] = str(tmp_path / "out")
    conf["problem_configs"] = problems
    pth = tmp_path / "conf.yaml"
    with open(pth, "w") as f:
        yaml.safe_dump(conf, f)
    return str(pth)


def test_all_three_optimizers_end_to_end(tmp_path):
    problems = {
        "p1": _prob_conf(
            "dinno",
            {
                "alg_name": "dinno",
                "rho_init": 0.5,
                "rho_scaling": 1.0003,
                "outer_iterations": 6,
                "primal_iterations": 2,
                "primal_optimizer": "adam",
                "persistant_primal_opt": False,
                "primal_lr_start": 0.005,
                "primal_lr_finish": 0.0005,
                "lr_decay_type": "log",
                "profile": False,
            },
        ),
        "p2": _prob_conf(
            "dsgd",
            {
                "alg_name": "dsgd",
                "outer_iterations": 6,
                "alpha0": 0.005,
                "mu": 0.001,
                "profile": False,
            },
        ),
        "p3": _prob_conf(
            "dsgt",
            {
                "alg_name": "dsgt",
                "outer_iterations": 6,
                "alpha": 0.005,
                "init_grads": True,
                "profile": False,
            },
        ),
    }
    pth = _write_conf(tmp_path, problems)
    dist_mnist_ex.experiment(pth)

    # checkpoint layout parity: run dir with yaml snapshot, graph pickle,
    # one results file per problem
    out_meta = tmp_path / "out"
    runs = list(out_meta.iterdir())
    assert len(runs) == 1
    files = {p.name for p in runs[0].iterdir()}
    assert "graph.gpickle" in files
    assert any(f.endswith(".yaml") for f in files)
    for name in ("dinno", "dsgd", "dsgt"):
        assert f"{name}_results.pt" in files
        res = torch.load(
            os.path.join(runs[0], f"{name}_results.pt"),
            weights_only=False,
        )
        # 6 iterations, eval every 5 -> evals at k=0, 5(=last)
        assert len(res["top1_accuracy"]) == 2
        assert len(res["consensus_error"]) == 2
        accs = res["top1_accuracy"][-1]
        assert accs.shape == (4,)
        assert (accs >= 0).all() and (accs <= 1).all()


def test_dinno_learns_synthetic_mnist(tmp_path):
    """Convergence sanity: 30 DiNNO rounds on the easy synthetic task
    should beat random-chance accuracy by a wide margin on every node."""
    problems = {
        "p1": _prob_conf(
            "dinno",
            {
                "alg_name": "dinno",
                "rho_init": 0.5,
                "rho_scaling": 1.0003,
                "outer_iterations": 60,
                "primal_iterations": 2,
                "primal_optimizer": "adam",
                "persistant_primal_opt": False,
                "primal_lr_start": 0.005,
                "primal_lr_finish": 0.001,
                "lr_decay_type": "log",
                "profile": False,
            },
        )
    }
    pth = _write_conf(tmp_path, problems)
    dist_mnist_ex.experiment(pth)
    runs = list((tmp_path / "out").iterdir())
    res = torch.load(
        os.path.join(runs[0], "dinno_results.pt"), weights_only=False
    )
    final_acc = res["top1_accuracy"][-1]
    assert final_acc.amin().item() > 0.5  # hetero split, 60 rounds
    # consensus error decays from its peak (the k=0 eval is exactly 0:
    # all replicas start identical, so compare against the peak instead)
    peak = max(ce[1].amax() for ce in res["consensus_error"][:-1])
    last = res["consensus_error"][-1][1].amax()
    assert last < peak


def test_scaling_driver(tmp_path):
    """dist_mnist_scaling sweep (tiny): per-trial graphs + results +
    rounds/sec summary (reference experiments/dist_mnist_scaling.py)."""
    import yaml as _yaml

    from nn_distributed_training_amd.experiments import dist_mnist_scaling

    conf = {
        "experiment": {
            "name": "tiny_scaling",
            "output_metadir": str(tmp_path / "out"),
            "use_cuda": False,
            "writeout": True,
            "data_source": "synthetic",
            "train_samples": 400,
            "val_samples": 100,
            "precision": "fp64",
            "engine": "torch",
            "seed": 0,
            "loss": "NLL",
            "sweep": {"type": "nodes", "num_nodes": [3, 5],
                      "fiedler": 1.0},
            "model": {"num_filters": 3, "kernel_size": 5,
                      "linear_width": 64},
        },
        "problem_configs": {
            "p1": {
                "problem_name": "dinno",
                "train_batch_size": 16,
                "val_batch_size": 64,
                "verbose_evals": False,
                "metrics": ["top1_accuracy", "consensus_error"],
                "metrics_config": {"evaluate_frequency": 100},
                "optimizer_config": {
                    "alg_name": "dinno", "rho_init": 0.5,
                    "rho_scaling": 1.0003, "outer_iterations": 3,
                    "primal_iterations": 1,
                    "primal_optimizer": "adam",
                    "persistant_primal_opt": False,
                    "primal_lr_start": 0.005,
                    "primal_lr_finish": 0.001,
                    "lr_decay_type": "log", "profile": False,
                },
            }
        },
    }
    pth = tmp_path / "scaling.yaml"
    with open(pth, "w") as f:
        _yaml.safe_dump(conf, f)
    dist_mnist_scaling.experiment(str(pth))

    runs = list((tmp_path / "out").iterdir())
    files = {p.name for p in runs[0].iterdir()}
    assert "0.gpickle" in files and "1.gpickle" in files
    assert "0_dinno_results.pt" in files and "1_dinno_results.pt" in files
    summary = torch.load(
        runs[0] / "scaling_summary.pt", weights_only=False
    )
    assert {(0, "dinno"), (1, "dinno")} <= set(summary)
    assert summary[(0, "dinno")]["N"] == 3
    assert summary[(1, "dinno")]["N"] == 5
    assert summary[(0, "dinno")]["rounds_per_sec"] > 0


def test_fixed_seed_metric_streams_are_identical(tmp_path):
    """fixed seed -> identical metric streams across two
    independent runs of the same driver (the reference's only notion of
    reproducibility is `torch.manual_seed` at the driver top)."""
    import glob
    import pickle

    problems = {
        "p1": _prob_conf(
            "dsgd",
            {
                "alg_name": "dsgd", "outer_iterations": 6,
                "alpha0": 0.005, "mu": 0.001, "profile": False,
            },
        )
    }

    def run(sub):
        base = tmp_path / sub
        os.makedirs(base)
        pth = _write_conf(base, problems)
        dist_mnist_ex.experiment(pth)
        run_dir = glob.glob(str(base / "out" / "*_tiny_mnist"))[0]
        res = torch.load(
            os.path.join(run_dir, "dsgd_results.pt"), weights_only=False
        )
        with open(os.path.join(run_dir, "graph.gpickle"), "rb") as f:
            graph = pickle.load(f)
        return res, graph

    res_a, g_a = run("a")
    res_b, g_b = run("b")

    assert sorted(g_a.edges()) == sorted(g_b.edges())
    assert res_a.keys() == res_b.keys()
    for key in res_a:
        for ea, eb in zip(res_a[key], res_b[key]):
            ta = torch.as_tensor(ea[1] if isinstance(ea, tuple) else ea,
                                 dtype=torch.float64)
            tb = torch.as_tensor(eb[1] if isinstance(eb, tuple) else eb,
                                 dtype=torch.float64)
            torch.testing.assert_close(ta, tb, rtol=0, atol=0)


def test_centralized_baseline(tmp_path):
    """`centralized_training:` trains one pooled-data model and writes
    centralized_results.pt with per-epoch curves (the reference's
    upper-bound line, centralized notebooks / mnist_four.ipynb)."""
    import copy
    import glob

    conf = {"experiment": dict(copy.deepcopy(TINY_CONF)["experiment"]),
            "problem_configs": {}}
    conf["experiment"]["output_metadir"] = str(tmp_path / "out")
    conf["experiment"]["individual_training"]["train_solo"] = True
    conf["experiment"]["centralized_training"] = {
        "train_centralized": True,
        "optimizer": "adam",
        "lr": 0.005,
        "epochs": 2,
        "train_batch_size": 64,
        "val_batch_size": 64,
        "verbose": False,
    }
    pth = tmp_path / "conf.yaml"
    with open(pth, "w") as f:
        yaml.safe_dump(conf, f)
    dist_mnist_ex.experiment(str(pth))

    run_dir = glob.glob(str(tmp_path / "out" / "*"))[0]
    cent = torch.load(os.path.join(run_dir, "centralized_results.pt"),
                      weights_only=False)
    assert len(cent["validation_accuracy"]) == 2
    assert len(cent["validation_loss"]) == 2
    # pooled training on the synthetic task should beat chance
    assert cent["validation_accuracy"][-1] > 0.2
    solo = torch.load(os.path.join(run_dir, "solo_results.pt"),
                      weights_only=False)
    assert len(solo) == 4

    # the comparability figure renders from these artifacts
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "plot_results",
        os.path.join(os.path.dirname(__file__), "..",
                     "visualization", "plot_results.py"),
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    figs = tmp_path / "figs"
    mod.plot_run_dir(run_dir, str(figs))
    # no decentralized runs -> the four-figure may or may not draw;
    # the call itself must succeed with centralized+solo present


def test_idx_files_source_real_mnist_format(tmp_path):
    """`data_source: idx_files` loads REAL MNIST from the official IDX
    file format via the first-party parser (torchvision is absent in
    this image; VERDICT r1 item 5 — comparability to the published
    anchors whenever the files are present)."""
    import gzip
    import struct

    import numpy as np

    from nn_distributed_training_amd.data.mnist import (
        MNIST_MEAN,
        MNIST_STD,
        load_mnist,
    )

    rng = np.random.default_rng(0)

    def write_idx_images(path, n, gz=False):
        imgs = rng.integers(0, 256, size=(n, 28, 28), dtype=np.uint8)
        payload = struct.pack(">IIII", 0x803, n, 28, 28) + imgs.tobytes()
        op = gzip.open if gz else open
        with op(path + (".gz" if gz else ""), "wb") as f:
            f.write(payload)
        return imgs

    def write_idx_labels(path, n, gz=False):
        labels = rng.integers(0, 10, size=(n,), dtype=np.uint8)
        payload = struct.pack(">II", 0x801, n) + labels.tobytes()
        op = gzip.open if gz else open
        with op(path + (".gz" if gz else ""), "wb") as f:
            f.write(payload)
        return labels

    d = tmp_path / "mnist"
    d.mkdir()
    # train plain, t10k gzipped — both layouts must parse
    tr_imgs = write_idx_images(str(d / "train-images-idx3-ubyte"), 32)
    tr_lbls = write_idx_labels(str(d / "train-labels-idx1-ubyte"), 32)
    write_idx_images(str(d / "t10k-images-idx3-ubyte"), 8, gz=True)
    write_idx_labels(str(d / "t10k-labels-idx1-ubyte"), 8, gz=True)

    train, val = load_mnist(str(d), source="idx_files")
    assert len(train) == 32 and len(val) == 8
    x, y = train[0]
    assert x.shape == (1, 28, 28)
    # normalization matches the reference's torchvision pipeline
    expect = (tr_imgs[0].astype(np.float64) / 255.0 - MNIST_MEAN) \
        / MNIST_STD
    np.testing.assert_allclose(x[0].numpy(), expect, rtol=1e-5)
    assert int(y) == int(tr_lbls[0])
    assert train.targets.shape == (32,)
