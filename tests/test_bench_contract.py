"""bench.py driver-contract test: flags accepted, one JSON line with the
required schema, whole-job value semantics."""

import json
import subprocess
import sys

REQUIRED = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup",
    "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
    "dtype", "data", "config",
}


def _run(args):
    import os

    env = {**os.environ, "NDTA_BENCH_SUSTAIN_S": "0.5"}
    out = subprocess.run(
        [sys.executable, "bench.py", *args],
        capture_output=True, text=True, timeout=420, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    return json.loads(line)


def test_bench_default_contract():
    d = _run(["--steps", "4", "--warmup", "1",
              "--samples-per-node", "256"])
    assert REQUIRED <= set(d)
    assert d["metric"] == "comm_rounds_per_sec"
    assert d["n_gpus"] == 1
    assert d["steps"] == 4 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "strong"
    assert d["data"] == "synthetic"
    assert d["dtype"] == "fp64"
    assert d["value"] > 0
    # internal consistency: ms_per_step == 1e3 / rounds_per_sec
    assert abs(d["ms_per_step"] * d["value"] - 1e3) < 1e-6 * 1e3
    cfg = d["config"]
    assert cfg["nodes"] == 8 and cfg["alg"] == "dinno"
    assert "consensus_err_max" in cfg and "val_acc_max" in cfg
    assert "parallelism" in cfg
    # sustained-window evidence keys
    assert d["timed_region_s"] > 0
    assert d["sustained_rounds_per_sec"] > 0
    assert d["sustained_s"] > 0 and d["sustained_steps"] >= 1
    assert "per_step_ms_p50" in d and "per_step_ms_std" in d


def test_bench_dsgd_config2():
    d = _run(["--alg", "dsgd", "--nodes", "4", "--steps", "3",
              "--warmup", "1", "--samples-per-node", "256"])
    assert d["config"]["alg"] == "dsgd"
    assert d["config"]["nodes"] == 4
