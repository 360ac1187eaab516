"""The driver's bench contract: `python bench.py` prints ONE JSON line
with the documented fields (BASELINE.json headline metric).  Runs the
real script on CPU with tiny flags so the contract can't rot.
"""
import json
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run(args, timeout=600):
    r = subprocess.run([sys.executable] + args, capture_output=True,
                       text=True, timeout=timeout, cwd=ROOT)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    return json.loads(r.stdout.strip().splitlines()[-1])


def test_bench_json_contract():
    d = _run(["bench.py", "--steps", "4", "--warmup", "1"])
    assert REQUIRED <= set(d)
    assert d["metric"] == "samples_per_sec"
    assert d["unit"] == "samples/s"
    assert d["n_gpus"] == 1 and d["steps"] == 4 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["dtype"] == "fp32"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    assert cfg["model"] == "mnist_mlp_784_256_256_10"
    assert cfg["global_batch"] == 1200  # 4 workers x 300 (BASELINE config)
    assert "parallelism" in cfg


def test_time_to_acc_json_contract():
    d = _run(["benchmarks/time_to_acc.py", "--max-rounds", "2",
              "--round-steps", "2", "--target", "0.0",
              "--train-rows", "400"])
    assert d["metric"] == "time_to_97pct_acc"
    assert d["unit"] == "s"
    assert d["higher_is_better"] is False
    assert d["value"] == d["value"]  # reached target -> not NaN
    assert "final_accuracy" in d
