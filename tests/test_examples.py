"""Examples stay runnable.

The reference ships runnable examples exercised by its CI
(tests run via .travis.yml); ours EXECUTE end to end as well:
``local_hogwild.py`` needs no Spark, and the three Spark examples run
on the real pyspark when installed or on the bundled localspark
runtime otherwise (with a synthetic MNIST-shaped DataFrame when
``examples/mnist_train.csv`` is absent — see examples/_data.py).
"""
import subprocess
import sys
from pathlib import Path

import pytest

EXAMPLES = Path(__file__).resolve().parent.parent / "examples"


def test_local_hogwild_example_runs():
    r = subprocess.run(
        [sys.executable, str(EXAMPLES / "local_hogwild.py")],
        capture_output=True, text=True, timeout=240)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert "trained 4 weight tensors" in r.stdout


@pytest.mark.parametrize("name,needle", [
    ("simple_dnn.py", "DenseVector"),
    ("cnn_example.py", "DenseVector"),
    ("autoencoder_example.py", "_c0"),
])
@pytest.mark.timeout(600)
def test_spark_examples_run(name, needle):
    r = subprocess.run(
        [sys.executable, str(EXAMPLES / name), "--rows", "200",
         "--iters", "2"],
        capture_output=True, text=True, timeout=540)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-3000:]
    assert needle in r.stdout, r.stdout[-2000:]


def test_cluster_demo_runs():
    r = subprocess.run(
        [sys.executable, str(EXAMPLES / "cluster_demo.py"),
         "--executors", "2", "--rows", "120", "--iters", "4"],
        capture_output=True, text=True, timeout=540)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-3000:]
    assert "[cluster-demo] PASS" in r.stdout
