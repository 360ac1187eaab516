"""Examples stay runnable.

The reference ships runnable examples (examples/simple_dnn.py,
cnn_example.py, autoencoder_example.py) exercised by its CI; ours must
not rot either.  `local_hogwild.py` needs no Spark and runs end-to-end
here; the pyspark-based ones are syntax-checked (pyspark is not in this
image — see docs/STATUS.md).
"""
import py_compile
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


@pytest.mark.parametrize("name", [
    "simple_dnn.py", "cnn_example.py", "autoencoder_example.py"])
def test_spark_examples_compile(name):
    py_compile.compile(str(EXAMPLES / name), doraise=True)
