"""Spark-ML API integration tests — the port of reference
tests/dl_runner.py.  They run wherever pyspark is installed (a Spark
local[2] session = 2 concurrent worker partitions against a real PS
process); in the pyspark-less CI container they skip."""
import pytest

pyspark = pytest.importorskip("pyspark")

import random

import numpy as np
from pyspark.ml.feature import VectorAssembler
from pyspark.ml.linalg import Vectors
from pyspark.sql import SparkSession

from sparkflow_amd.graph_utils import build_adam_config, build_graph
from sparkflow_amd.models import gaussian_mlp, tabular_autoencoder, xor_mlp
from sparkflow_amd.spark import SparkAsyncDL
from sparkflow_amd.spark.pipeline_util import PysparkPipelineWrapper

random.seed(12345)


@pytest.fixture(scope="module")
def spark():
    session = (SparkSession.builder.master("local[2]")
               .appName("sparkflow-amd-tests")
               .config("spark.driver.host", "127.0.0.1")
               .getOrCreate())
    yield session
    session.stop()


def _gaussian_df(spark, n=400):
    rng = np.random.default_rng(12345)
    rows = []
    for i in range(n):
        label = i % 2
        mean = -2.0 if label == 0 else 2.0
        rows.append((label, Vectors.dense(rng.normal(mean, 0.7, 10))))
    return spark.createDataFrame(rows, ["label", "features"])


def _estimator(graph_json, port, **kw):
    defaults = dict(
        inputCol="features", tensorflowGraph=graph_json, tfInput="x:0",
        tfLabel="y:0", tfOutput="out:0", tfLearningRate=.01, iters=15,
        predictionCol="predicted", labelCol="label", verbose=0,
        partitions=2, miniBatchSize=-1, port=port,
    )
    defaults.update(kw)
    return SparkAsyncDL(**defaults)


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def test_fit_transform_gaussians(spark):
    df = _gaussian_df(spark)
    graph_json = build_graph(gaussian_mlp)
    model = _estimator(graph_json, _free_port()).fit(df)
    preds = model.transform(df).collect()
    assert len(preds) == 400
    errors = sum(1 for r in preds
                 if round(min(max(r["predicted"], 0.0), 1.0)) != r["label"])
    assert errors < len(preds)  # reference-style loose assertion


def test_adam_optimizer_options(spark):
    df = _gaussian_df(spark, 200)
    graph_json = build_graph(gaussian_mlp)
    opts = build_adam_config(learning_rate=0.01, beta1=0.85, beta2=0.998)
    model = _estimator(graph_json, _free_port(), iters=10,
                       optimizerOptions=opts).fit(df)
    assert model.getOrDefault(model.modelWeights)


def test_rmsprop(spark):
    df = _gaussian_df(spark, 200)
    graph_json = build_graph(gaussian_mlp)
    model = _estimator(graph_json, _free_port(), iters=10,
                       tfOptimizer="rmsprop").fit(df)
    assert model.getOrDefault(model.modelWeights)


def test_save_pipeline_roundtrip(spark, tmp_path):
    from pyspark.ml import Pipeline, PipelineModel

    df = _gaussian_df(spark, 200)
    graph_json = build_graph(gaussian_mlp)
    est = _estimator(graph_json, _free_port(), iters=5)
    pipeline = Pipeline(stages=[est]).fit(df)
    path = str(tmp_path / "pipe")
    pipeline.write().overwrite().save(path)
    loaded = PysparkPipelineWrapper.unwrap(PipelineModel.load(path))
    preds = loaded.transform(df).collect()
    assert len(preds) == 200


def test_autoencoder(spark):
    rng = np.random.default_rng(0)
    rows = [(Vectors.dense(rng.random(784)),) for _ in range(64)]
    df = spark.createDataFrame(rows, ["features"])
    graph_json = build_graph(tabular_autoencoder)
    est = SparkAsyncDL(inputCol="features", tensorflowGraph=graph_json,
                       tfInput="x:0", tfLabel=None, tfOutput="out:0",
                       iters=5, partitions=2, miniBatchSize=16,
                       miniStochasticIters=2, port=_free_port())
    model = est.fit(df)
    assert model.getOrDefault(model.modelWeights)
