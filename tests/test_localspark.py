"""Unit tests for the bundled localspark runtime (sparkflow_amd/localspark):
pyspark-API semantics, executor process model, and the Spark 2.4 on-disk
pipeline layout it writes."""
import json
import os

import numpy as np
import pytest

from sparkflow_amd import localspark

localspark.install(quiet=True)

from pyspark.ml import Pipeline, PipelineModel  # noqa: E402
from pyspark.ml.evaluation import MulticlassClassificationEvaluator  # noqa: E402
from pyspark.ml.feature import (  # noqa: E402
    Normalizer,
    OneHotEncoder,
    StopWordsRemover,
    VectorAssembler,
)
from pyspark.ml.linalg import Vectors  # noqa: E402
from pyspark.sql import Row, SparkSession  # noqa: E402


@pytest.fixture(scope="module")
def spark():
    s = (SparkSession.builder.master("local[2]").appName("ls-tests")
         .config("spark.driver.host", "127.0.0.1").getOrCreate())
    yield s
    s.stop()


def test_row_semantics():
    r = Row(b=2, a=1)  # Spark 2.x sorts kwarg fields
    assert r.__fields__ == ["a", "b"] and tuple(r) == (1, 2)
    assert r["a"] == 1 and r.b == 2
    assert r.asDict() == {"a": 1, "b": 2}
    import pickle

    r2 = pickle.loads(pickle.dumps(r))
    assert r2.asDict() == r.asDict()


def test_rdd_partitioning_and_actions(spark):
    df = spark.createDataFrame([(i, i * i) for i in range(10)], ["a", "b"])
    rdd = df.rdd
    assert rdd.getNumPartitions() == 2
    assert rdd.coalesce(1).getNumPartitions() == 1
    assert sorted(r["a"] for r in rdd.collect()) == list(range(10))
    assert rdd.map(lambda r: r["b"]).filter(lambda v: v > 10).count() == 6
    idx_seen = rdd.mapPartitionsWithIndex(
        lambda i, it: [(i, len(list(it)))]).collect()
    assert sorted(idx_seen) == [(0, 5), (1, 5)]
    rp = rdd.repartition(4)
    assert rp.getNumPartitions() == 4 and rp.count() == 10


def test_localproc_executors_are_processes():
    from sparkflow_amd.localspark.context import SparkConf, SparkContext
    from sparkflow_amd.localspark.sql import SparkSession as LS

    prev = SparkContext._active_spark_context
    try:
        s = LS(SparkContext("localproc[2]", "pids", SparkConf()))
        df = s.createDataFrame([(i,) for i in range(8)], ["v"])
        pids = set(df.rdd.mapPartitions(
            lambda it: [os.getpid()]).collect())
        assert os.getpid() not in pids  # tasks ran in child processes
        assert len(pids) >= 1
    finally:
        SparkContext._active_spark_context = prev


def test_vector_assembler_and_encoder(spark):
    df = spark.createDataFrame(
        [(1, 2.5, Vectors.dense([1.0, 2.0])),
         (0, -1.0, Vectors.sparse(2, [1], [3.0]))],
        ["label", "scalar", "vec"])
    out = VectorAssembler(inputCols=["scalar", "vec"],
                          outputCol="f").transform(df).collect()
    np.testing.assert_allclose(np.asarray(out[0]["f"]), [2.5, 1.0, 2.0])
    np.testing.assert_allclose(np.asarray(out[1]["f"]), [-1.0, 0.0, 3.0])

    oh = OneHotEncoder(inputCol="label", outputCol="oh",
                       dropLast=False).transform(df).collect()
    np.testing.assert_allclose(np.asarray(oh[0]["oh"]), [0.0, 1.0])
    # dropLast=True (Spark default) drops the final category
    oh2 = OneHotEncoder(inputCol="label",
                        outputCol="oh").transform(df).collect()
    assert len(np.asarray(oh2[0]["oh"])) == 1

    nm = Normalizer(inputCol="vec", outputCol="n",
                    p=2.0).transform(df).collect()
    np.testing.assert_allclose(np.linalg.norm(np.asarray(nm[0]["n"])), 1.0,
                               atol=1e-6)


def test_evaluator_matches_sklearn(spark):
    rng = np.random.default_rng(0)
    y = rng.integers(0, 3, 200)
    p = np.where(rng.random(200) < 0.7, y, rng.integers(0, 3, 200))
    df = spark.createDataFrame(
        [(float(a), float(b)) for a, b in zip(y, p)],
        ["label", "prediction"])
    ev_acc = MulticlassClassificationEvaluator(metricName="accuracy")
    ev_f1 = MulticlassClassificationEvaluator(metricName="f1")
    from sklearn.metrics import accuracy_score, f1_score

    assert abs(ev_acc.evaluate(df) - accuracy_score(y, p)) < 1e-9
    assert abs(ev_f1.evaluate(df)
               - f1_score(y, p, average="weighted")) < 1e-9


def test_pipeline_on_disk_layout(spark, tmp_path):
    """The written directory matches Spark's PipelineSharedReadWrite
    layout: metadata/part-00000 JSON (+_SUCCESS) with java class names
    and stageUids, per-stage dirs named <idx>_<uid> (reference
    pipeline_util.py smuggles python stages through exactly this
    format)."""
    sw = StopWordsRemover(inputCol="in", outputCol="out",
                          stopWords=["a", "b"])
    p = Pipeline(stages=[sw])
    path = str(tmp_path / "p")
    p.write().overwrite().save(path)

    meta = json.loads(open(os.path.join(path, "metadata",
                                        "part-00000")).readline())
    assert meta["class"] == "org.apache.spark.ml.Pipeline"
    assert meta["paramMap"]["stageUids"] == [sw.uid]
    assert os.path.exists(os.path.join(path, "metadata", "_SUCCESS"))
    stage_dir = os.path.join(path, "stages", "0_%s" % sw.uid)
    smeta = json.loads(open(os.path.join(stage_dir, "metadata",
                                         "part-00000")).readline())
    assert smeta["class"] == "org.apache.spark.ml.feature.StopWordsRemover"
    assert smeta["paramMap"]["stopWords"] == ["a", "b"]

    loaded = Pipeline.load(path)
    st = loaded.getStages()[0]
    assert isinstance(st, StopWordsRemover)
    assert st.getStopWords() == ["a", "b"] and st.uid == sw.uid


def test_stopwords_carrier_roundtrip_bytes(tmp_path):
    """A python object smuggled through the carrier (the reference's
    dill+zlib+GUID stopwords encoding) survives save/load byte-exactly."""
    from sparkflow_amd.spark.pipeline_util import (
        PY_OBJ_GUID,
        decode_py_obj,
        encode_py_obj,
    )

    obj = {"weights": [1.5, 2.5], "name": "stage"}
    words = encode_py_obj(obj)
    assert words[-1] == PY_OBJ_GUID
    sw = StopWordsRemover(inputCol="x", outputCol="y")
    sw.setStopWords(words)
    path = str(tmp_path / "carrier")
    from sparkflow_amd.localspark.ml_util import load_stage, save_stage

    save_stage(sw, path)
    back = load_stage(path)
    assert back.getStopWords()[-1] == PY_OBJ_GUID
    assert decode_py_obj(back.getStopWords()) == obj


def test_spark_read_csv(spark, tmp_path):
    p = tmp_path / "d.csv"
    p.write_text("1,0.5,a\n2,1.5,b\n")
    df = spark.read.option("inferSchema", "true").csv(str(p))
    assert df.columns == ["_c0", "_c1", "_c2"]
    r = df.collect()
    assert r[0]["_c0"] == 1 and r[0]["_c1"] == 0.5 and r[1]["_c2"] == "b"


def test_nested_pipeline_unwrap(spark, tmp_path):
    """unwrap recurses into sub-pipelines (reference
    pipeline_util.py:62-64)."""
    from sparkflow_amd.spark.pipeline_util import (
        PysparkPipelineWrapper,
        encode_py_obj,
    )

    inner_payload = {"kind": "python-stage"}
    sw = StopWordsRemover(inputCol="a", outputCol="b")
    sw.setStopWords(encode_py_obj(inner_payload))
    inner = PipelineModel([sw])
    outer = PipelineModel([inner])
    path = str(tmp_path / "nested")
    outer.write().overwrite().save(path)
    loaded = PysparkPipelineWrapper.unwrap(PipelineModel.load(path))
    assert isinstance(loaded.stages[0], PipelineModel)
    assert loaded.stages[0].stages[0] == inner_payload
