"""MNIST MLP on Spark (port of reference examples/simple_dnn.py).

Requires: pyspark, an mnist_train.csv with label in column 0.
Each Spark executor should own one MI355X (set HIP_VISIBLE_DEVICES per
executor); the driver hosts the parameter server.
"""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from sparkflow_amd import localspark

localspark.install(quiet=True)  # no-op when a real pyspark is installed

from pyspark.ml.feature import OneHotEncoder, VectorAssembler
from pyspark.ml.pipeline import Pipeline, PipelineModel
from pyspark.sql import SparkSession

from examples._data import load_mnist_df
from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_mlp
from sparkflow_amd.spark import SparkAsyncDL
from sparkflow_amd.spark.pipeline_util import PysparkPipelineWrapper


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2000)
    ap.add_argument("--iters", type=int, default=None)
    args = ap.parse_args()

    spark = (SparkSession.builder.master("local[2]")
             .appName("sparkflow-amd simple_dnn")
             .config("spark.driver.host", "127.0.0.1").getOrCreate())
    df = load_mnist_df(spark, rows=args.rows)

    graph_json = build_graph(mnist_mlp)
    va = VectorAssembler(inputCols=df.columns[1:785], outputCol="features")
    encoded = OneHotEncoder(inputCol="_c0", outputCol="labels", dropLast=False)

    spark_model = SparkAsyncDL(
        inputCol="features",
        tensorflowGraph=graph_json,
        tfInput="x:0",
        tfLabel="_c0",          # class-index labels
        tfOutput="out:0",
        tfLearningRate=.001,
        iters=args.iters or 50,
        predictionCol="predicted",
        labelCol="_c0",
        miniBatchSize=300,
        verbose=1,
        partitions=4,
    )

    pipeline = Pipeline(stages=[va, encoded, spark_model]).fit(df)
    pipeline.write().overwrite().save("mnist_dnn_pipeline")

    loaded = PysparkPipelineWrapper.unwrap(
        PipelineModel.load("mnist_dnn_pipeline"))
    preds = loaded.transform(df)
    preds.select("predicted", "_c0").show(10)


if __name__ == "__main__":
    main()
