"""Unsupervised autoencoder on Spark (port of reference
examples/autoencoder_example.py — ``tfLabel=None``)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from sparkflow_amd import localspark

localspark.install(quiet=True)  # no-op when a real pyspark is installed

from pyspark.ml.feature import VectorAssembler
from pyspark.sql import SparkSession

from examples._data import load_mnist_df
from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_autoencoder
from sparkflow_amd.spark import SparkAsyncDL


def main():
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=2000)
    ap.add_argument("--iters", type=int, default=None)
    args = ap.parse_args()

    spark = (SparkSession.builder.master("local[2]")
             .appName("sparkflow-amd ae")
             .config("spark.driver.host", "127.0.0.1").getOrCreate())
    df = load_mnist_df(spark, rows=args.rows)
    va = VectorAssembler(inputCols=df.columns[1:785], outputCol="features")

    spark_model = SparkAsyncDL(
        inputCol="features",
        tensorflowGraph=build_graph(mnist_autoencoder),
        tfInput="x:0",
        tfLabel=None,           # unsupervised: loss vs the input batch
        tfOutput="out:0",
        tfLearningRate=.001,
        iters=args.iters or 20,
        predictionCol="predicted",
        partitions=4,
        miniBatchSize=256,
        miniStochasticIters=1,
        verbose=1,
    )
    model = spark_model.fit(va.transform(df))
    model.transform(va.transform(df)).show(2)


if __name__ == "__main__":
    main()
