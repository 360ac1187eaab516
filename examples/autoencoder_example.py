"""Unsupervised autoencoder on Spark (port of reference
examples/autoencoder_example.py — ``tfLabel=None``)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from pyspark.ml.feature import VectorAssembler
from pyspark.sql import SparkSession

from sparkflow_amd.graph_utils import build_graph
from sparkflow_amd.models import mnist_autoencoder
from sparkflow_amd.spark import SparkAsyncDL


def main():
    spark = SparkSession.builder.appName("sparkflow-amd ae").getOrCreate()
    df = (spark.read.option("inferSchema", "true")
          .csv("examples/mnist_train.csv"))
    va = VectorAssembler(inputCols=df.columns[1:785], outputCol="features")

    spark_model = SparkAsyncDL(
        inputCol="features",
        tensorflowGraph=build_graph(mnist_autoencoder),
        tfInput="x:0",
        tfLabel=None,           # unsupervised: loss vs the input batch
        tfOutput="out:0",
        tfLearningRate=.001,
        iters=20,
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
