#!/usr/bin/env python3
"""One-command multi-process cluster demo (parity with the reference's
docker-compose Spark master + worker cluster, docker-compose.yml:1-22,
re-homed for the single-node MI355X deployment model).

Stands up a ``localproc[N]`` session — N spawned EXECUTOR PROCESSES, the
process shape of a real Spark worker set — and runs the full SparkFlow
story end to end:

  1. ``SparkAsyncDL.fit`` on a 2-class Gaussian DataFrame.  With
     SPARKFLOW_FIT_TRANSPORT=dist (default here) each executor becomes a
     torch.distributed rank: rank 0 hosts the parameter server + a local
     worker, the other ranks pull/push over their DistTransport pair
     links (RCCL over xGMI when world <= #GPUs; gloo wires otherwise).
  2. ``model.transform`` (batched per-partition inference).
  3. Pipeline save -> load -> unwrap round trip through the
     StopWordsRemover carrier format.

Run:  python examples/cluster_demo.py [--executors 2] [--tcp]
"""
import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--executors", type=int, default=2)
    ap.add_argument("--rows", type=int, default=400)
    ap.add_argument("--iters", type=int, default=10)
    ap.add_argument("--tcp", action="store_true",
                    help="use the TCP PS fallback instead of the "
                    "torch.distributed star")
    args = ap.parse_args()
    os.environ.setdefault("SPARKFLOW_FIT_TRANSPORT",
                          "tcp" if args.tcp else "dist")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("GLOO_SOCKET_IFNAME", "lo")

    import numpy as np

    from sparkflow_amd import localspark

    localspark.install()
    from pyspark.ml import Pipeline, PipelineModel
    from pyspark.ml.linalg import Vectors
    from pyspark.sql import SparkSession

    from sparkflow_amd.graph_utils import build_graph
    from sparkflow_amd.models import gaussian_mlp
    from sparkflow_amd.spark import SparkAsyncDL
    from sparkflow_amd.spark.pipeline_util import PysparkPipelineWrapper
    from sparkflow_amd.parallel.transport import find_free_port

    spark = (SparkSession.builder
             .master(f"localproc[{args.executors}]")
             .appName("sparkflow-amd-cluster-demo")
             .config("spark.driver.host", "127.0.0.1")
             .getOrCreate())
    print(f"[cluster-demo] driver pid={os.getpid()} "
          f"executors={args.executors} (spawned processes) "
          f"fit transport={os.environ['SPARKFLOW_FIT_TRANSPORT']}")

    rng = np.random.default_rng(12345)
    rows = [(i % 2, Vectors.dense(
        rng.normal(-2.0 if i % 2 == 0 else 2.0, 0.7, 10)))
        for i in range(args.rows)]
    df = spark.createDataFrame(rows, ["label", "features"])

    est = SparkAsyncDL(
        inputCol="features", tensorflowGraph=build_graph(gaussian_mlp),
        tfInput="x:0", tfLabel="y:0", tfOutput="out:0", tfOptimizer="adam",
        tfLearningRate=.01, iters=args.iters, predictionCol="predicted",
        labelCol="label", partitions=args.executors, miniBatchSize=-1,
        port=find_free_port())
    pipeline = Pipeline(stages=[est])
    model = pipeline.fit(df)

    preds = model.transform(df).collect()
    errors = sum(1 for r in preds
                 if round(min(max(float(r["predicted"]), 0.0), 1.0))
                 != r["label"])
    print(f"[cluster-demo] fit+transform: {len(preds)} rows, "
          f"{errors} errors")

    path = os.path.join(tempfile.mkdtemp(prefix="sparkflow_demo_"), "pipe")
    model.write().overwrite().save(path)
    loaded = PysparkPipelineWrapper.unwrap(PipelineModel.load(path))
    reloaded = loaded.transform(df).take(5)
    print(f"[cluster-demo] pipeline save/load round trip OK "
          f"({path}); sample prediction: "
          f"{float(reloaded[0]['predicted']):.3f}")
    assert errors < len(preds)
    print("[cluster-demo] PASS")
    spark.stop()


if __name__ == "__main__":
    main()
