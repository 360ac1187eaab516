"""localspark — a JVM-free, pure-Python runtime exposing the subset of
the pyspark API that SparkFlow programs use.

Why this exists: the reference framework (lifeomic/sparkflow) is a
Spark-ML library, and its entire test story is end-to-end ``fit`` /
``transform`` / pipeline save-load on a Spark ``local[2]`` session
(reference tests/dl_runner.py:26-31).  The MI355X deployment target is a
single 8-GPU node, where the JVM+py4j stack adds nothing to the compute
path — every FLOP runs in our HIP kernels and every byte of training
traffic moves over RCCL.  localspark supplies the Spark *programming
model* (DataFrame/RDD partitions, ML Pipeline, Params, the
StopWordsRemover-carrier persistence format) natively:

* ``local[N]``   — N concurrent task threads per action, the exact
  concurrency shape of the reference's test harness (2 partitions -> 2
  concurrent worker closures against one PS).
* ``localproc[N]`` — N spawned *processes*, one per partition: the shape
  of a real Spark executor set, used to run one rank per GPU so
  ``SparkAsyncDL.fit`` drives the RCCL star (see
  sparkflow_amd/spark/async_dl.py + parallel/dist_engine.py).

``install()`` registers these modules under the ``pyspark`` names ONLY
when the real pyspark is not importable, so the same user code (and the
ported reference test-suite) runs unchanged in either environment.  The
on-disk pipeline format written here is the reference's own: Spark-style
``metadata/part-00000`` JSON + ``stages/<i>_<uid>/`` dirs, with python
stages dill+zlib-encoded into a StopWordsRemover's stopwords
(reference pipeline_util.py:109-127).
"""
from __future__ import annotations

import sys


def using_localspark() -> bool:
    """True when ``pyspark`` in sys.modules is this package's shim."""
    mod = sys.modules.get("pyspark")
    return mod is not None and getattr(mod, "__is_localspark__", False)


def install(force: bool = False, quiet: bool = False) -> bool:
    """Register localspark under the ``pyspark`` module names.

    No-op (returns False) if a real pyspark is importable and ``force``
    is not set.  Returns True when the shim is (already) active."""
    if using_localspark():
        return True
    if not force:
        try:
            import pyspark  # noqa: F401  (the real one)

            return False
        except ImportError:
            pass

    import types

    from sparkflow_amd.localspark import context as _context
    from sparkflow_amd.localspark import sql as _sql
    from sparkflow_amd.localspark import rdd as _rdd
    from sparkflow_amd.localspark import ml_base as _ml_base
    from sparkflow_amd.localspark import ml_evaluation as _ml_evaluation
    from sparkflow_amd.localspark import ml_feature as _ml_feature
    from sparkflow_amd.localspark import ml_linalg as _ml_linalg
    from sparkflow_amd.localspark import ml_param as _ml_param
    from sparkflow_amd.localspark import ml_pipeline as _ml_pipeline
    from sparkflow_amd.localspark import ml_util as _ml_util
    from sparkflow_amd.localspark import ml_wrapper as _ml_wrapper
    from sparkflow_amd.localspark import sql_functions as _sql_functions

    top = types.ModuleType("pyspark")
    top.__is_localspark__ = True
    top.__path__ = []  # mark as package
    top.__version__ = "2.4.3+localspark"
    top.SparkContext = _context.SparkContext
    top.SparkConf = _context.SparkConf
    top.TaskContext = _context.TaskContext
    top.keyword_only = _ml_param.keyword_only
    top.context = _context
    top.sql = _sql
    top.rdd = _rdd

    ml = types.ModuleType("pyspark.ml")
    ml.__path__ = []
    ml.Pipeline = _ml_pipeline.Pipeline
    ml.PipelineModel = _ml_pipeline.PipelineModel
    ml.Estimator = _ml_base.Estimator
    ml.Transformer = _ml_base.Transformer
    ml.Model = _ml_base.Model
    ml.base = _ml_base
    ml.param = _ml_param
    ml.pipeline = _ml_pipeline
    ml.feature = _ml_feature
    ml.linalg = _ml_linalg
    ml.util = _ml_util
    ml.wrapper = _ml_wrapper
    ml.evaluation = _ml_evaluation
    top.ml = ml

    _sql.functions = _sql_functions

    mods = {
        "pyspark": top,
        "pyspark.context": _context,
        "pyspark.conf": _context,
        "pyspark.rdd": _rdd,
        "pyspark.sql": _sql,
        "pyspark.sql.session": _sql,
        "pyspark.sql.dataframe": _sql,
        "pyspark.sql.types": _sql,
        "pyspark.sql.functions": _sql_functions,
        "pyspark.ml": ml,
        "pyspark.ml.base": _ml_base,
        "pyspark.ml.param": _ml_param,
        "pyspark.ml.param.shared": _ml_param,
        "pyspark.ml.pipeline": _ml_pipeline,
        "pyspark.ml.feature": _ml_feature,
        "pyspark.ml.linalg": _ml_linalg,
        "pyspark.ml.util": _ml_util,
        "pyspark.ml.wrapper": _ml_wrapper,
        "pyspark.ml.evaluation": _ml_evaluation,
    }
    sys.modules.update(mods)
    if not quiet:
        print("[sparkflow_amd] pyspark not found: using the built-in "
              "localspark runtime (JVM-free pyspark-compatible API)")
    return True
