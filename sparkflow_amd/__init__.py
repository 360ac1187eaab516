"""sparkflow_amd — an MI355X-native asynchronous parameter-server training framework.

A from-scratch re-design of the capabilities of lifeomic/sparkflow for AMD
Instinct MI355X (gfx950, CDNA4) hardware:

* the driver-side Flask/HTTP parameter server of the reference
  (``sparkflow/HogwildSparkModel.py``) becomes a rank-0 **GPU parameter
  server** whose gradient-push / weight-pull loop runs as RCCL
  point-to-point send/recv over xGMI (``sparkflow_amd.parallel``);
* TensorFlow-1.x graph execution is replaced by PyTorch-ROCm modules whose
  hot ops are hand-written CDNA4 HIP kernels with MFMA/LDS tiling
  (``sparkflow_amd.ops``);
* the Spark-ML Pipeline API surface (``SparkAsyncDL`` / ``SparkAsyncDLModel``
  with the identical Param list, ``build_graph``, the dill/zlib
  StopWordsRemover pipeline on-disk format) is preserved
  (``sparkflow_amd.spark``, ``sparkflow_amd.graph_utils``).

Reference parity map (file:line refer to the lifeomic/sparkflow reference):
  sparkflow/graph_utils.py        -> sparkflow_amd/graph_utils.py
  sparkflow/ml_util.py            -> sparkflow_amd/ml_util.py
  sparkflow/RWLock.py             -> sparkflow_amd/parallel/rwlock.py
  sparkflow/HogwildSparkModel.py  -> sparkflow_amd/parallel/{ps,worker,hogwild}.py
  sparkflow/tensorflow_async.py   -> sparkflow_amd/spark/async_dl.py
  sparkflow/pipeline_util.py      -> sparkflow_amd/spark/pipeline_util.py
  sparkflow/tensorflow_model_loader.py -> sparkflow_amd/spark/model_loader.py
"""

__version__ = "0.1.0"

from sparkflow_amd.graph_utils import build_graph  # noqa: F401
