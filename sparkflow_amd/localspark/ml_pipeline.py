"""pyspark.ml.pipeline subset: Pipeline / PipelineModel with Spark's
directory persistence (PipelineSharedReadWrite layout)."""
from __future__ import annotations

import os
from typing import List

from sparkflow_amd.localspark.ml_base import Estimator, Model, Transformer
from sparkflow_amd.localspark.ml_param import (
    Param,
    Params,
    TypeConverters,
    keyword_only,
)
from sparkflow_amd.localspark import ml_util as _mlu


class Pipeline(Estimator):
    stages = Param(Params._dummy(), "stages", "pipeline stages")

    @keyword_only
    def __init__(self, stages=None):
        super().__init__()
        self._set(stages=stages or [])

    def setStages(self, value: List):
        return self._set(stages=value)

    def getStages(self) -> List:
        return self.getOrDefault(self.stages)

    def _fit(self, dataset) -> "PipelineModel":
        stages = self.getStages()
        transformers = []
        for i, stage in enumerate(stages):
            if isinstance(stage, Estimator):
                model = stage.fit(dataset)
                transformers.append(model)
                if i < len(stages) - 1:
                    dataset = model.transform(dataset)
            elif isinstance(stage, Transformer):
                transformers.append(stage)
                if i < len(stages) - 1:
                    dataset = stage.transform(dataset)
            else:
                raise TypeError("stage %r is not Estimator/Transformer"
                                % (stage,))
        return PipelineModel(transformers)

    def write(self):
        return _mlu.MLWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def read(cls):
        return _mlu.MLReader(cls)

    @classmethod
    def load(cls, path: str) -> "Pipeline":
        obj = _mlu.load_stage(path)
        if not isinstance(obj, Pipeline):
            raise TypeError("%s is not a Pipeline" % path)
        return obj


class PipelineModel(Model):
    def __init__(self, stages: List):
        super().__init__()
        self.stages = list(stages)

    def _transform(self, dataset):
        for stage in self.stages:
            dataset = stage.transform(dataset)
        return dataset

    def write(self):
        return _mlu.MLWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def read(cls):
        return _mlu.MLReader(cls)

    @classmethod
    def load(cls, path: str) -> "PipelineModel":
        obj = _mlu.load_stage(path)
        if not isinstance(obj, PipelineModel):
            raise TypeError("%s is not a PipelineModel" % path)
        return obj


# ---------------------------------------------------------------------------
# persistence impl shared with ml_util.save_stage/load_stage
# ---------------------------------------------------------------------------


def save_pipeline(obj, path: str) -> None:
    is_model = isinstance(obj, PipelineModel)
    stages = obj.stages if is_model else obj.getStages()
    java_class = ("org.apache.spark.ml.PipelineModel" if is_model
                  else "org.apache.spark.ml.Pipeline")
    carriers = []
    for stage in stages:
        if isinstance(stage, (Pipeline, PipelineModel)):
            carriers.append(stage)
        elif hasattr(stage, "_to_java") and not hasattr(stage, "_java_class"):
            carriers.append(stage._to_java())
        else:
            carriers.append(stage)
    _mlu.write_metadata(
        path, java_class, obj.uid, {},
        extra={"paramMap": {"stageUids": [c.uid for c in carriers]}})
    for i, c in enumerate(carriers):
        _mlu.save_stage(c, os.path.join(path, "stages",
                                        "%d_%s" % (i, c.uid)))


def load_pipeline(path: str, meta: dict):
    stage_dir = os.path.join(path, "stages")
    uids = meta.get("paramMap", {}).get("stageUids", [])
    stages = []
    for i, uid in enumerate(uids):
        stages.append(_mlu.load_stage(
            os.path.join(stage_dir, "%d_%s" % (i, uid))))
    if meta["class"] == "org.apache.spark.ml.PipelineModel":
        obj = PipelineModel(stages)
    else:
        obj = Pipeline(stages=stages)
    obj.uid = meta["uid"]
    return obj
