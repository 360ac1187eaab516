"""pyspark.ml.evaluation subset: MulticlassClassificationEvaluator."""
from __future__ import annotations

import numpy as np

from sparkflow_amd.localspark.ml_param import (
    HasLabelCol,
    HasPredictionCol,
    Param,
    Params,
    TypeConverters,
    keyword_only,
)


class MulticlassClassificationEvaluator(HasPredictionCol, HasLabelCol):
    metricName = Param(Params._dummy(), "metricName",
                       "metric: f1|accuracy|weightedPrecision|weightedRecall",
                       typeConverter=TypeConverters.toString)

    @keyword_only
    def __init__(self, predictionCol="prediction", labelCol="label",
                 metricName="f1"):
        super().__init__()
        self._setDefault(predictionCol="prediction", labelCol="label",
                         metricName="f1")
        kwargs = self._input_kwargs
        self._set(**{k: v for k, v in kwargs.items() if v is not None})

    def evaluate(self, dataset) -> float:
        pred_col = self.getPredictionCol()
        label_col = self.getLabelCol()
        rows = dataset.collect()
        y = np.array([float(r[label_col]) for r in rows])
        p = np.array([float(r[pred_col]) for r in rows])
        metric = self.getOrDefault(self.metricName)
        if metric == "accuracy":
            return float((y == p).mean())
        labels = np.unique(np.concatenate([y, p]))
        precisions, recalls, f1s, weights = [], [], [], []
        for c in labels:
            tp = float(((p == c) & (y == c)).sum())
            fp = float(((p == c) & (y != c)).sum())
            fn = float(((p != c) & (y == c)).sum())
            prec = tp / (tp + fp) if tp + fp > 0 else 0.0
            rec = tp / (tp + fn) if tp + fn > 0 else 0.0
            f1 = 2 * prec * rec / (prec + rec) if prec + rec > 0 else 0.0
            precisions.append(prec)
            recalls.append(rec)
            f1s.append(f1)
            weights.append(float((y == c).sum()) / len(y))
        w = np.array(weights)
        if metric == "weightedPrecision":
            return float((np.array(precisions) * w).sum())
        if metric == "weightedRecall":
            return float((np.array(recalls) * w).sum())
        return float((np.array(f1s) * w).sum())  # f1 (default)

    def isLargerBetter(self) -> bool:
        return True
