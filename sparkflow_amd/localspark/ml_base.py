"""pyspark.ml.base subset: Estimator / Transformer / Model."""
from __future__ import annotations

from sparkflow_amd.localspark.ml_param import Params


class Estimator(Params):
    def fit(self, dataset, params=None):
        if params:
            return self.copy(params)._fit(dataset)
        return self._fit(dataset)

    def _fit(self, dataset):
        raise NotImplementedError


class Transformer(Params):
    def transform(self, dataset, params=None):
        if params:
            return self.copy(params)._transform(dataset)
        return self._transform(dataset)

    def _transform(self, dataset):
        raise NotImplementedError


class Model(Transformer):
    pass
