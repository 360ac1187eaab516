"""Import a pre-trained PyTorch model as a SparkAsyncDLModel — parity with
reference sparkflow/tensorflow_model_loader.py (which imports a TF saver
checkpoint; here the inbound artifact is a torch checkpoint).

Accepted checkpoint forms for ``load_torch_model``:
* a ``build_graph`` JSON artifact file (``*.json``), optionally with a
  separate weights JSON;
* a ``torch.save``'d ``nn.Module`` instance (``*.pt`` / ``*.pth``);
* a ``torch.save``'d ``state_dict`` plus a ``builder`` callable.
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Callable, Optional

import torch

from sparkflow_amd.ml_util import convert_weights_to_json, torch_get_weights
from sparkflow_amd.utils.serialization import capture_module, module_from_json


def load_torch_model(model_path: str,
                     inputCol: str = "encoded",
                     predictionCol: str = "predicted",
                     tfInput: str = "x:0",
                     tfOutput: str = "out:0",
                     tfDropout: Optional[str] = None,
                     toKeepDropout: bool = False,
                     builder: Optional[Callable] = None):
    """Restore a trained model and wrap it as a ``SparkAsyncDLModel``
    (parity: tensorflow_model_loader.load_tensorflow_model,
    tensorflow_model_loader.py:8-32)."""
    from sparkflow_amd.spark.async_dl import SparkAsyncDLModel

    path = Path(model_path)
    if path.suffix == ".json":
        graph_json = path.read_text()
        module = module_from_json(graph_json)
    else:
        obj = torch.load(str(path), map_location="cpu", weights_only=False)
        if isinstance(obj, torch.nn.Module):
            module = obj
        else:  # state_dict + builder
            if builder is None:
                raise ValueError(
                    "a state_dict checkpoint needs a builder callable")
            module = builder()
            module.load_state_dict(obj)
        graph_json = capture_module(module)

    weights_json = convert_weights_to_json(torch_get_weights(module))
    return SparkAsyncDLModel(
        inputCol=inputCol,
        modelJson=graph_json,
        modelWeights=weights_json,
        tfInput=tfInput,
        tfOutput=tfOutput,
        tfDropout=tfDropout,
        toKeepDropout=toKeepDropout,
        predictionCol=predictionCol,
    )


def attach_torch_model_to_pipeline(model_path: str, pipelineModel,
                                   inputCol: str = "encoded",
                                   predictionCol: str = "predicted",
                                   tfInput: str = "x:0",
                                   tfOutput: str = "out:0",
                                   builder: Optional[Callable] = None):
    """Append the restored model to an existing PipelineModel (parity:
    attach_tensorflow_model_to_pipeline, tensorflow_model_loader.py:35-45)."""
    from pyspark.ml import PipelineModel

    spark_model = load_torch_model(model_path, inputCol=inputCol,
                                   predictionCol=predictionCol,
                                   tfInput=tfInput, tfOutput=tfOutput,
                                   builder=builder)
    return PipelineModel(stages=[pipelineModel, spark_model])
