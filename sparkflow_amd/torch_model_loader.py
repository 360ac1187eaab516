"""Drop-in parity module for reference ``sparkflow.tensorflow_model_loader``
(torch checkpoints instead of TF saver checkpoints)."""
from sparkflow_amd.spark.model_loader import (  # noqa: F401
    attach_torch_model_to_pipeline,
    load_torch_model,
)
