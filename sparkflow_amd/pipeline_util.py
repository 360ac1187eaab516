"""Drop-in parity module for reference ``sparkflow.pipeline_util``."""
from sparkflow_amd.spark.pipeline_util import (  # noqa: F401
    PY_OBJ_GUID,
    PysparkObjId,
    PysparkPipelineWrapper,
    PysparkReaderWriter,
    decode_py_obj,
    encode_py_obj,
    load_byte_array,
)
