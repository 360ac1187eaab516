"""Property-based round-trip tests for the persistence codecs
(hypothesis): weights<->JSON (the reference's model checkpoint format,
ml_util.py:31-40) and flat-buffer packing must be exact inverses for any
shape mix."""
import hypothesis.strategies as st
import numpy as np
import torch
from hypothesis import given, settings

from sparkflow_amd.ml_util import (
    convert_json_to_weights,
    convert_weights_to_json,
)
from sparkflow_amd.utils.serialization import flat_to_list, list_to_flat


shapes = st.lists(
    st.lists(st.integers(1, 7), min_size=1, max_size=3),
    min_size=1, max_size=5,
)


@settings(max_examples=25, deadline=None)
@given(shapes, st.integers(0, 2 ** 31 - 1))
def test_weights_json_roundtrip(shape_list, seed):
    rng = np.random.default_rng(seed)
    ws = [rng.standard_normal(s).astype(np.float32) for s in shape_list]
    back = convert_json_to_weights(convert_weights_to_json(ws))
    assert len(back) == len(ws)
    for a, b in zip(ws, back):
        np.testing.assert_allclose(a, np.asarray(b, dtype=np.float32),
                                   rtol=1e-6, atol=1e-6)


@settings(max_examples=25, deadline=None)
@given(shapes, st.integers(0, 2 ** 31 - 1))
def test_flat_pack_roundtrip(shape_list, seed):
    rng = np.random.default_rng(seed)
    ws = [rng.standard_normal(s).astype(np.float32) for s in shape_list]
    flat = list_to_flat(ws)
    assert flat.numel() == sum(int(np.prod(s)) for s in shape_list)

    class Holder(torch.nn.Module):
        def __init__(self):
            super().__init__()
            for i, w in enumerate(ws):
                self.register_parameter(
                    f"p{i}", torch.nn.Parameter(torch.as_tensor(w)))

    back = flat_to_list(Holder(), flat)
    for a, b in zip(ws, back):
        np.testing.assert_array_equal(a, b)
