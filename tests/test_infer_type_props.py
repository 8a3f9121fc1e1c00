"""Property-based V2 tensor codec round-trips: random dtype/shape/data
through set_data_from_numpy -> to_rest -> from_bytes (binary extension)
and the JSON path must reproduce the tensors exactly. BYTES tensors use
random byte strings incl. empty and non-UTF8."""

import numpy as np

import hypothesis.strategies as st
from hypothesis import given, settings
from hypothesis.extra.numpy import arrays, array_shapes

from kserve_amd.protocol.infer_type import InferInput, InferRequest

DTYPES = [
    ("FP32", np.float32),
    ("FP64", np.float64),
    ("INT32", np.int32),
    ("INT64", np.int64),
    ("UINT8", np.uint8),
    ("BOOL", np.bool_),
]


@st.composite
def tensor(draw):
    name, np_dtype = draw(st.sampled_from(DTYPES))
    shape = draw(array_shapes(min_dims=1, max_dims=3, min_side=1,
                              max_side=4))
    if np_dtype is np.bool_:
        arr = draw(arrays(np_dtype, shape))
    elif np.issubdtype(np_dtype, np.integer):
        info = np.iinfo(np_dtype)
        arr = draw(arrays(np_dtype, shape,
                          elements=st.integers(info.min, info.max)))
    else:
        arr = draw(arrays(
            np_dtype, shape,
            elements=st.floats(-1e6, 1e6, allow_nan=False, width=32),
        ))
    return name, arr


@settings(max_examples=60, deadline=None)
@given(st.lists(tensor(), min_size=1, max_size=3), st.booleans())
def test_binary_and_json_roundtrip(tensors, use_binary):
    inputs = []
    for i, (dt, arr) in enumerate(tensors):
        inp = InferInput(f"t{i}", list(arr.shape), dt)
        inp.set_data_from_numpy(arr, binary_data=use_binary)
        inputs.append(inp)
    req = InferRequest("m", inputs)
    body, json_len = req.to_rest()
    if json_len is None:
        decoded = InferRequest.from_inference_request("m", body)
    else:
        decoded = InferRequest.from_bytes(body, json_len, "m")
    assert len(decoded.inputs) == len(tensors)
    for (dt, arr), out in zip(tensors, decoded.inputs):
        got = out.as_numpy()
        assert got.dtype == arr.dtype, (dt, got.dtype)
        np.testing.assert_array_equal(got.reshape(arr.shape), arr)


@settings(max_examples=60, deadline=None)
@given(st.lists(st.binary(min_size=0, max_size=12), min_size=1,
                max_size=6))
def test_bytes_tensor_roundtrip(items):
    arr = np.array(items, dtype=np.object_)
    inp = InferInput("b", [len(items)], "BYTES")
    inp.set_data_from_numpy(arr, binary_data=True)
    req = InferRequest("m", [inp])
    body, json_len = req.to_rest()
    decoded = InferRequest.from_bytes(body, json_len, "m")
    assert list(decoded.inputs[0].as_numpy()) == items
