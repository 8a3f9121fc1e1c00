"""V2 tensor codec tests (behavioral parity with reference
python/kserve/test/test_infer_type.py — written fresh)."""

import json

import numpy as np
import pytest

from kserve_amd.errors import InvalidInput
from kserve_amd.protocol.infer_type import (
    InferInput,
    InferOutput,
    InferRequest,
    InferResponse,
    RequestedOutput,
    deserialize_bytes_tensor,
    serialize_bytes_tensor,
)


class TestBytesTensor:
    def test_roundtrip(self):
        arr = np.array([b"hello", b"", b"world!"], dtype=np.object_)
        raw = serialize_bytes_tensor(arr)
        # 4-byte LE length prefix per element
        assert raw[:4] == (5).to_bytes(4, "little")
        out = deserialize_bytes_tensor(raw)
        assert list(out) == [b"hello", b"", b"world!"]

    def test_truncated(self):
        with pytest.raises(InvalidInput):
            deserialize_bytes_tensor(b"\x05\x00\x00\x00abc")

    def test_strings(self):
        arr = np.array(["a", "bc"], dtype=np.object_)
        raw = serialize_bytes_tensor(arr)
        out = deserialize_bytes_tensor(raw)
        assert [x.decode() for x in out] == ["a", "bc"]


class TestInferInput:
    def test_json_data_numpy(self):
        inp = InferInput("x", [2, 2], "FP32", data=[1.0, 2.0, 3.0, 4.0])
        arr = inp.as_numpy()
        assert arr.dtype == np.float32
        assert arr.shape == (2, 2)

    def test_binary_numpy_roundtrip(self):
        orig = np.arange(12, dtype=np.float32).reshape(3, 4)
        inp = InferInput("x", [3, 4], "FP32")
        inp.set_data_from_numpy(orig, binary_data=True)
        assert inp.parameters["binary_data_size"] == orig.nbytes
        np.testing.assert_array_equal(inp.as_numpy(), orig)

    def test_fp16_json_rejected(self):
        with pytest.raises(InvalidInput):
            InferInput.from_dict(
                {"name": "x", "shape": [2], "datatype": "FP16", "data": [1.0, 2.0]}
            )

    def test_int64(self):
        orig = np.array([[1, 2], [3, 4]], dtype=np.int64)
        inp = InferInput("ids", [2, 2], "INT64")
        inp.set_data_from_numpy(orig, binary_data=False)
        assert inp.data == [1, 2, 3, 4]
        np.testing.assert_array_equal(inp.as_numpy(), orig)


class TestBinaryExtension:
    def _make_request(self):
        x = np.arange(6, dtype=np.float32).reshape(2, 3)
        ids = np.array([b"a", b"b"], dtype=np.object_)
        i1 = InferInput("x", [2, 3], "FP32")
        i1.set_data_from_numpy(x, binary_data=True)
        i2 = InferInput("ids", [2], "BYTES")
        i2.set_data_from_numpy(ids, binary_data=True)
        return InferRequest("m", [i1, i2]), x, ids

    def test_to_rest_binary(self):
        req, x, ids = self._make_request()
        body, json_len = req.to_rest()
        assert isinstance(body, bytes)
        meta = json.loads(body[:json_len])
        assert meta["inputs"][0]["parameters"]["binary_data_size"] == x.nbytes
        # raw tensors concatenated after JSON prefix, in input order
        got_x = np.frombuffer(
            body[json_len : json_len + x.nbytes], dtype=np.float32
        ).reshape(2, 3)
        np.testing.assert_array_equal(got_x, x)

    def test_from_bytes_roundtrip(self):
        req, x, ids = self._make_request()
        body, json_len = req.to_rest()
        decoded = InferRequest.from_bytes(body, json_len, "m")
        np.testing.assert_array_equal(decoded.inputs[0].as_numpy(), x)
        assert list(decoded.inputs[1].as_numpy()) == [b"a", b"b"]

    def test_from_bytes_truncated(self):
        req, _, _ = self._make_request()
        body, json_len = req.to_rest()
        with pytest.raises(InvalidInput):
            InferRequest.from_bytes(body[:-4], json_len, "m")

    def test_json_only_roundtrip(self):
        i = InferInput("x", [2], "FP32", data=[1.5, 2.5])
        req = InferRequest("m", [i])
        body, json_len = req.to_rest()
        assert json_len is None
        assert isinstance(body, dict)
        decoded = InferRequest.from_inference_request("m", body)
        np.testing.assert_array_equal(
            decoded.inputs[0].as_numpy(), np.array([1.5, 2.5], dtype=np.float32)
        )

    def test_mixed_json_and_binary(self):
        x = np.arange(4, dtype=np.float32)
        i1 = InferInput("a", [4], "FP32", data=[9.0, 8.0, 7.0, 6.0])
        i2 = InferInput("b", [4], "FP32")
        i2.set_data_from_numpy(x, binary_data=True)
        req = InferRequest("m", [i1, i2])
        body, json_len = req.to_rest()
        decoded = InferRequest.from_bytes(body, json_len, "m")
        np.testing.assert_array_equal(
            decoded.inputs[0].as_numpy(), np.array([9.0, 8.0, 7.0, 6.0], dtype=np.float32)
        )
        np.testing.assert_array_equal(decoded.inputs[1].as_numpy(), x)


class TestInferResponse:
    def test_response_binary_roundtrip(self):
        y = np.arange(8, dtype=np.float32).reshape(2, 4)
        out = InferOutput("y", [2, 4], "FP32")
        out.set_data_from_numpy(y, binary_data=True)
        resp = InferResponse("id-1", "m", [out])
        body, json_len = resp.to_rest()
        assert json_len is not None
        decoded = InferResponse.from_bytes(body, json_len)
        np.testing.assert_array_equal(decoded.outputs[0].as_numpy(), y)

    def test_response_json_when_requested(self):
        y = np.arange(4, dtype=np.float32)
        out = InferOutput("y", [4], "FP32")
        out.set_data_from_numpy(y, binary_data=True)
        resp = InferResponse("id-1", "m", [out])
        body, json_len = resp.to_rest([RequestedOutput("y", {"binary_data": False})])
        assert json_len is None
        assert body["outputs"][0]["data"] == [0.0, 1.0, 2.0, 3.0]

    def test_bf16_binary(self):
        bits = np.array([16256, 16384], dtype=np.uint16)  # 1.0, 2.0 in bf16
        out = InferOutput("y", [2], "BF16")
        out.set_raw_data(bits.tobytes())
        resp = InferResponse("id", "m", [out])
        body, json_len = resp.to_rest()
        decoded = InferResponse.from_bytes(body, json_len)
        np.testing.assert_array_equal(decoded.outputs[0].as_numpy(), bits)


class TestDataFrame:
    def test_as_dataframe(self):
        i1 = InferInput("a", [2], "FP32", data=[1.0, 2.0])
        i2 = InferInput("b", [2], "BYTES", data=["x", "y"])
        df = InferRequest("m", [i1, i2]).as_dataframe()
        assert list(df.columns) == ["a", "b"]
        assert df["b"].tolist() == ["x", "y"]


class TestBytesEdgeCases:
    def test_empty_and_unicode_bytes_roundtrip(self):
        """BYTES tensors with empty and multibyte elements survive the
        4-byte-LE length-prefixed binary codec."""
        import numpy as np

        from kserve_amd.protocol.infer_type import InferInput, InferRequest

        vals = np.array(["", "héllo", "日本語", "x" * 300], dtype=object)
        inp = InferInput(name="s", shape=[4], datatype="BYTES")
        inp.set_data_from_numpy(vals, binary_data=True)
        req = InferRequest(model_name="m", infer_inputs=[inp])
        body, json_len = req.to_rest()
        assert isinstance(body, bytes)
        back = InferRequest.from_bytes(body, json_len, "m")
        out = back.inputs[0].as_numpy()
        assert [
            v.decode("utf-8") if isinstance(v, bytes) else v for v in out
        ] == list(vals)

    def test_zero_element_tensor(self):
        import numpy as np

        from kserve_amd.protocol.infer_type import InferInput

        inp = InferInput(name="e", shape=[0], datatype="FP32")
        inp.set_data_from_numpy(np.zeros((0,), dtype=np.float32), binary_data=True)
        assert inp.parameters["binary_data_size"] == 0
