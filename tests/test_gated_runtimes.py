"""Import-gated runtime predict paths exercised against stub libraries.

xgboost/lightgbm/paddle/pypmml are absent from this offline image, so
round 1 never ran these runtimes' predict logic. These tests inject
minimal stand-ins implementing exactly the API surface each runtime
calls (the reference's mock-based storage tests play the same role), so
the V1/V2 plumbing, file discovery, and error mapping are verified even
without the real libraries."""

import sys
import types

import numpy as np
import pytest

from kserve_amd.protocol.infer_type import InferInput, InferRequest


@pytest.fixture
def stub_xgboost(monkeypatch, tmp_path):
    mod = types.ModuleType("xgboost")

    class DMatrix:
        def __init__(self, data, nthread=1):
            self.data = np.asarray(data)

    class Booster:
        def __init__(self, params=None, model_file=None):
            assert model_file and model_file.endswith(".bst")
            self.model_file = model_file

        def predict(self, dmatrix):
            # deterministic fake: row sums
            return dmatrix.data.sum(axis=1)

    mod.DMatrix = DMatrix
    mod.Booster = Booster
    monkeypatch.setitem(sys.modules, "xgboost", mod)
    (tmp_path / "model.bst").write_bytes(b"stub")
    return str(tmp_path)


class TestXGBoostRuntime:
    def test_v1_predict(self, stub_xgboost):
        from kserve_amd.runtimes.xgbserver import XGBoostModel

        m = XGBoostModel("xgb", stub_xgboost)
        assert m.load()
        out = m.predict({"instances": [[1.0, 2.0], [3.0, 4.0]]})
        assert out["predictions"] == [3.0, 7.0]

    def test_v2_predict_binary(self, stub_xgboost):
        from kserve_amd.runtimes.xgbserver import XGBoostModel

        m = XGBoostModel("xgb", stub_xgboost)
        m.load()
        x = np.array([[1.0, 1.5]], dtype=np.float32)
        inp = InferInput("input-0", [1, 2], "FP32")
        inp.set_data_from_numpy(x, binary_data=True)
        resp = m.predict(InferRequest("xgb", [inp]))
        got = resp.outputs[0].as_numpy()
        assert np.allclose(got, [2.5])

    def test_missing_booster_file(self, monkeypatch, tmp_path):
        mod = types.ModuleType("xgboost")
        monkeypatch.setitem(sys.modules, "xgboost", mod)
        from kserve_amd.runtimes.xgbserver import XGBoostModel

        with pytest.raises(RuntimeError, match="No booster"):
            XGBoostModel("xgb", str(tmp_path)).load()

    def test_inference_error_mapped(self, stub_xgboost):
        from kserve_amd.errors import InferenceError
        from kserve_amd.runtimes.xgbserver import XGBoostModel

        m = XGBoostModel("xgb", stub_xgboost)
        m.load()

        def boom(_):
            raise ValueError("bad shape")

        m._booster.predict = boom
        with pytest.raises(InferenceError, match="bad shape"):
            m.predict({"instances": [[1.0]]})


@pytest.fixture
def stub_lightgbm(monkeypatch, tmp_path):
    mod = types.ModuleType("lightgbm")

    class Booster:
        def __init__(self, model_file=None, params=None):
            assert model_file
            self.model_file = model_file

        def predict(self, data):
            return np.asarray(data).mean(axis=1)

    mod.Booster = Booster
    monkeypatch.setitem(sys.modules, "lightgbm", mod)
    (tmp_path / "model.txt").write_text("stub")
    return str(tmp_path)


class TestLightGBMRuntime:
    def test_v1_predict(self, stub_lightgbm):
        from kserve_amd.runtimes.lgbserver import LightGBMModel

        m = LightGBMModel("lgb", stub_lightgbm)
        assert m.load()
        out = m.predict({"instances": [[2.0, 4.0]]})
        assert out["predictions"] == [3.0]


@pytest.fixture
def stub_pmml(monkeypatch, tmp_path):
    pypmml = types.ModuleType("pypmml")

    class PmmlModel:
        @classmethod
        def load(cls, path):
            assert path.endswith(".pmml")
            return cls()

        def predict(self, row):
            return {"prediction": sum(row)}

    pypmml.Model = PmmlModel
    monkeypatch.setitem(sys.modules, "pypmml", pypmml)
    (tmp_path / "model.pmml").write_text("<PMML/>")
    return str(tmp_path)


class TestPMMLRuntime:
    def test_v1_predict(self, stub_pmml):
        from kserve_amd.runtimes.pmmlserver import PMMLModel

        m = PMMLModel("pmml", stub_pmml)
        assert m.load()
        out = m.predict({"instances": [[1.0, 2.0, 3.0]]})
        assert out["predictions"][0]["prediction"] == 6.0
