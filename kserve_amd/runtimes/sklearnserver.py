"""scikit-learn predictor runtime.

Reference parity: python/sklearnserver (model.py:69 LoC + __main__) —
joblib/pickle model from /mnt/models, V1 instances + V2 tensor protocol.
Run: python -m kserve_amd.runtimes.sklearnserver --model_dir ... --model_name ...
"""

from __future__ import annotations

import os
import pickle
from typing import Dict, Union

import numpy as np

from kserve_amd.errors import InferenceError, InvalidInput
from kserve_amd.model import Model
from kserve_amd.protocol.infer_type import (
    InferOutput,
    InferRequest,
    InferResponse,
    from_np_dtype,
)

MODEL_EXTENSIONS = (".joblib", ".pkl", ".pickle")


class SKLearnModel(Model):
    def __init__(self, name: str, model_dir: str):
        super().__init__(name)
        self.model_dir = model_dir
        self._model = None

    def load(self) -> bool:
        model_path = None
        for f in sorted(os.listdir(self.model_dir)):
            if f.endswith(MODEL_EXTENSIONS):
                model_path = os.path.join(self.model_dir, f)
                break
        if model_path is None:
            raise RuntimeError(
                f"No model file ({MODEL_EXTENSIONS}) under {self.model_dir}"
            )
        if model_path.endswith(".joblib"):
            import joblib

            self._model = joblib.load(model_path)
        else:
            with open(model_path, "rb") as f:
                self._model = pickle.load(f)
        self.ready = True
        return self.ready

    def predict(self, payload: Union[Dict, InferRequest], headers=None):
        try:
            if isinstance(payload, InferRequest):
                instances = payload.inputs[0].as_numpy()
            else:
                instances = np.asarray(payload["instances"])
        except (KeyError, IndexError):
            raise InvalidInput('Expected "instances" or V2 inputs')
        try:
            result = self._model.predict(instances)
        except Exception as e:
            raise InferenceError(str(e))
        if isinstance(payload, InferRequest):
            result = np.asarray(result)
            out = InferOutput(
                "output-0",
                list(result.shape),
                from_np_dtype(result.dtype),
            )
            out.set_data_from_numpy(result, binary_data=payload.inputs[0].raw_data is not None)
            return InferResponse(payload.id, self.name, [out])
        return {"predictions": np.asarray(result).tolist()}


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    args = build_arg_parser().parse_args(argv)
    model = SKLearnModel(args.model_name, args.model_dir)
    model.load()
    ModelServer(
        http_port=args.http_port,
        grpc_port=args.grpc_port,
        enable_grpc=args.enable_grpc,
        workers=args.workers,
    ).start([model])


if __name__ == "__main__":
    main()
