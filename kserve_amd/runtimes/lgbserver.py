"""LightGBM predictor runtime (reference parity: python/lgbserver/model.py:71).

lightgbm is not installed in the offline image; import is gated at load().
"""

from __future__ import annotations

import os
from typing import Dict, Union

import numpy as np

from kserve_amd.errors import InferenceError, InvalidInput
from kserve_amd.model import Model
from kserve_amd.protocol.infer_type import InferOutput, InferRequest, InferResponse, from_np_dtype


class LightGBMModel(Model):
    def __init__(self, name: str, model_dir: str, nthread: int = 1):
        super().__init__(name)
        self.model_dir = model_dir
        self.nthread = nthread
        self._booster = None

    def load(self) -> bool:
        try:
            import lightgbm as lgb
        except ImportError as e:
            raise RuntimeError(
                "lightgbm is not installed in this image; install it to use "
                "the lgbserver runtime"
            ) from e
        path = None
        for f in sorted(os.listdir(self.model_dir)):
            if f.endswith((".bst", ".txt", ".model")):
                path = os.path.join(self.model_dir, f)
                break
        if path is None:
            raise RuntimeError(f"No booster file under {self.model_dir}")
        self._booster = lgb.Booster(params={"nthread": self.nthread}, model_file=path)
        self.ready = True
        return self.ready

    def predict(self, payload: Union[Dict, InferRequest], headers=None):
        if isinstance(payload, InferRequest):
            instances = payload.inputs[0].as_numpy()
        else:
            try:
                instances = np.asarray(payload["instances"])
            except KeyError:
                raise InvalidInput('Expected "instances"')
        try:
            result = self._booster.predict(instances)
        except Exception as e:
            raise InferenceError(str(e))
        if isinstance(payload, InferRequest):
            result = np.asarray(result)
            out = InferOutput("output-0", list(result.shape), from_np_dtype(result.dtype))
            out.set_data_from_numpy(result, binary_data=payload.inputs[0].raw_data is not None)
            return InferResponse(payload.id, self.name, [out])
        return {"predictions": np.asarray(result).tolist()}


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    args = build_arg_parser().parse_args(argv)
    model = LightGBMModel(args.model_name, args.model_dir)
    model.load()
    ModelServer(
        http_port=args.http_port,
        grpc_port=args.grpc_port,
        enable_grpc=args.enable_grpc,
        workers=args.workers,
    ).start([model])


if __name__ == "__main__":
    main()
