"""PMML runtime (reference parity: python/pmmlserver/model.py:75, pypmml/JPype).

pypmml is not installed in this offline image; import gated at load().
"""

from __future__ import annotations

import os
from typing import Dict, Union

import numpy as np

from kserve_amd.errors import InferenceError, InvalidInput
from kserve_amd.model import Model
from kserve_amd.protocol.infer_type import InferRequest


class PMMLModel(Model):
    def __init__(self, name: str, model_dir: str):
        super().__init__(name)
        self.model_dir = model_dir
        self._model = None

    def load(self) -> bool:
        try:
            from pypmml import Model as PmmlModel
        except ImportError as e:
            raise RuntimeError(
                "pypmml is not installed in this image; install it to use "
                "the pmmlserver runtime"
            ) from e
        path = None
        for f in sorted(os.listdir(self.model_dir)):
            if f.endswith(".pmml") or f.endswith(".xml"):
                path = os.path.join(self.model_dir, f)
                break
        if path is None:
            raise RuntimeError(f"No .pmml file under {self.model_dir}")
        self._model = PmmlModel.load(path)
        self.ready = True
        return self.ready

    def predict(self, payload: Union[Dict, InferRequest], headers=None):
        if isinstance(payload, InferRequest):
            instances = payload.inputs[0].as_numpy().tolist()
        else:
            try:
                instances = payload["instances"]
            except KeyError:
                raise InvalidInput('Expected "instances"')
        try:
            results = [self._model.predict(row) for row in instances]
        except Exception as e:
            raise InferenceError(str(e))
        return {"predictions": results}


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    args = build_arg_parser().parse_args(argv)
    model = PMMLModel(args.model_name, args.model_dir)
    model.load()
    ModelServer(
        http_port=args.http_port,
        grpc_port=args.grpc_port,
        enable_grpc=args.enable_grpc,
    ).start([model])


if __name__ == "__main__":
    main()
