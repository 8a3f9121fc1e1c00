"""Paddle inference runtime (reference parity: python/paddleserver/model.py:91).

paddlepaddle is not installed in this offline image; import gated at load().
"""

from __future__ import annotations

import os
from typing import Dict, Union

import numpy as np

from kserve_amd.errors import InferenceError, InvalidInput
from kserve_amd.model import Model
from kserve_amd.protocol.infer_type import InferOutput, InferRequest, InferResponse, from_np_dtype


class PaddleModel(Model):
    def __init__(self, name: str, model_dir: str):
        super().__init__(name)
        self.model_dir = model_dir
        self._predictor = None
        self._input_names = None

    def load(self) -> bool:
        try:
            import paddle.inference as paddle_infer
        except ImportError as e:
            raise RuntimeError(
                "paddlepaddle is not installed in this image; install it to "
                "use the paddleserver runtime"
            ) from e
        model_file = None
        params_file = None
        for f in sorted(os.listdir(self.model_dir)):
            if f.endswith(".pdmodel"):
                model_file = os.path.join(self.model_dir, f)
            elif f.endswith(".pdiparams"):
                params_file = os.path.join(self.model_dir, f)
        if model_file is None:
            raise RuntimeError(f"No .pdmodel under {self.model_dir}")
        config = paddle_infer.Config(model_file, params_file)
        self._predictor = paddle_infer.create_predictor(config)
        self._input_names = self._predictor.get_input_names()
        self.ready = True
        return self.ready

    def predict(self, payload: Union[Dict, InferRequest], headers=None):
        if isinstance(payload, InferRequest):
            instances = payload.inputs[0].as_numpy().astype(np.float32)
        else:
            try:
                instances = np.asarray(payload["instances"], dtype=np.float32)
            except KeyError:
                raise InvalidInput('Expected "instances"')
        try:
            handle = self._predictor.get_input_handle(self._input_names[0])
            handle.reshape(instances.shape)
            handle.copy_from_cpu(instances)
            self._predictor.run()
            out_names = self._predictor.get_output_names()
            result = self._predictor.get_output_handle(out_names[0]).copy_to_cpu()
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
    model = PaddleModel(args.model_name, args.model_dir)
    model.load()
    ModelServer(
        http_port=args.http_port,
        grpc_port=args.grpc_port,
        enable_grpc=args.enable_grpc,
    ).start([model])


if __name__ == "__main__":
    main()
