"""Custom transformer sample: pre/post-processing in front of a predictor.

Reference parity: python/custom_transformer/model.py — preprocess converts
the user payload to the predictor's tensor format, forwards via
--predictor_host, postprocess reshapes the response.

Run: python examples/custom_transformer.py --predictor_host pred:8080
"""

import argparse
from typing import Dict

from kserve_amd.model import Model, PredictorConfig
from kserve_amd.model_server import ModelServer
from kserve_amd.errors import InvalidInput


class ImageTransformer(Model):
    def __init__(self, name: str, predictor_host: str):
        super().__init__(
            name, predictor_config=PredictorConfig(predictor_host=predictor_host)
        )
        self.ready = True

    def preprocess(self, payload: Dict, headers=None) -> Dict:
        instances = payload.get("instances")
        if instances is None:
            raise InvalidInput('Expected "instances"')
        # normalize [0,255] pixel rows to [0,1] floats for the predictor
        norm = [[float(p) / 255.0 for p in row] for row in instances]
        return {"instances": norm}

    async def predict(self, payload: Dict, headers=None) -> Dict:
        return await self._forward_predict(payload, headers)

    def postprocess(self, response: Dict, headers=None) -> Dict:
        return {"predictions": response.get("predictions", [])}


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_name", default="custom-transformer")
    parser.add_argument("--predictor_host", required=True)
    parser.add_argument("--http_port", type=int, default=8080)
    args = parser.parse_args()
    ModelServer(http_port=args.http_port).start(
        [ImageTransformer(args.model_name, args.predictor_host)]
    )
