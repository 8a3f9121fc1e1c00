"""Custom predictor sample: a user-defined Model served over V1/V2/gRPC.

Reference parity: python/custom_model/model.py (AlexNet sample) — the
shape of a custom predictor: subclass Model, implement load/preprocess/
predict, run it under ModelServer. This sample uses a small torch
image-classifier-like network (random weights, offline image) so it runs
anywhere; swap `build_net` for a real model.

Run: python examples/custom_model.py --model_name custom-model
Then: curl localhost:8080/v1/models/custom-model:predict \
        -d '{"instances": [[...784 floats...]]}'
"""

import argparse
from typing import Dict

import torch

from kserve_amd.model import Model
from kserve_amd.model_server import ModelServer
from kserve_amd.errors import InvalidInput


def build_net() -> torch.nn.Module:
    torch.manual_seed(0)
    return torch.nn.Sequential(
        torch.nn.Linear(784, 128), torch.nn.ReLU(), torch.nn.Linear(128, 10)
    )


class CustomModel(Model):
    def __init__(self, name: str):
        super().__init__(name)
        self.net = None
        self.load()

    def load(self) -> bool:
        self.net = build_net().eval()
        self.ready = True
        return self.ready

    def preprocess(self, payload: Dict, headers=None) -> torch.Tensor:
        instances = payload.get("instances")
        if not instances:
            raise InvalidInput('Expected "instances"')
        return torch.tensor(instances, dtype=torch.float32)

    def predict(self, batch: torch.Tensor, headers=None) -> Dict:
        with torch.no_grad():
            logits = self.net(batch)
        return {"predictions": logits.argmax(-1).tolist()}


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--model_name", default="custom-model")
    parser.add_argument("--http_port", type=int, default=8080)
    args = parser.parse_args()
    ModelServer(http_port=args.http_port).start([CustomModel(args.model_name)])
