"""AutoGluon tabular predictor runtime (import-gated).

Reference parity: python/autogluonserver — TabularPredictor.load(model_dir)
+ predict over V1/V2 tabular payloads. The autogluon library is not in this
offline image, so the wrapper raises a clear error at load() unless it is
installed; payload handling (records vs matrix) matches the reference.
"""

from __future__ import annotations

from typing import Dict

from kserve_amd.errors import InvalidInput
from kserve_amd.model import Model


class AutoGluonModel(Model):
    def __init__(self, name: str, model_dir: str):
        super().__init__(name)
        self.model_dir = model_dir
        self.predictor = None

    def load(self) -> bool:
        try:
            from autogluon.tabular import TabularPredictor
        except ImportError as e:
            raise RuntimeError(
                "autogluon is not installed in this offline image; install "
                "autogluon.tabular to serve AutoGluon models"
            ) from e
        self.predictor = TabularPredictor.load(self.model_dir)
        self.ready = True
        return self.ready

    def predict(self, payload: Dict, headers=None) -> Dict:
        import pandas as pd

        instances = payload.get("instances")
        if instances is None:
            raise InvalidInput('Expected "instances"')
        if instances and isinstance(instances[0], dict):
            frame = pd.DataFrame.from_records(instances)
        else:
            frame = pd.DataFrame(instances)
        result = self.predictor.predict(frame)
        return {"predictions": result.tolist()}


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    args = build_arg_parser().parse_args(argv)
    model = AutoGluonModel(args.model_name, args.model_dir)
    model.load()
    ModelServer(http_port=args.http_port).start([model])


if __name__ == "__main__":
    main()
