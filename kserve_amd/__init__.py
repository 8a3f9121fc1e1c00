"""kserve_amd — MI355X-native model-serving platform.

A from-scratch serving framework with KServe's capabilities: V1/V2/OpenAI
inference protocols, a ModelServer data plane, and a native LLM engine whose
hot path is hand-written CDNA4 HIP kernels (paged attention, RMSNorm, RoPE,
sampling) with RCCL tensor parallelism over xGMI.
"""

__version__ = "0.1.0"

from kserve_amd.model import (  # noqa: F401
    BaseModel,
    InferenceModel,
    Model,
    OpenAIModel,
    PredictorConfig,
    PredictorProtocol,
)
from kserve_amd.model_repository import ModelRepository  # noqa: F401
from kserve_amd.model_server import ModelServer, build_arg_parser  # noqa: F401
from kserve_amd.protocol.infer_type import (  # noqa: F401
    InferInput,
    InferOutput,
    InferRequest,
    InferResponse,
    RequestedOutput,
)
