"""Stable platform constants.

Mirrors the reference's pod/data-plane contract so control-plane presets carry
over bit-for-bit (reference: pkg/constants/constants.go, python constants.py:104-105,
SURVEY.md §7.1):
  - model server HTTP 8080 / gRPC 8081, agent proxy 9081, router 8080
  - models mounted at /mnt/models, agent model configs at /mnt/configs
"""

# --- ports (reference: pkg/constants/constants.go:396, cmd/agent/main.go:56) ---
HTTP_PORT = 8080
GRPC_PORT = 8081
AGENT_PORT = 9081
LOG_MARSHALLER_PORT = 9083
ROUTER_PORT = 8080

# --- paths (reference: constants.go:510-511, kserve_storage.py:62) ---
MODEL_MOUNT_PATH = "/mnt/models"
MODEL_CONFIG_MOUNT_PATH = "/mnt/configs"
PVC_MOUNT_PATH = "/mnt/pvc"

# --- container names (reference: constants.go:94,496-503) ---
INFERENCE_CONTAINER = "kserve-container"
STORAGE_INITIALIZER_CONTAINER = "storage-initializer"
TRANSFORMER_CONTAINER = "transformer-container"
WORKER_CONTAINER = "worker-container"
AGENT_CONTAINER = "agent"

# --- protocol headers ---
# V2 binary tensor extension (reference: python/kserve protocol/dataplane.py:393-405)
INFERENCE_CONTENT_LENGTH_HEADER = "inference-content-length"
# headers forwarded to downstream predictors (reference: model.py:45 _FORWARDABLE_HEADERS)
FORWARDABLE_HEADERS = ("x-request-id", "x-b3-traceid", "authorization")
REQUEST_ID_HEADER = "x-request-id"

# --- V1/V2 route shapes (reference: v1_endpoints.py:155-171, v2_endpoints.py:236-305) ---
V1_ROUTE_PREFIX = "/v1"
V2_ROUTE_PREFIX = "/v2"
OPENAI_ROUTE_PREFIX = "/openai/v1"

# --- misc defaults ---
DEFAULT_MODEL_NAME = "model"
# batcher defaults (reference: pkg/batcher/handler.go:33-37)
DEFAULT_MAX_BATCH_SIZE = 32
DEFAULT_MAX_LATENCY_MS = 5000

# --- engine defaults (MI355X sizing: 288 GB HBM3E per GPU) ---
HBM_BYTES_PER_GPU = 288 * (1 << 30)
DEFAULT_GPU_MEMORY_UTILIZATION = 0.90
DEFAULT_KV_BLOCK_SIZE = 16
