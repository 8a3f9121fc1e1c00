"""Prometheus metrics for the data plane.

Reference parity: python/kserve/kserve/metrics.py:19-40 (per-stage histograms)
plus engine-level LLM serving metrics the native engine exposes.
"""

from prometheus_client import Counter, Gauge, Histogram

PRE_HIST = Histogram(
    "request_preprocess_seconds", "pre-process request latency", ["model_name"]
)
PREDICT_HIST = Histogram(
    "request_predict_seconds", "predict request latency", ["model_name"]
)
POST_HIST = Histogram(
    "request_postprocess_seconds", "post-process request latency", ["model_name"]
)
EXPLAIN_HIST = Histogram(
    "request_explain_seconds", "explain request latency", ["model_name"]
)

# ---- native LLM engine metrics ----
LLM_NUM_RUNNING = Gauge("llm_num_running_requests", "requests in the running batch")
LLM_NUM_WAITING = Gauge("llm_num_waiting_requests", "requests queued for prefill")
LLM_GENERATION_TOKENS = Counter(
    "llm_generation_tokens_total", "output tokens generated"
)
LLM_PROMPT_TOKENS = Counter("llm_prompt_tokens_total", "prompt tokens processed")
LLM_SPEC_ACCEPTED = Counter(
    "llm_spec_decode_accepted_tokens_total",
    "draft tokens accepted by speculative decoding",
)
LLM_TTFT_HIST = Histogram(
    "llm_time_to_first_token_seconds",
    "time to first token",
    buckets=(0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0),
)
LLM_TPOT_HIST = Histogram(
    "llm_time_per_output_token_seconds",
    "inter-token latency",
    buckets=(0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25),
)
LLM_E2E_HIST = Histogram("llm_e2e_request_seconds", "end-to-end request latency")
LLM_KV_USAGE = Gauge("llm_kv_cache_usage_ratio", "fraction of GPU KV blocks in use")
LLM_PREEMPTIONS = Counter("llm_preemptions_total", "scheduler preemptions")


def get_labeled_histograms(model_name: str):
    return {
        "preprocess": PRE_HIST.labels(model_name=model_name),
        "predict": PREDICT_HIST.labels(model_name=model_name),
        "postprocess": POST_HIST.labels(model_name=model_name),
        "explain": EXPLAIN_HIST.labels(model_name=model_name),
    }
