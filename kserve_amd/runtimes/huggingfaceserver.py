"""HuggingFace-model server: backend selection between the native LLM engine
(generative, Llama-family) and the native encoder (BERT-class).

Reference parity: python/huggingfaceserver/__main__.py:60-323 (backend
select: vllm-if-supported else HF; ours: native engine if decoder-only
architecture, native encoder if BERT-class).
Run: python -m kserve_amd.runtimes.huggingfaceserver --model_dir ... \
     [--model_name m] [--task fill_mask] [--tensor-parallel-size N]
"""

from __future__ import annotations

import json
import os

from kserve_amd.logging import configure_logging, logger

GENERATIVE_ARCHITECTURES = (
    "LlamaForCausalLM",
    "MistralForCausalLM",
    "MixtralForCausalLM",
    "Qwen2ForCausalLM",
)
ENCODER_SUFFIXES = (
    "ForMaskedLM",
    "ForSequenceClassification",
    "ForTokenClassification",
)


def detect_backend(model_dir: str) -> str:
    with open(os.path.join(model_dir, "config.json")) as f:
        cfg = json.load(f)
    archs = cfg.get("architectures") or []
    for a in archs:
        if a in GENERATIVE_ARCHITECTURES:
            return "engine"
        for sfx in ENCODER_SUFFIXES:
            if a.endswith(sfx):
                return "encoder"
        if a.endswith("Model") and a.startswith(("Bert", "Roberta", "Distil")):
            return "encoder"
    model_type = cfg.get("model_type", "")
    if model_type in ("llama", "mistral", "mixtral", "qwen2"):
        return "engine"
    if any(a.endswith(("ForCausalLM", "LMHeadModel")) for a in archs):
        # decoder the native engine doesn't implement: HF generate fallback
        # (reference generative_model.py behavior — request-serial)
        return "hf"
    return "encoder"


def build_model(args):
    backend = getattr(args, "backend", None) or detect_backend(args.model_dir)
    logger.info("huggingfaceserver backend: %s", backend)
    if backend == "engine":
        import torch

        from kserve_amd.engine.config import (
            CacheConfig,
            EngineConfig,
            ModelConfig,
            ParallelConfig,
            SchedulerConfig,
        )
        from kserve_amd.parallel import comm
        from kserve_amd.runtimes.llm_model import LLMModel

        comm.init_distributed(
            tp_size=getattr(args, "tensor_parallel_size", 1),
            pp_size=getattr(args, "pipeline_parallel_size", 1),
        )
        from transformers import AutoTokenizer

        tokenizer = AutoTokenizer.from_pretrained(args.model_dir)
        model_cfg = ModelConfig.from_hf_config(
            os.path.join(args.model_dir, "config.json")
        )
        if getattr(args, "expert_parallel", False):
            model_cfg.expert_parallel = True
        # draft-model speculation (vLLM --speculative-config equivalent)
        draft_cfg = None
        draft_dir = getattr(args, "speculative_draft_model", None)
        if draft_dir:
            draft_cfg = ModelConfig.from_hf_config(
                os.path.join(draft_dir, "config.json")
            )
        if model_cfg.sliding_window:
            logger.info(
                "sliding_window=%d: attention bounded in-kernel to the last "
                "%d positions (Mistral-family numerics beyond the window)",
                model_cfg.sliding_window, model_cfg.sliding_window,
            )
        cfg = EngineConfig(
            model=model_cfg,
            cache=CacheConfig(
                cpu_offload_bytes=getattr(args, "kv_offload_bytes", 0),
                kv_cache_dtype=getattr(args, "kv_cache_dtype", "auto"),
            ),
            scheduler=SchedulerConfig(
                max_num_seqs=getattr(args, "max_num_seqs", 256),
                max_model_len=getattr(args, "max_model_len", 8192),
                speculative_ngram=getattr(args, "speculative_ngram", 0),
                speculative_k=(
                    getattr(args, "num_speculative_tokens", 3)
                    if draft_cfg is not None
                    else 0
                ),
            ),
            draft_model=draft_cfg,
            draft_model_path=draft_dir,
            parallel=ParallelConfig(
                tensor_parallel_size=getattr(args, "tensor_parallel_size", 1),
                pipeline_parallel_size=getattr(args, "pipeline_parallel_size", 1),
            ),
            model_path=args.model_dir,
            device="cuda" if torch.cuda.is_available() else "cpu",
        )
        lora_modules = {}
        if getattr(args, "enable_lora", False) or getattr(args, "lora_modules", None):
            for spec in getattr(args, "lora_modules", []) or []:
                name, _, path = spec.partition("=")
                if not path:
                    raise ValueError(f"--lora-modules expects name=path, got {spec!r}")
                lora_modules[name] = path
        return LLMModel(
            args.model_name, cfg, tokenizer=tokenizer, lora_modules=lora_modules
        )
    if backend == "hf":
        from kserve_amd.runtimes.hf_generative import HFGenerativeModel

        model = HFGenerativeModel(args.model_name, model_dir=args.model_dir)
        model.load()
        return model
    from kserve_amd.runtimes.encoder_model import EncoderModel

    model = EncoderModel(
        args.model_name, args.model_dir, task=getattr(args, "task", None)
    )
    model.load()
    return model


def main(argv=None):
    from kserve_amd.model_server import ModelServer, build_arg_parser

    parser = build_arg_parser()
    parser.add_argument("--backend", default=None, choices=["engine", "encoder", "hf"])
    parser.add_argument("--task", default=None)
    parser.add_argument("--tensor-parallel-size", dest="tensor_parallel_size", type=int, default=1)
    parser.add_argument("--pipeline-parallel-size", dest="pipeline_parallel_size", type=int, default=1)
    parser.add_argument("--max_model_len", type=int, default=8192)
    parser.add_argument("--max_num_seqs", type=int, default=256)
    # data-parallel contract flags (reference preset
    # config-llm-worker-data-parallel.yaml:188-199)
    parser.add_argument("--data-parallel-size", dest="data_parallel_size", type=int, default=1)
    parser.add_argument("--data-parallel-size-local", dest="data_parallel_size_local", type=int, default=None)
    parser.add_argument("--data-parallel-rpc-port", dest="data_parallel_rpc_port", type=int, default=5555)
    parser.add_argument("--data-parallel-address", dest="data_parallel_address", default="127.0.0.1")
    parser.add_argument("--data-parallel-start-rank", dest="data_parallel_start_rank", type=int, default=0)
    parser.add_argument("--enable-expert-parallel", dest="expert_parallel", action="store_true")
    # host-DRAM KV offload tier (LLMInferenceService KVCacheOffloadingSpec)
    parser.add_argument("--kv-offload-bytes", dest="kv_offload_bytes", type=int, default=0)
    parser.add_argument("--kv-cache-dtype", dest="kv_cache_dtype", default="auto",
                        choices=["auto", "fp8"])
    # LoRA adapter serving (reference: --enable-lora --lora-modules name=path)
    parser.add_argument("--speculative-draft-model",
                        dest="speculative_draft_model", default=None,
                        help="HF dir of a small draft model (draft-model "
                             "speculation; weights loaded from its "
                             "safetensors)")
    parser.add_argument("--num-speculative-tokens",
                        dest="num_speculative_tokens", type=int, default=3)
    parser.add_argument("--speculative-ngram", dest="speculative_ngram",
                        type=int, default=0)
    parser.add_argument("--enable-lora", dest="enable_lora", action="store_true")
    parser.add_argument(
        "--lora-modules", dest="lora_modules", nargs="*", default=[],
        help="name=path pairs; each name becomes a served model id",
    )
    args = parser.parse_args(argv)
    configure_logging()
    model = build_model(args)
    ModelServer(
        http_port=args.http_port,
        grpc_port=args.grpc_port,
        enable_grpc=args.enable_grpc,
    ).start([model])


if __name__ == "__main__":
    main()
