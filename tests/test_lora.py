"""Multi-LoRA serving: adapter loading, per-request application, equivalence
with merged weights (reference: huggingfaceserver --enable-lora/--lora-modules,
vllm LoRA request routing — __main__.py:334-337)."""

import json
import os

import pytest
import torch

from kserve_amd.engine.config import (
    CacheConfig,
    EngineConfig,
    ModelConfig,
    SchedulerConfig,
)
from kserve_amd.engine.engine import LLMEngine
from kserve_amd.engine.sampling_params import SamplingParams

RANK = 4
ALPHA = 8.0


def make_adapter_dir(tmp_path, cfg: ModelConfig, seed=123, modules=None):
    """Write a synthetic PEFT adapter for the tiny model."""
    from safetensors.torch import save_file

    modules = modules or ["q_proj", "v_proj", "o_proj", "down_proj", "gate_proj"]
    gen = torch.Generator().manual_seed(seed)
    tensors = {}
    dims = {
        "q_proj": (cfg.num_heads * cfg.head_dim, cfg.hidden_size),
        "k_proj": (cfg.num_kv_heads * cfg.head_dim, cfg.hidden_size),
        "v_proj": (cfg.num_kv_heads * cfg.head_dim, cfg.hidden_size),
        "o_proj": (cfg.hidden_size, cfg.num_heads * cfg.head_dim),
        "gate_proj": (cfg.intermediate_size, cfg.hidden_size),
        "up_proj": (cfg.intermediate_size, cfg.hidden_size),
        "down_proj": (cfg.hidden_size, cfg.intermediate_size),
    }
    for i in range(cfg.num_layers):
        for m in modules:
            out, inp = dims[m]
            parent = "self_attn" if "proj" in m and m[0] in "qkvo" else "mlp"
            base = f"base_model.model.model.layers.{i}.{parent}.{m}"
            tensors[f"{base}.lora_A.weight"] = (
                torch.randn(RANK, inp, generator=gen) * 0.05
            )
            tensors[f"{base}.lora_B.weight"] = (
                torch.randn(out, RANK, generator=gen) * 0.05
            )
    d = tmp_path / "adapter"
    d.mkdir()
    save_file(tensors, str(d / "adapter_model.safetensors"))
    (d / "adapter_config.json").write_text(
        json.dumps({"r": RANK, "lora_alpha": ALPHA, "target_modules": modules})
    )
    return str(d), tensors


def make_engine(**kw):
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
        **kw,
    )
    return LLMEngine(cfg)


def test_adapter_loading(tmp_path):
    torch.manual_seed(0)
    engine = make_engine()
    path, tensors = make_adapter_dir(tmp_path, engine.config.model)
    aid = engine.register_lora("my-adapter", path)
    assert aid == 1
    mgr = engine.lora_manager
    w = mgr.layer_weights(aid, 0, "q_proj")
    assert w is not None and w.a.shape[0] == RANK
    assert w.scale == ALPHA / RANK
    assert mgr.layer_weights(aid, 0, "up_proj") is None  # not targeted
    # idempotent re-register
    assert engine.register_lora("my-adapter", path) == aid


def test_lora_changes_output_and_matches_merged(tmp_path):
    """Runtime LoRA == merging B@A*scale into the base weights."""
    torch.manual_seed(0)
    engine = make_engine()
    path, tensors = make_adapter_dir(tmp_path, engine.config.model)
    engine.register_lora("adapt", path)
    prompts = [[1, 2, 3, 4, 5], [9, 8, 7]]
    sp_base = SamplingParams(temperature=0.0, max_tokens=6)
    sp_lora = SamplingParams(temperature=0.0, max_tokens=6, lora_name="adapt")
    base_out = [o.output_token_ids for o in engine.generate(prompts, sp_base).values()]
    lora_out = [o.output_token_ids for o in engine.generate(prompts, sp_lora).values()]
    assert base_out != lora_out  # the adapter must change generations

    # merged-weight engine: same seed => same base weights; fold BA in
    torch.manual_seed(0)
    merged = make_engine()
    scale = ALPHA / RANK
    with torch.no_grad():
        for i, layer in enumerate(merged.model.layers):
            pre = f"base_model.model.model.layers.{i}."
            for mod, target, row_off in [
                ("self_attn.q_proj", layer.self_attn.qkv_proj, 0),
                (
                    "self_attn.v_proj",
                    layer.self_attn.qkv_proj,
                    layer.self_attn.qkv_proj.q_size
                    + layer.self_attn.qkv_proj.kv_size,
                ),
                ("self_attn.o_proj", layer.self_attn.o_proj, 0),
                ("mlp.gate_proj", layer.mlp.gate_up_proj, 0),
                ("mlp.down_proj", layer.mlp.down_proj, 0),
            ]:
                a = tensors[pre + mod + ".lora_A.weight"]
                b = tensors[pre + mod + ".lora_B.weight"]
                delta = (b @ a) * scale
                w = target.weight.data
                w[row_off : row_off + delta.shape[0]] += delta.to(w.dtype)
    merged_out = [
        o.output_token_ids for o in merged.generate(prompts, sp_base).values()
    ]
    assert lora_out == merged_out


def test_mixed_batch_isolation(tmp_path):
    """Base and LoRA requests in one batch: base rows must be unaffected."""
    torch.manual_seed(0)
    engine = make_engine()
    path, _ = make_adapter_dir(tmp_path, engine.config.model)
    engine.register_lora("adapt", path)
    sp_base = SamplingParams(temperature=0.0, max_tokens=5)
    prompts = [[1, 2, 3], [4, 5, 6]]
    pure = [o.output_token_ids for o in engine.generate(prompts, sp_base).values()]
    # same prompts, one with the adapter, submitted together
    rid_a = engine.add_request(prompts[0], sp_base)
    rid_b = engine.add_request(
        prompts[1], SamplingParams(temperature=0.0, max_tokens=5, lora_name="adapt")
    )
    outs = {}
    while engine.has_unfinished():
        for out in engine.step():
            if out.finished:
                outs[out.request_id] = out.output_token_ids
    assert outs[rid_a] == pure[0]
    assert outs[rid_b] != pure[1]


def test_unknown_adapter_rejected(tmp_path):
    torch.manual_seed(0)
    engine = make_engine()
    with pytest.raises(ValueError):
        engine.add_request(
            [1, 2, 3], SamplingParams(lora_name="nope", max_tokens=2)
        )


def test_lora_served_as_model_name(tmp_path):
    """Adapter names are routed through /v1/completions and /v1/models
    (reference registers each --lora-modules name with the model server)."""
    from fastapi.testclient import TestClient

    from kserve_amd.model_repository import ModelRepository
    from kserve_amd.protocol.dataplane import DataPlane
    from kserve_amd.protocol.rest.openai.endpoints import register_openai_endpoints
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.runtimes.llm_model import LLMModel

    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
    )
    path, _ = make_adapter_dir(tmp_path, cfg.model)
    model = LLMModel("tiny", cfg, lora_modules={"tiny-sql": path})
    repo = ModelRepository()
    repo.update(model)
    dp = DataPlane(repo)
    app = create_app(dp)
    register_openai_endpoints(app, dp, [model])

    with TestClient(app) as c:
        import asyncio

        # start engine manually (TestClient doesn't run ModelServer lifecycle)
        asyncio.new_event_loop().run_until_complete(model.start_engine())

        names = {m["id"] for m in c.get("/openai/v1/models").json()["data"]}
        assert {"tiny", "tiny-sql"} <= names

        base = c.post(
            "/openai/v1/completions",
            json={"model": "tiny", "prompt": [1, 2, 3, 4], "max_tokens": 5,
                  "temperature": 0.0},
        )
        assert base.status_code == 200, base.text
        lora = c.post(
            "/openai/v1/completions",
            json={"model": "tiny-sql", "prompt": [1, 2, 3, 4], "max_tokens": 5,
                  "temperature": 0.0},
        )
        assert lora.status_code == 200, lora.text
        assert base.json()["choices"][0]["text"] != lora.json()["choices"][0]["text"]
        model.stop()


def test_lora_with_chunked_prefill(tmp_path):
    """LoRA deltas must apply identically when the prompt is chunked
    (paged-context attention path)."""
    torch.manual_seed(0)
    full = make_engine()
    path, _ = make_adapter_dir(tmp_path, full.config.model)
    full.register_lora("adapt", path)
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=4,
            max_num_batched_tokens=8,  # chunks the 19-token prompt
            max_model_len=128,
            enable_chunked_prefill=True,
        ),
        device="cpu",
        eos_token_id=-1,
    )
    chunked = LLMEngine(cfg)
    chunked.register_lora("adapt", path)
    sp = SamplingParams(temperature=0.0, max_tokens=6, lora_name="adapt")
    prompts = [list(range(1, 20))]
    a = [o.output_token_ids for o in full.generate(prompts, sp).values()]
    b = [o.output_token_ids for o in chunked.generate(prompts, sp).values()]
    assert a == b
