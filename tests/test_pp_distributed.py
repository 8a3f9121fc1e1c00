"""Pipeline-parallel correctness without GPUs: gloo backend, world_size=2.

PP=2 splits the layer stack across two processes (stage 0: embeddings +
first half; stage 1: second half + norm + lm_head), activations travel by
p2p send/recv and sampled tokens broadcast back so both schedulers stay
in lockstep. Greedy outputs must match the single-process engine exactly.
(Reference parity: WorkerSpec.PipelineParallelSize / ParallelismSpec.Pipeline,
SURVEY.md §2.5 — vLLM PP over Ray there; RCCL p2p here.)
"""

import multiprocessing as mp
import os

import pytest
import torch


def _set_env(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)


def _hf_weights(cfg):
    """Deterministic full-model weights (same in every process; random_init
    draws differ per PP stage, so explicit weights are required)."""
    import transformers

    torch.manual_seed(7)
    hf_cfg = transformers.LlamaConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        intermediate_size=cfg.intermediate_size,
        num_hidden_layers=cfg.num_layers,
        num_attention_heads=cfg.num_heads,
        num_key_value_heads=cfg.num_kv_heads,
        rms_norm_eps=cfg.rms_norm_eps,
        rope_theta=cfg.rope_theta,
        tie_word_embeddings=False,
    )
    hf = transformers.LlamaForCausalLM(hf_cfg).eval().float()
    return dict(hf.state_dict())


def _engine_config():
    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )

    return EngineConfig(
        model=ModelConfig.tiny(vocab_size=128),
        cache=CacheConfig(block_size=4, num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
        device="cpu",
        eos_token_id=-1,
    )


def _build_engine():
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.models.llama import LlamaForCausalLM

    cfg = _engine_config()
    model = LlamaForCausalLM(cfg.model, dtype=torch.float32, device="cpu")
    model.load_hf_state_dict(_hf_weights(cfg.model))
    return LLMEngine(cfg, model=model)


def _reference_outputs():
    """Single-process greedy outputs with the shared weights."""
    from kserve_amd.engine.sampling_params import SamplingParams

    engine = _build_engine()
    sp = SamplingParams(temperature=0.0, max_tokens=8)
    out = engine.generate([[1, 2, 3, 4, 5], [9, 8, 7]], sp)
    toks = [o.output_token_ids for o in out.values()]
    # mirror the workers' second (chunked-prefill) run
    engine.config.scheduler.enable_chunked_prefill = True
    engine.scheduler.config.max_num_batched_tokens = 4
    out2 = engine.generate([list(range(1, 12))], sp)
    toks.append(list(out2.values())[0].output_token_ids)
    return toks


def _pp2_engine_worker(rank, world, port, q):
    try:
        _set_env(rank, world, port)
        from kserve_amd.engine.sampling_params import SamplingParams
        from kserve_amd.parallel import comm

        comm.init_distributed(tp_size=1, pp_size=2, backend="gloo")
        st = comm.get_state()
        assert st.pp_size == 2 and st.tp_size == 1
        engine = _build_engine()
        cfg = engine.config
        # stage layer partition sanity
        assert engine.model.num_local_layers == cfg.model.num_layers // 2
        assert (engine.model.embed_tokens is not None) == (rank == 0)
        assert (engine.model.lm_head is not None) == (rank == 1)
        sp = SamplingParams(temperature=0.0, max_tokens=8)
        out = engine.generate([[1, 2, 3, 4, 5], [9, 8, 7]], sp)
        toks = [o.output_token_ids for o in out.values()]
        # chunked prefill under PP: partial chunks sample nothing on any
        # stage; logits slicing must tolerate None on non-final stages
        engine.config.scheduler.enable_chunked_prefill = True
        engine.scheduler.config.max_num_batched_tokens = 4
        out2 = engine.generate([list(range(1, 12))], sp)
        toks.append(list(out2.values())[0].output_token_ids)
        comm.destroy_distributed()
        q.put((rank, ("ok", toks)))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, (f"FAIL: {e}\n{traceback.format_exc()}", None)))


def _pp2_weights_worker(rank, world, port, q):
    """PP=2 forward logits (last stage) vs HF reference with REAL weights."""
    try:
        _set_env(rank, world, port)
        import transformers

        from kserve_amd.engine.config import ModelConfig
        from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM
        from kserve_amd.parallel import comm

        comm.init_distributed(tp_size=1, pp_size=2, backend="gloo")
        cfg = ModelConfig.tiny(vocab_size=128)
        torch.manual_seed(7)
        hf_cfg = transformers.LlamaConfig(
            vocab_size=cfg.vocab_size,
            hidden_size=cfg.hidden_size,
            intermediate_size=cfg.intermediate_size,
            num_hidden_layers=cfg.num_layers,
            num_attention_heads=cfg.num_heads,
            num_key_value_heads=cfg.num_kv_heads,
            rms_norm_eps=cfg.rms_norm_eps,
            rope_theta=cfg.rope_theta,
            tie_word_embeddings=False,
        )
        hf = transformers.LlamaForCausalLM(hf_cfg).eval().float()
        sd = dict(hf.state_dict())
        model = LlamaForCausalLM(cfg, dtype=torch.float32, device="cpu")
        model.load_hf_state_dict(sd)
        token_ids = list(range(10))
        T = len(token_ids)
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.zeros(T, dtype=torch.int32),
            cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
            max_seqlen=T,
        )
        caches = [(torch.empty(0), torch.empty(0))] * model.num_local_layers
        hidden = model(torch.tensor(token_ids), torch.arange(T), caches, meta)
        logits = model.compute_logits(hidden)
        if rank == world - 1:
            with torch.no_grad():
                ref = hf(torch.tensor([token_ids]), use_cache=False).logits[0]
            torch.testing.assert_close(logits, ref, atol=5e-4, rtol=5e-4)
        else:
            assert logits is None
        comm.destroy_distributed()
        q.put((rank, ("ok", None)))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, (f"FAIL: {e}\n{traceback.format_exc()}", None)))


def _run(fn, port, expect_tokens=None):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=fn, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=240) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    toks_by_rank = {}
    for rank, (status, toks) in results:
        assert status == "ok", f"rank {rank}: {status}"
        toks_by_rank[rank] = toks
    return toks_by_rank


@pytest.mark.timeout(300)
def test_pp2_llama_matches_hf():
    _run(_pp2_weights_worker, 29621)


@pytest.mark.timeout(300)
def test_pp2_engine_matches_single_process():
    ref = _reference_outputs()
    toks = _run(_pp2_engine_worker, 29623)
    # every stage's lockstep scheduler must report the same tokens, equal
    # to the single-process engine
    assert toks[0] == toks[1] == ref


def _ep2_worker(rank, world, port, q):
    """Expert-parallel Mixtral (tp group=2, 2 experts/rank) vs HF."""
    try:
        _set_env(rank, world, port)
        import transformers

        from kserve_amd.engine.config import ModelConfig
        from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM
        from kserve_amd.parallel import comm

        comm.init_distributed(tp_size=2, backend="gloo")
        torch.manual_seed(21)
        hf_cfg = transformers.MixtralConfig(
            vocab_size=128,
            hidden_size=64,
            intermediate_size=128,
            num_hidden_layers=2,
            num_attention_heads=4,
            num_key_value_heads=2,
            num_local_experts=4,
            num_experts_per_tok=2,
            rms_norm_eps=1e-5,
            rope_theta=10000.0,
            sliding_window=None,
            tie_word_embeddings=False,
        )
        hf = transformers.MixtralForCausalLM(hf_cfg).eval().float()
        cfg = ModelConfig(
            vocab_size=128, hidden_size=64, intermediate_size=128,
            num_layers=2, num_heads=4, num_kv_heads=2, head_dim=16,
            rms_norm_eps=1e-5, rope_theta=10000.0,
            num_local_experts=4, num_experts_per_tok=2,
            expert_parallel=True, model_name="mixtral-ep",
        )
        model = LlamaForCausalLM(cfg, dtype=torch.float32, device="cpu")
        model.load_hf_state_dict(dict(hf.state_dict()))
        moe = model.layers[0].mlp
        assert moe.expert_parallel and moe.experts_per_rank == 2
        assert moe.expert_lo == rank * 2
        token_ids = list(range(9))
        T = len(token_ids)
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.zeros(T, dtype=torch.int32),
            cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
            max_seqlen=T,
        )
        caches = [(torch.empty(0), torch.empty(0))] * model.num_local_layers
        hidden = model(torch.tensor(token_ids), torch.arange(T), caches, meta)
        logits = model.compute_logits(hidden)
        with torch.no_grad():
            ref = hf(torch.tensor([token_ids]), use_cache=False).logits[0]
        torch.testing.assert_close(logits, ref, atol=5e-4, rtol=5e-4)
        comm.destroy_distributed()
        q.put((rank, ("ok", None)))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, (f"FAIL: {e}\n{traceback.format_exc()}", None)))


@pytest.mark.timeout(300)
def test_ep2_mixtral_matches_hf():
    """Expert parallelism: experts partitioned across ranks, outputs summed
    by the block all-reduce (reference --enable-expert-parallel)."""
    _run(_ep2_worker, 29627)
