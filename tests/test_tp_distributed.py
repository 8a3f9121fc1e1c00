"""Tensor-parallel correctness without GPUs: gloo backend, world_size=2.

The TP=2 sharded model must produce the same logits as the TP=1 model with
identical weights (SURVEY.md §7 hard part #7: mock-RCCL/CPU harness the
reference never had)."""

import multiprocessing as mp
import os

import numpy as np
import pytest
import torch


def _set_env(rank, world, port):
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)


def _tp2_linear_worker(rank, world, port, q):
    try:
        _set_env(rank, world, port)
        from kserve_amd.parallel import comm
        from kserve_amd.parallel.layers import (
            ColumnParallelLinear,
            RowParallelLinear,
        )

        comm.init_distributed(tp_size=2, backend="gloo")
        torch.manual_seed(0)
        full_w1 = torch.randn(32, 16)
        full_w2 = torch.randn(16, 32)
        x = torch.randn(4, 16)

        col = ColumnParallelLinear(16, 32, dtype=torch.float32, gather_output=True)
        col.load_shard(full_w1)
        row = RowParallelLinear(32, 16, dtype=torch.float32)
        row.load_shard(full_w2)

        y = col(x)
        ref_y = x @ full_w1.t()
        torch.testing.assert_close(y, ref_y, atol=1e-5, rtol=1e-5)

        # row-parallel consumes the col-sharded (ungathered) activation
        col2 = ColumnParallelLinear(16, 32, dtype=torch.float32, gather_output=False)
        col2.load_shard(full_w1)
        z = row(col2(x))
        ref_z = (x @ full_w1.t()) @ full_w2.t()
        torch.testing.assert_close(z, ref_z, atol=1e-4, rtol=1e-4)
        comm.destroy_distributed()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        q.put((rank, f"FAIL: {type(e).__name__}: {e}"))


def _tp2_llama_worker(rank, world, port, q):
    try:
        _set_env(rank, world, port)
        from kserve_amd.engine.config import ModelConfig
        from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM
        from kserve_amd.parallel import comm

        comm.init_distributed(tp_size=2, backend="gloo")
        cfg = ModelConfig.tiny(vocab_size=128)
        torch.manual_seed(7)
        # build the full reference state dict once (same on both ranks)
        import transformers

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
        caches = [(torch.empty(0), torch.empty(0))] * cfg.num_layers
        hidden = model(
            torch.tensor(token_ids), torch.arange(T), caches, meta
        )
        logits = model.compute_logits(hidden)
        with torch.no_grad():
            ref = hf(torch.tensor([token_ids]), use_cache=False).logits[0]
        torch.testing.assert_close(logits, ref, atol=5e-4, rtol=5e-4)
        comm.destroy_distributed()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def _run_workers(fn, world=2, port=29611):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=fn, args=(r, world, port, q)) for r in range(world)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=180) for _ in range(world)]
    for p in procs:
        p.join(timeout=60)
        if p.is_alive():
            p.terminate()
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


@pytest.mark.timeout(300)
def test_tp2_parallel_linears():
    _run_workers(_tp2_linear_worker, port=29611)


@pytest.mark.timeout(300)
def test_tp2_llama_matches_hf():
    pytest.importorskip("transformers")
    _run_workers(_tp2_llama_worker, port=29613)


def _tp2_engine_worker(rank, world, port, q):
    try:
        _set_env(rank, world, port)
        import torch

        from kserve_amd.engine.config import (
            CacheConfig,
            EngineConfig,
            ModelConfig,
            SchedulerConfig,
        )
        from kserve_amd.engine.engine import LLMEngine
        from kserve_amd.engine.sampling_params import SamplingParams
        from kserve_amd.parallel import comm

        comm.init_distributed(tp_size=2, backend="gloo")
        cfg = EngineConfig(
            model=ModelConfig.tiny(vocab_size=128),
            cache=CacheConfig(block_size=4, num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=128, max_model_len=64
            ),
            device="cpu",
            seed=7,
            eos_token_id=-1,
        )
        engine = LLMEngine(cfg)
        out = engine.generate(
            [[1, 2, 3], [9, 8, 7, 6]],
            SamplingParams(temperature=0.0, max_tokens=6),
        )
        toks = [o.output_token_ids for o in out.values()]
        comm.destroy_distributed()
        q.put((rank, ("ok", toks)))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, (f"FAIL: {e}\n{traceback.format_exc()}", None)))


@pytest.mark.timeout(300)
def test_tp2_engine_generate_consistent():
    """TP=2 engine must produce identical greedy tokens on both ranks.

    (Random-init weights are seeded per-rank via tp_rank offsets in
    random_init — ranks hold different SHARDS of one logical model, so the
    all-reduced logits and therefore tokens must agree across ranks.)"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_tp2_engine_worker, args=(r, 2, 29617, q))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, (status, toks) = q.get(timeout=240)
        assert status == "ok", f"rank {rank}: {status}"
        results[rank] = toks
    for p in procs:
        p.join(timeout=60)
    assert results[0] == results[1]
    assert all(len(t) == 6 for t in results[0])


def _tp2_overlap_worker(rank, world, port, q):
    try:
        _set_env(rank, world, port)
        from kserve_amd.parallel import comm
        from kserve_amd.parallel.layers import RowParallelLinear

        comm.init_distributed(tp_size=2, backend="gloo")
        torch.manual_seed(1)
        full_w = torch.randn(64, 32)
        x = torch.randn(16, 16)  # this rank's shard input

        sync = RowParallelLinear(32, 64, dtype=torch.float32,
                                 overlap_chunks=1)
        sync.load_shard(full_w)
        over = RowParallelLinear(32, 64, dtype=torch.float32,
                                 overlap_chunks=2)
        over.load_shard(full_w)
        over.OVERLAP_MIN_NUMEL = 1  # force the overlap path at test sizes
        y_sync = sync(x)
        y_over = over(x)
        torch.testing.assert_close(y_over, y_sync, atol=1e-5, rtol=1e-5)
        # odd chunking still exact
        over3 = RowParallelLinear(32, 64, dtype=torch.float32,
                                  overlap_chunks=3)
        over3.load_shard(full_w)
        over3.OVERLAP_MIN_NUMEL = 1
        torch.testing.assert_close(over3(x), y_sync, atol=1e-5, rtol=1e-5)
        comm.destroy_distributed()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


@pytest.mark.timeout(300)
def test_tp2_overlapped_all_reduce_matches_sync():
    """Chunked async all-reduce (comm/compute overlap) must be numerically
    identical to the synchronous reduction."""
    _run_workers(_tp2_overlap_worker, port=29617)
