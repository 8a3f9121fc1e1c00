"""GPU engine integration: full native path (HIP kernels + hipGraphs)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _cfg(enforce_eager=False, num_blocks=128):
    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )

    return EngineConfig(
        model=ModelConfig(
            vocab_size=2048,
            hidden_size=512,
            intermediate_size=1024,
            num_layers=4,
            num_heads=4,
            num_kv_heads=2,
            head_dim=128,
            max_position_embeddings=1024,
            model_name="gpu-tiny",
        ),
        cache=CacheConfig(block_size=16, num_gpu_blocks=num_blocks),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=2048, max_model_len=512
        ),
        device="cuda",
        seed=0,
        eos_token_id=-1,
        enforce_eager=enforce_eager,
    )


def test_engine_generates_eager():
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    engine = LLMEngine(_cfg(enforce_eager=True))
    out = engine.generate(
        [[1, 2, 3, 4], [10, 11, 12, 13, 14, 15]],
        SamplingParams(temperature=0.0, max_tokens=12),
    )
    assert len(out) == 2
    for o in out.values():
        assert len(o.output_token_ids) == 12


def test_graph_matches_eager():
    """hipGraph-captured decode must produce the same greedy tokens as eager."""
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    prompts = [[1, 2, 3, 4, 5], [100, 200, 300], [7, 8, 9, 10]]
    sp = SamplingParams(temperature=0.0, max_tokens=16)

    engine_e = LLMEngine(_cfg(enforce_eager=True))
    out_e = engine_e.generate(prompts, sp)
    del engine_e
    torch.cuda.empty_cache()

    engine_g = LLMEngine(_cfg(enforce_eager=False))
    out_g = engine_g.generate(prompts, sp)
    toks_e = [o.output_token_ids for o in out_e.values()]
    toks_g = [o.output_token_ids for o in out_g.values()]
    assert toks_e == toks_g


def test_logits_match_cpu_reference():
    """GPU bf16 prefill logits vs CPU fp32 reference of the same weights."""
    from kserve_amd.engine.config import ModelConfig
    from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM

    cfg = ModelConfig(
        vocab_size=512,
        hidden_size=256,
        intermediate_size=512,
        num_layers=2,
        num_heads=2,
        num_kv_heads=1,
        head_dim=128,
        max_position_embeddings=256,
    )
    torch.manual_seed(0)
    cpu_model = LlamaForCausalLM(cfg, dtype=torch.float32, device="cpu")
    cpu_model.random_init(seed=5)
    gpu_model = LlamaForCausalLM(cfg, dtype=torch.bfloat16, device="cuda")
    # copy weights
    with torch.no_grad():
        for (n1, p1), (n2, p2) in zip(
            gpu_model.named_parameters(), cpu_model.named_parameters()
        ):
            assert n1 == n2
            p1.copy_(p2.to(torch.bfloat16))

    T = 33
    ids = torch.randint(0, cfg.vocab_size, (T,))
    pos = torch.arange(T)

    def run(model, device):
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.zeros(T, dtype=torch.int32, device=device),
            cu_seqlens=torch.tensor([0, T], dtype=torch.int32, device=device),
            max_seqlen=T,
        )
        caches = [
            (
                torch.zeros(8, cfg.num_kv_heads, 16, 128, dtype=model.dtype, device=device),
                torch.zeros(8, cfg.num_kv_heads, 16, 128, dtype=model.dtype, device=device),
            )
            for _ in range(cfg.num_layers)
        ]
        # slot mapping 0..T-1 valid (block 0..2 exist)
        meta.slot_mapping = torch.arange(T, dtype=torch.int32, device=device)
        hidden = model(ids.to(device), pos.to(device), caches, meta)
        return model.compute_logits(hidden)

    ref = run(cpu_model, "cpu")
    got = run(gpu_model, "cuda")
    # bf16 end-to-end tolerance: compare top-1 agreement + correlation
    agree = (got.float().cpu().argmax(-1) == ref.argmax(-1)).float().mean()
    assert agree > 0.9, f"top-1 agreement {agree}"
    cos = torch.nn.functional.cosine_similarity(
        got.float().cpu().flatten(), ref.flatten(), dim=0
    )
    assert cos > 0.99, f"cosine {cos}"


def test_preemption_on_gpu():
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    engine = LLMEngine(_cfg(enforce_eager=True, num_blocks=20))
    out = engine.generate(
        [[i, i + 1, i + 2, i + 3] for i in range(6)],
        SamplingParams(temperature=0.0, max_tokens=30),
    )
    assert len(out) == 6
    for o in out.values():
        assert len(o.output_token_ids) == 30


def test_bert_fill_mask_gpu_matches_cpu():
    """BERT encoder path on GPU bf16 (layer_norm/gelu/bidirectional flash
    HIP kernels) vs CPU fp32 reference."""
    import pytest

    transformers = pytest.importorskip("transformers")
    from kserve_amd.models.bert import BertConfig, BertForMaskedLM

    cfg = transformers.BertConfig(
        vocab_size=512,
        hidden_size=128,
        num_hidden_layers=2,
        num_attention_heads=2,  # head_dim 64
        intermediate_size=256,
        max_position_embeddings=128,
    )
    torch.manual_seed(0)
    hf = transformers.BertForMaskedLM(cfg).eval().float()
    sd = dict(hf.state_dict())
    ours_cpu = BertForMaskedLM(BertConfig.tiny(), dtype=torch.float32)
    ours_cpu.load_hf_state_dict(sd)
    ours_gpu = BertForMaskedLM(BertConfig.tiny(), dtype=torch.bfloat16, device="cuda")
    ours_gpu.load_hf_state_dict(sd)
    ours_gpu = ours_gpu.to("cuda")

    ids = torch.randint(0, 512, (24,))
    cu = torch.tensor([0, 10, 24], dtype=torch.int32)
    ref = ours_cpu(ids, cu)
    got = ours_gpu(ids.cuda(), cu.cuda())
    agree = (got.float().cpu().argmax(-1) == ref.argmax(-1)).float().mean()
    assert agree > 0.9, f"top-1 agreement {agree}"


def test_kv_offload_swap_on_gpu():
    """Swap-based preemption over pinned host DRAM (hipMemcpyAsync side
    stream) preserves greedy outputs on the GPU path."""
    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    def cfg(num_blocks, cpu_blocks):
        return EngineConfig(
            model=ModelConfig(
                vocab_size=2048,
                hidden_size=512,
                intermediate_size=1024,
                num_layers=4,
                num_heads=4,
                num_kv_heads=2,
                head_dim=128,
                max_position_embeddings=1024,
            ),
            cache=CacheConfig(
                block_size=16, num_gpu_blocks=num_blocks, num_cpu_blocks=cpu_blocks
            ),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=2048, max_model_len=512
            ),
            device="cuda",
            seed=0,
            eos_token_id=-1,
            enforce_eager=True,
        )

    # 1) KV pages round-trip bit-exactly through the pinned host tier
    engine = LLMEngine(cfg(32, 32))
    runner = engine.runner
    for layer, (gk, gv) in enumerate(runner.kv_caches):
        gk[3].normal_()
        gv[3].normal_()
    snap_k = [gk[3].clone() for gk, _ in runner.kv_caches]
    snap_v = [gv[3].clone() for _, gv in runner.kv_caches]
    runner.swap_blocks([(3, 5)], to_gpu=False)   # gpu block 3 -> cpu block 5
    for gk, gv in runner.kv_caches:
        gk[3].zero_()
        gv[3].zero_()
    runner.swap_blocks([(5, 7)], to_gpu=True)    # cpu block 5 -> gpu block 7
    torch.cuda.synchronize()
    for layer, (gk, gv) in enumerate(runner.kv_caches):
        assert torch.equal(gk[7], snap_k[layer])
        assert torch.equal(gv[7], snap_v[layer])
    del engine
    torch.cuda.empty_cache()

    # 2) engine-level: swaps occur and all requests complete (exact token
    # equality vs an unconstrained engine is not required on GPU: hipBLASLt
    # picks batch-size-dependent algorithms, so bf16 greedy outputs can
    # legitimately differ when preemption changes batch composition)
    prompts = [[i + 1, i + 2, i + 3, i + 4] for i in range(3)]
    sp = SamplingParams(temperature=0.0, max_tokens=60)
    small = LLMEngine(cfg(10, 64))
    swaps = {"n": 0}
    orig = small.scheduler.schedule

    def counting():
        b = orig()
        swaps["n"] += len(b.swap_out)
        return b

    small.scheduler.schedule = counting
    got = small.generate(prompts, sp)
    assert len(got) == 3
    for o in got.values():
        assert len(o.output_token_ids) == 60
    assert swaps["n"] > 0


def test_chunked_prefill_on_gpu():
    """Chunked prefill runs the native paged-context kernel end-to-end.
    hipBLASLt bf16 rounding varies with GEMM row count, so later greedy
    tokens can diverge; the FIRST sampled token goes through 4 layers of
    full-context attention and must agree with the full-prefill engine."""
    from kserve_amd.engine.config import SchedulerConfig
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    prompts = [list(range(1, 100)), list(range(200, 250)), [5, 6, 7]]
    sp = SamplingParams(temperature=0.0, max_tokens=8)

    cfg = _cfg(enforce_eager=True)
    torch.manual_seed(0)
    full = LLMEngine(cfg)
    a = full.generate(prompts, sp)
    del full
    torch.cuda.empty_cache()

    cfg2 = _cfg(enforce_eager=True)
    cfg2.scheduler = SchedulerConfig(
        max_num_seqs=8,
        max_num_batched_tokens=32,  # forces 99-token prompt into 4 chunks
        max_model_len=512,
        enable_chunked_prefill=True,
    )
    torch.manual_seed(0)
    chunked = LLMEngine(cfg2)
    b = chunked.generate(prompts, sp)

    outs_a = [o.output_token_ids for o in a.values()]
    outs_b = [o.output_token_ids for o in b.values()]
    for ta, tb in zip(outs_a, outs_b):
        assert len(tb) == 8
        # bf16 GEMM rounding differs with row-batch shape; the first token of
        # each completion is far from any tie for random weights and must match
        assert ta[0] == tb[0], (ta, tb)


def test_lora_on_gpu(tmp_path):
    """LoRA requests run the eager decode path on GPU and match the
    merged-weight engine's first token."""
    from tests.test_lora import make_adapter_dir
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    cfg = _cfg(enforce_eager=True)
    torch.manual_seed(0)
    engine = LLMEngine(cfg)
    path, _ = make_adapter_dir(tmp_path, cfg.model)
    engine.register_lora("adapt", path)
    prompts = [[1, 2, 3, 4, 5]]
    base = list(engine.generate(prompts, SamplingParams(temperature=0.0, max_tokens=6)).values())[0]
    lora = list(
        engine.generate(
            prompts, SamplingParams(temperature=0.0, max_tokens=6, lora_name="adapt")
        ).values()
    )[0]
    assert len(lora.output_token_ids) == 6
    assert base.output_token_ids != lora.output_token_ids


def test_spec_decode_on_gpu():
    """Spec decode verify runs the paged-context kernel on GPU; greedy
    outputs must match the plain hipGraph decode engine exactly (both paths
    sample argmax from the same bf16 logits)."""
    from kserve_amd.engine.config import SchedulerConfig
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    prompts = [[1, 2, 3, 4, 1, 2, 3, 4, 1, 2], [7, 8, 9]]
    sp = SamplingParams(temperature=0.0, max_tokens=20)

    torch.manual_seed(0)
    plain = LLMEngine(_cfg(enforce_eager=True))
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    del plain
    torch.cuda.empty_cache()

    cfg = _cfg(enforce_eager=True)
    cfg.scheduler = SchedulerConfig(
        max_num_seqs=8, max_num_batched_tokens=2048, max_model_len=512,
        speculative_ngram=4,
    )
    torch.manual_seed(0)
    spec = LLMEngine(cfg)
    b = [o.output_token_ids for o in spec.generate(prompts, sp).values()]
    assert a == b


def test_prefix_cache_on_gpu():
    """Prefix-cached prompts (suffix via the paged-context kernel) match the
    uncached engine's greedy outputs; second wave hits the cache."""
    from kserve_amd.engine.config import CacheConfig
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    shared = list(range(1, 40))  # 39-token shared prefix (2 full 16-blocks)
    prompts = [shared + [50 + i] for i in range(3)]
    sp = SamplingParams(temperature=0.0, max_tokens=8)

    torch.manual_seed(0)
    plain = LLMEngine(_cfg(enforce_eager=True))
    a = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    del plain
    torch.cuda.empty_cache()

    cfg = _cfg(enforce_eager=True)
    cfg.cache = CacheConfig(
        block_size=16, num_gpu_blocks=128, enable_prefix_caching=True
    )
    torch.manual_seed(0)
    cached = LLMEngine(cfg)
    b0 = [o.output_token_ids for o in cached.generate(prompts, sp).values()]
    b1 = [o.output_token_ids for o in cached.generate(prompts, sp).values()]
    assert cached.scheduler.block_manager.cache_hit_tokens > 0
    # hipBLASLt bf16 rounding varies with GEMM row count (suffix-only
    # prefill has fewer rows): compare the first token, which dominates
    for ta, tb0, tb1 in zip(a, b0, b1):
        assert ta[0] == tb0[0] == tb1[0]
        assert len(tb1) == 8


def test_mixtral_moe_on_gpu():
    """MoE engine path on GPU: decode is hipGraph-captured via the
    dense-bmm path (round 2); batched greedy equals single greedy."""
    from kserve_amd.engine.config import (
        CacheConfig,
        EngineConfig,
        ModelConfig,
        SchedulerConfig,
    )
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    torch.manual_seed(3)
    cfg = EngineConfig(
        model=ModelConfig(
            vocab_size=2048, hidden_size=256, intermediate_size=512,
            num_layers=2, num_heads=2, num_kv_heads=1, head_dim=128,
            max_position_embeddings=512, num_local_experts=4,
            num_experts_per_tok=2, model_name="moe-gpu-tiny",
        ),
        cache=CacheConfig(block_size=16, num_gpu_blocks=64),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=1024, max_model_len=256
        ),
        device="cuda",
        seed=0,
        eos_token_id=-1,
    )
    engine = LLMEngine(cfg)
    assert engine.runner._graphs, "MoE decode graphs must be captured"
    sp = SamplingParams(temperature=0.0, max_tokens=10)
    out = engine.generate([[1, 2, 3], [9, 8, 7, 6]], sp)
    assert all(len(o.output_token_ids) == 10 for o in out.values())
    single = engine.generate([[1, 2, 3]], sp)
    assert (
        list(single.values())[0].output_token_ids
        == list(out.values())[0].output_token_ids
    )
    # graph path must agree with a fully-eager engine
    del engine
    torch.cuda.empty_cache()
    cfg.enforce_eager = True
    eager = LLMEngine(cfg)
    out_e = eager.generate([[1, 2, 3], [9, 8, 7, 6]], sp)
    assert [o.output_token_ids for o in out_e.values()] == [
        o.output_token_ids for o in out.values()
    ]


def test_sampled_window_graph_topk1_matches_greedy():
    """In-graph sampled decode (round 2): with top_k=1 the fused sampler
    must reproduce greedy tokens exactly, through the SAMPLED graph variant
    (multi-step windows at temperature>0)."""
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    prompts = [[1, 2, 3, 4, 5], [100, 200, 300], [7, 8, 9, 10]]

    engine = LLMEngine(_cfg(enforce_eager=False))
    greedy = engine.generate(
        prompts, SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True)
    )
    toks_greedy = [o.output_token_ids for o in greedy.values()]

    # temperature>0 but top_k=1: the random path must still pick argmax
    sampled = engine.generate(
        prompts,
        SamplingParams(
            temperature=0.7, top_k=1, max_tokens=16, seed=5, ignore_eos=True
        ),
    )
    toks_sampled = [o.output_token_ids for o in sampled.values()]
    assert toks_greedy == toks_sampled
    # the sampled graph variant must actually have been captured
    assert engine.runner._sampled_graphs, "sampled window graph not used"


def test_sampled_window_graph_seeded_determinism():
    """Same seed + same batch => same sampled tokens across two engines
    (counter-based Gumbel in-graph); different seeds diverge."""
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    prompts = [[11, 12, 13, 14]]

    def run(seed):
        eng = LLMEngine(_cfg(enforce_eager=False))
        out = eng.generate(
            prompts,
            SamplingParams(
                temperature=1.0, top_p=0.95, max_tokens=24, seed=seed,
                ignore_eos=True,
            ),
        )
        toks = [o.output_token_ids for o in out.values()][0]
        del eng
        torch.cuda.empty_cache()
        return toks

    a = run(42)
    b = run(42)
    c = run(43)
    assert len(a) == 24
    assert a == b
    assert a != c  # 24 draws over vocab 2048: astronomically unlikely equal


def test_sliding_window_engine_graph_matches_eager():
    """Sliding-window attention through the ENGINE on GPU: the hipGraph
    path must match eager with the window active, and windowed generation
    must diverge from full attention once the context exceeds the window."""
    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    def cfg_w(window, eager):
        c = _cfg(enforce_eager=eager)
        c.model.sliding_window = window
        c.model.model_name = f"sw-{window}-{eager}"
        return c

    prompt = [list(range(1, 41))]  # 40-token prompt
    sp = SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True)

    eng_eager = LLMEngine(cfg_w(32, eager=True))
    toks_eager = [
        o.output_token_ids for o in eng_eager.generate(prompt, sp).values()
    ]
    del eng_eager
    torch.cuda.empty_cache()

    eng_graph = LLMEngine(cfg_w(32, eager=False))
    toks_graph = [
        o.output_token_ids for o in eng_graph.generate(prompt, sp).values()
    ]
    del eng_graph
    torch.cuda.empty_cache()
    assert toks_eager == toks_graph

    # full attention with identical weights must produce DIFFERENT logits
    # for a context beyond the window (token streams can coincide when a
    # random model collapses to an absorbing argmax, so compare logits)
    from kserve_amd.engine.config import ModelConfig
    from kserve_amd.models.llama import AttentionMetadata, LlamaForCausalLM

    mcfg = _cfg().model
    torch.manual_seed(0)
    m_full = LlamaForCausalLM(mcfg, dtype=torch.bfloat16, device="cuda")
    m_full.random_init(seed=3)
    mcfg_w = _cfg().model
    mcfg_w.sliding_window = 32
    torch.manual_seed(0)
    m_win = LlamaForCausalLM(mcfg_w, dtype=torch.bfloat16, device="cuda")
    m_win.random_init(seed=3)

    T = 64
    ids = torch.randint(0, mcfg.vocab_size, (T,), device="cuda")
    pos = torch.arange(T, device="cuda")

    def prefill_logits(model):
        caches = [
            (
                torch.zeros(8, mcfg.num_kv_heads, 16, 128,
                            dtype=torch.bfloat16, device="cuda"),
                torch.zeros(8, mcfg.num_kv_heads, 16, 128,
                            dtype=torch.bfloat16, device="cuda"),
            )
            for _ in range(mcfg.num_layers)
        ]
        meta = AttentionMetadata(
            is_prefill=True,
            slot_mapping=torch.arange(T, dtype=torch.int32, device="cuda"),
            cu_seqlens=torch.tensor([0, T], dtype=torch.int32,
                                    device="cuda"),
            max_seqlen=T,
        )
        h = model(ids, pos, caches, meta)
        return model.compute_logits(h)[-1]

    lf = prefill_logits(m_full).float()
    lw = prefill_logits(m_win).float()
    assert not torch.allclose(lf, lw, atol=1e-2, rtol=1e-2)


def test_draft_model_speculation_on_gpu():
    """Draft-model speculation through the native GPU verify path.

    Exact token equality with the plain decode engine holds in fp32 (CPU
    test) but not under bf16 on a RANDOM model: the verify forward runs
    width-k GEMM shapes whose bf16 results flip near-tie argmaxes
    (profiles gotcha: never exact-compare tokens across different
    batching). The GPU properties asserted instead: deterministic runs,
    the prefill-sampled first token matches the plain engine, requests
    reach max_tokens, and a same-weights draft compresses engine steps."""
    import dataclasses

    from kserve_amd.engine.engine import LLMEngine
    from kserve_amd.engine.sampling_params import SamplingParams

    base = _cfg(enforce_eager=False)
    base.scheduler.multi_step = 1
    prompts = [[1, 2, 3, 4, 5], [7, 8, 9]]
    sp = SamplingParams(temperature=0.0, max_tokens=20, ignore_eos=True)

    plain = LLMEngine(base)
    ref = [o.output_token_ids for o in plain.generate(prompts, sp).values()]
    del plain
    torch.cuda.empty_cache()

    def run_spec():
        cfg = _cfg(enforce_eager=False)
        cfg.scheduler.multi_step = 1
        cfg.scheduler.speculative_k = 3
        cfg.draft_model = dataclasses.replace(cfg.model)
        eng = LLMEngine(cfg)
        eng.draft.model.load_state_dict(eng.runner.model.state_dict())
        steps = 0
        outs = {}
        for p, rid in zip(prompts, ("a", "b")):
            eng.add_request(p, sp, request_id=rid)
        while eng.scheduler.has_unfinished() and steps < 60:
            for o in eng.step():
                outs.setdefault(o.request_id, []).extend(o.new_token_ids)
            steps += 1
        del eng
        torch.cuda.empty_cache()
        return [outs["a"], outs["b"]], steps

    toks1, steps1 = run_spec()
    toks2, steps2 = run_spec()
    assert toks1 == toks2, "spec decode must be deterministic"
    assert all(len(t) == 20 for t in toks1)
    # the first token comes from the identical prefill forward
    assert [t[0] for t in toks1] == [t[0] for t in ref]
    # perfect draft at k=3: ~20/4 spec rounds + prefill per wave
    assert steps1 <= 12, f"expected compressed steps, got {steps1}"
