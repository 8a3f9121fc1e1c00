#!/usr/bin/env python3
"""Flagship serving benchmark: Llama-3-8B decode throughput (output tok/s).

Contract (driver): `python bench.py --gpus N --steps K --warmup W` — for N>1
launched under torch.distributed.run with one rank per GPU. Each rank runs an
independent engine replica (data-parallel serving, the deployment shape of
BASELINE.json config "Llama-3-8B /v1/completions at 1/2/4/8 MI355X"); the
reported value is the WHOLE-JOB aggregate output tok/s.

A "step" = one continuous-batching decode iteration over the full running
batch. Synthetic random prompts, random-init bf16 weights (no network).
TTFT p50 is measured on the prefill wave and reported alongside.
"""

import argparse
import json
import os
import statistics
import time

# hipBLASLt GEMM algo selection via TunableOp: load the committed gfx950
# tuning table (profiles/tunableop_gfx950<ordinal>.csv) when present.
_TUNE_BASE = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "profiles", "tunableop_gfx950.csv")
if os.path.exists(_TUNE_BASE.replace(".csv", "0.csv")):
    os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
    os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
    os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", _TUNE_BASE)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    # defaults sized so a flagless run finishes in minutes AND its KV
    # demand (concurrency * (prompt + 8*(steps+warmup) + 64) tokens) fits
    # the pool with no preemption churn
    parser.add_argument("--steps", type=int, default=16)
    parser.add_argument("--warmup", type=int, default=4)
    parser.add_argument("--concurrency", type=int, default=1536,
                        help="concurrent sequences per GPU")
    parser.add_argument("--prompt-len", type=int, default=512)
    parser.add_argument("--model", type=str, default="llama-3-8b",
                        choices=["llama-3-8b", "llama-3-70b", "mixtral-8x7b",
                                 "mistral-7b", "tiny"])
    parser.add_argument("--tp", type=int, default=1,
                        help="tensor-parallel degree (ranks per engine)")
    parser.add_argument("--kv-blocks", type=int, default=0,
                        help="override KV pool size (0 = derive from HBM)")
    parser.add_argument("--eager", action="store_true",
                        help="disable hipGraph capture")
    parser.add_argument("--temperature", type=float, default=0.0)
    parser.add_argument("--multi-step", dest="multi_step", type=int, default=8,
                        help="decode iterations per hipGraph window")
    parser.add_argument("--kv-cache-dtype", type=str, default="auto",
                        choices=["auto", "fp8"],
                        help="opt-in fp8 E4M3 KV cache (NOT the headline "
                             "config; bf16 compute is unchanged)")
    parser.add_argument("--profile-cpu", action="store_true",
                        help="cProfile the timed loop and print hot functions")
    args = parser.parse_args()

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

    state = comm.init_distributed(tp_size=args.tp)
    rank, world = state.rank, state.world_size
    use_gpu = torch.cuda.is_available()
    device = state.device if use_gpu else "cpu"

    if args.model == "llama-3-8b":
        mcfg = ModelConfig.llama3_8b()
    elif args.model == "llama-3-70b":
        mcfg = ModelConfig.llama3_70b()
    elif args.model == "mixtral-8x7b":
        mcfg = ModelConfig.mixtral_8x7b()
    elif args.model == "mistral-7b":
        mcfg = ModelConfig.mistral_7b()
    else:
        mcfg = ModelConfig(
            vocab_size=1024, hidden_size=512, intermediate_size=1024,
            num_layers=2, num_heads=4, num_kv_heads=2, head_dim=128,
            max_position_embeddings=2048, model_name="tiny",
        )

    # each timed "step" is one scheduling iteration = up to `multi_step`
    # decode iterations (hipGraph window); budget the KV/model length for it
    window = args.multi_step
    max_model_len = args.prompt_len + (args.steps + args.warmup) * window + 64
    cfg = EngineConfig(
        model=mcfg,
        cache=CacheConfig(
            block_size=16,
            num_gpu_blocks=args.kv_blocks or None,
            gpu_memory_utilization=0.9,
            kv_cache_dtype=args.kv_cache_dtype,
        ),
        scheduler=SchedulerConfig(
            max_num_seqs=args.concurrency,
            max_num_batched_tokens=16384,
            max_model_len=max_model_len,
            multi_step=args.multi_step,
        ),
        device=device,
        seed=1234 + state.dp_rank,
        enforce_eager=args.eager or not use_gpu,
        eos_token_id=-1,
    )

    engine = LLMEngine(cfg)

    gen = torch.Generator().manual_seed(42 + state.dp_rank)
    sp = SamplingParams(
        temperature=args.temperature,
        max_tokens=max_model_len,  # never finishes inside the run
        ignore_eos=True,
    )
    t_submit = time.perf_counter()
    for i in range(args.concurrency):
        prompt = torch.randint(
            0, mcfg.vocab_size, (args.prompt_len,), generator=gen
        ).tolist()
        engine.add_request(prompt, sp, request_id=f"bench-{rank}-{i}")

    # ---- prefill wave (also measures TTFT) ----
    ttfts = []
    while engine.scheduler.num_waiting > 0 or any(
        r.first_token_time is None for r in engine.scheduler.running
    ):
        engine.step()
    for r in engine.scheduler.running:
        ttfts.append(r.first_token_time - t_submit)
    ttft_p50_ms = statistics.median(ttfts) * 1000 if ttfts else None
    prefill_done = time.perf_counter()

    # ---- warmup decode steps ----
    for _ in range(args.warmup):
        engine.step()

    # ---- timed region ----
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    tokens = 0
    prof = None
    if args.profile_cpu:
        import cProfile

        prof = cProfile.Profile()
        prof.enable()
    for _ in range(args.steps):
        outs = engine.step()
        tokens += sum(len(o.new_token_ids) for o in outs)
    if prof is not None:
        prof.disable()
    comm.barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    elapsed = t1 - t0

    # max elapsed over ranks; sum tokens over ranks
    if world > 1:
        import torch.distributed as dist

        # NCCL/RCCL requires device tensors for collectives
        red_dev = device if use_gpu else "cpu"
        te = torch.tensor([elapsed], dtype=torch.float64, device=red_dev)
        tk = torch.tensor([float(tokens)], dtype=torch.float64, device=red_dev)
        dist.all_reduce(te, op=dist.ReduceOp.MAX)
        dist.all_reduce(tk, op=dist.ReduceOp.SUM)
        elapsed = float(te[0])
        tokens = int(tk[0])

    if prof is not None and rank == 0:
        import pstats
        import sys

        stats = pstats.Stats(prof, stream=sys.stderr)
        stats.sort_stats("cumulative").print_stats(25)

    if rank == 0:
        value = tokens / elapsed
        n_gpus = world if use_gpu else args.gpus
        result = {
            "metric": f"output tok/s, {mcfg.model_name} /v1/completions decode",
            "value": round(value, 1),
            "unit": "tok/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bfloat16",
            "data": "synthetic random prompts, random-init weights",
            "ttft_p50_ms": round(ttft_p50_ms, 1) if ttft_p50_ms else None,
            "config": {
                "model": mcfg.model_name,
                "global_batch": args.concurrency * (world // args.tp),
                "seq_len": args.prompt_len,
                "parallelism": f"dp{world // args.tp}"
                + (f"xtp{args.tp}" if args.tp > 1 else ""),
                "kv_block_size": 16,
                "hipgraph": not cfg.enforce_eager,
                # headline runs use the bf16 KV default; fp8 KV is an
                # explicit opt-in and must be visible in the record
                "kv_cache_dtype": args.kv_cache_dtype,
            },
        }
        print(json.dumps(result), flush=True)

    comm.destroy_distributed()


if __name__ == "__main__":
    main()
