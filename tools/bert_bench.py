#!/usr/bin/env python3
"""BERT-base fill-mask bench (BASELINE config 2: V2 predict, bf16,
1x MI355X): batched encoder forward through the native varlen kernels,
random-init weights, synthetic token batches."""
import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    args = ap.parse_args()

    import torch

    from kserve_amd.models.bert import BertConfig, BertForMaskedLM

    use_gpu = torch.cuda.is_available()
    dev = "cuda:0" if use_gpu else "cpu"
    dtype = torch.bfloat16 if use_gpu else torch.float32
    cfg = BertConfig()  # bert-base: 12 layers, H=768, 12 heads
    torch.manual_seed(0)
    model = BertForMaskedLM(cfg, dtype=dtype, device=dev)
    with torch.no_grad():
        for p in model.parameters():
            if p.dim() >= 2:
                p.normal_(0.0, 0.02)
            else:
                p.fill_(0.01)

    B, T = args.batch, args.seq_len
    # varlen contract: flat tokens + cumulative boundaries
    ids = torch.randint(0, cfg.vocab_size, (B * T,), device=dev)
    cu = torch.arange(0, (B + 1) * T, T, dtype=torch.int32, device=dev)

    @torch.no_grad()
    def run():
        return model(ids, cu)

    for _ in range(args.warmup):
        run()
    if use_gpu:
        torch.cuda.synchronize()
    times = []
    for _ in range(args.iters):
        t0 = time.perf_counter()
        run()
        if use_gpu:
            torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    med = statistics.median(times)
    seq_s = B / med
    print(
        {
            "metric": "bert-base fill-mask sequences/s (V2 predict shape)",
            "batch": B,
            "seq_len": T,
            "ms_per_batch": round(med * 1000, 3),
            "sequences_s": round(seq_s, 1),
            "tokens_s": round(seq_s * T, 1),
            "dtype": str(dtype).split(".")[-1],
        }
    )


if __name__ == "__main__":
    main()
