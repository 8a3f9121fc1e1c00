#!/usr/bin/env python3
"""BASELINE config 2 at the SERVING level: BERT-base fill-mask through
the full REST stack — V2 `infer` requests (input_ids INT64 [B, S]) over
real localhost HTTP into the native varlen encoder on cuda:0.

The server runs in its own process (model + kernels on the GPU); an
aiohttp open-loop client drives concurrent V2 requests. Reports
sequences/s and request latency percentiles. Synthetic token batches,
random-init weights (BASELINE "synthetic data / random-init")."""

import argparse
import asyncio
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

VOCAB = 30522


def start_server(port: int, seq_len: int):
    import numpy as np
    import torch
    import uvicorn

    from kserve_amd.model import Model
    from kserve_amd.model_repository import ModelRepository
    from kserve_amd.protocol.dataplane import DataPlane
    from kserve_amd.protocol.infer_type import (
        InferOutput,
        InferRequest,
        InferResponse,
    )
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.models.bert import BertConfig, BertForMaskedLM

    use_gpu = torch.cuda.is_available()
    dev = "cuda:0" if use_gpu else "cpu"
    cfg = BertConfig()  # bert-base
    torch.manual_seed(0)
    dtype = torch.bfloat16 if use_gpu else torch.float32
    model = BertForMaskedLM(cfg, dtype=dtype, device=dev)
    with torch.no_grad():
        for p in model.parameters():
            if p.dim() >= 2:
                p.normal_(0.0, 0.02)
            else:
                p.fill_(0.01)

    class BertV2(Model):
        def __init__(self):
            super().__init__("bert")
            self.ready = True

        def predict(self, payload, headers=None):
            if isinstance(payload, dict):  # V1 :predict (batcher path)
                arr = np.asarray(payload["instances"], dtype=np.int64)
                b, s = arr.shape
                t = torch.from_numpy(arr).to(dev)
                cu = torch.arange(0, (b + 1) * s, s, dtype=torch.int32,
                                  device=dev)
                with torch.no_grad():
                    logits = model(t.reshape(-1), cu)
                pred = logits.argmax(-1).to(torch.int64)
                if use_gpu:
                    torch.cuda.synchronize()
                return {"predictions":
                        pred.reshape(b, s).cpu().tolist()}
            assert isinstance(payload, InferRequest)
            ids = payload.inputs[0].as_numpy().astype(np.int64)
            b, s = ids.shape
            t = torch.from_numpy(ids).to(dev)
            flat = t.reshape(-1)
            cu = torch.arange(0, (b + 1) * s, s, dtype=torch.int32,
                              device=dev)
            with torch.no_grad():
                logits = model(flat, cu)
            pred = logits.argmax(-1).to(torch.int64)
            if use_gpu:
                torch.cuda.synchronize()
            out_np = pred.reshape(b, s).cpu().numpy()
            out = InferOutput("predictions", [b, s], "INT64")
            out.set_data_from_numpy(out_np, binary_data=False)
            return InferResponse(payload.id, self.name, [out])

    repo = ModelRepository()
    repo.update(BertV2())
    app = create_app(DataPlane(repo))
    if os.environ.get("KS_BERT_BATCHER") == "1":
        import threading

        from kserve_amd.agent.batcher import create_batcher_proxy_app

        payload_logger = None
        if os.environ.get("KS_BERT_LOGGER") == "1":
            from kserve_amd.agent.payload_logger import PayloadLogger

            payload_logger = PayloadLogger(
                store_path="/tmp/bert-payload-logs", num_workers=2)
        proxy = create_batcher_proxy_app(
            f"http://127.0.0.1:{port}", "bert",
            max_batch_size=int(os.environ.get("KS_BATCH_MAX", "64")),
            max_latency_ms=int(os.environ.get("KS_BATCH_LAT_MS", "20")),
            payload_logger=payload_logger,
        )
        pcfg = uvicorn.Config(proxy, host="127.0.0.1", port=port + 1,
                              log_level="error")
        threading.Thread(target=uvicorn.Server(pcfg).run,
                         daemon=True).start()
    uvicorn.run(app, host="127.0.0.1", port=port, log_level="error")


async def drive(port, batch, seq_len, requests, concurrency,
                binary=False):
    import random

    import aiohttp
    import numpy as np

    rng = random.Random(0)
    url = f"http://127.0.0.1:{port}/v2/models/bert/infer"
    headers = {}
    if binary:
        # V2 binary tensor extension: JSON prefix + raw INT64 bytes,
        # prefix length in the inference-content-length header
        from kserve_amd.protocol.infer_type import InferInput, InferRequest

        arr = np.array(
            [[rng.randrange(VOCAB) for _ in range(seq_len)]
             for _ in range(batch)], dtype=np.int64)
        inp = InferInput("input_ids", [batch, seq_len], "INT64")
        inp.set_data_from_numpy(arr, binary_data=True)
        body_bytes, json_len = InferRequest("bert", [inp]).to_rest()
        headers = {
            "inference-content-length": str(json_len),
            "content-type": "application/octet-stream",
        }
        post_kwargs = {"data": body_bytes, "headers": headers}
    else:
        body = {
            "inputs": [{
                "name": "input_ids", "shape": [batch, seq_len],
                "datatype": "INT64",
                "data": [[rng.randrange(VOCAB) for _ in range(seq_len)]
                         for _ in range(batch)],
            }]
        }
        post_kwargs = {"json": body}
    lat = []
    conn = aiohttp.TCPConnector(limit=concurrency)
    timeout = aiohttp.ClientTimeout(total=120)
    async with aiohttp.ClientSession(connector=conn,
                                     timeout=timeout) as client:
        # warmup
        for _ in range(4):
            async with client.post(url, **post_kwargs) as r:
                assert r.status == 200, await r.text()

        sem = asyncio.Semaphore(concurrency)

        async def one():
            async with sem:
                t0 = time.perf_counter()
                async with client.post(url, **post_kwargs) as r:
                    await r.read()
                    assert r.status == 200
                lat.append(time.perf_counter() - t0)

        t0 = time.perf_counter()
        await asyncio.gather(*[one() for _ in range(requests)])
        elapsed = time.perf_counter() - t0
    lat.sort()
    return {
        "metric": "bert-base fill-mask seq/s through V2 HTTP"
                  + (" (binary tensors)" if binary else " (JSON)"),
        "value": round(requests * batch / elapsed, 1),
        "batch": batch, "seq_len": seq_len, "requests": requests,
        "concurrency": concurrency,
        "latency_p50_ms": round(lat[len(lat) // 2] * 1000, 1),
        "latency_p99_ms": round(lat[int(0.99 * len(lat)) - 1] * 1000, 1),
        "elapsed_s": round(elapsed, 2),
    }


async def drive_v1(port, seq_len, requests, concurrency, batcher):
    """1-sequence V1 :predict requests — the shape the in-pod batcher
    sidecar exists to coalesce (reference pkg/batcher)."""
    import random

    import aiohttp

    rng = random.Random(0)
    url = f"http://127.0.0.1:{port}/v1/models/bert:predict"
    lat = []
    conn = aiohttp.TCPConnector(limit=concurrency)
    timeout = aiohttp.ClientTimeout(total=120)
    async with aiohttp.ClientSession(connector=conn,
                                     timeout=timeout) as client:
        def body():
            return {"instances":
                    [[rng.randrange(VOCAB) for _ in range(seq_len)]]}

        for _ in range(4):
            async with client.post(url, json=body()) as r:
                assert r.status == 200, await r.text()

        sem = asyncio.Semaphore(concurrency)

        async def one():
            async with sem:
                t0 = time.perf_counter()
                async with client.post(url, json=body()) as r:
                    await r.read()
                    assert r.status == 200
                lat.append(time.perf_counter() - t0)

        t0 = time.perf_counter()
        await asyncio.gather(*[one() for _ in range(requests)])
        elapsed = time.perf_counter() - t0
    lat.sort()
    return {
        "metric": "bert 1-seq V1 requests/s "
                  + ("via batcher sidecar" if batcher else "direct")
                  + (" + payload logger"
                     if os.environ.get("KS_BERT_LOGGER") == "1" else ""),
        "value": round(requests / elapsed, 1),
        "seq_len": seq_len, "requests": requests,
        "concurrency": concurrency,
        "latency_p50_ms": round(lat[len(lat) // 2] * 1000, 1),
        "latency_p99_ms": round(lat[int(0.99 * len(lat)) - 1] * 1000, 1),
        "elapsed_s": round(elapsed, 2),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=18150)
    ap.add_argument("--batch", type=int, default=32)
    ap.add_argument("--seq-len", type=int, default=128)
    ap.add_argument("--requests", type=int, default=200)
    ap.add_argument("--concurrency", type=int, default=8)
    ap.add_argument("--serve", action="store_true")
    ap.add_argument("--binary", action="store_true",
                    help="V2 binary tensor extension request bodies")
    ap.add_argument("--v1-singles", action="store_true",
                    help="drive 1-sequence V1 :predict requests")
    ap.add_argument("--batcher", action="store_true",
                    help="with --v1-singles: route through the batcher "
                         "proxy sidecar (port+1)")
    ap.add_argument("--logger", action="store_true",
                    help="with --batcher: attach the payload logger "
                         "(CloudEvents to a local file sink)")
    args = ap.parse_args()
    if args.serve:
        start_server(args.port, args.seq_len)
        return

    import subprocess

    import requests as rq

    env = dict(os.environ)
    if args.batcher:
        env["KS_BERT_BATCHER"] = "1"
    if args.logger:
        env["KS_BERT_LOGGER"] = "1"
        os.environ["KS_BERT_LOGGER"] = "1"  # label only
    proc = subprocess.Popen(
        [sys.executable, os.path.abspath(__file__), "--serve",
         "--port", str(args.port), "--seq-len", str(args.seq_len)],
        env=env,
    )
    try:
        for _ in range(600):  # model init + first import can take a while
            try:
                if rq.get(f"http://127.0.0.1:{args.port}/",
                          timeout=1).status_code == 200:
                    break
            except Exception:
                time.sleep(0.5)
        else:
            raise RuntimeError("bert server did not come up")
        if args.v1_singles:
            res = asyncio.new_event_loop().run_until_complete(
                drive_v1(args.port + (1 if args.batcher else 0),
                         args.seq_len, args.requests, args.concurrency,
                         batcher=args.batcher)
            )
        else:
            res = asyncio.new_event_loop().run_until_complete(
                drive(args.port, args.batch, args.seq_len, args.requests,
                      args.concurrency, binary=args.binary)
            )
        print(json.dumps(res), flush=True)
    finally:
        proc.terminate()
        proc.wait(timeout=15)


if __name__ == "__main__":
    main()
