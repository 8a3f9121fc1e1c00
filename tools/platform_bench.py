#!/usr/bin/env python3
"""Platform-overhead latency benchmark: sklearn-iris V1 :predict and V2
infer through the FULL REST server over real localhost HTTP, driven
open-loop at fixed QPS — the same shape as the reference's published
vegeta tables (BASELINE.md test/benchmark/README.md:61-91; the
RawDeployment rows, 1.3-2.7 ms p50 at QPS 5-500 on n1-standard GKE
nodes, are the bar for our data-plane overhead). CPU-only."""

import argparse
import asyncio
import json
import os
import statistics
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def make_model_dir() -> str:
    import joblib
    from sklearn.datasets import load_iris
    from sklearn.linear_model import LogisticRegression

    X, y = load_iris(return_X_y=True)
    clf = LogisticRegression(max_iter=200).fit(X, y)
    d = tempfile.mkdtemp(prefix="iris-")
    joblib.dump(clf, os.path.join(d, "model.joblib"))
    return d


def start_server(port: int):
    import uvicorn

    from kserve_amd.model_repository import ModelRepository
    from kserve_amd.protocol.dataplane import DataPlane
    from kserve_amd.protocol.rest.server import create_app
    from kserve_amd.runtimes.sklearnserver import SKLearnModel

    model = SKLearnModel("iris", make_model_dir())
    model.load()
    repo = ModelRepository()
    repo.update(model)
    app = create_app(DataPlane(repo))
    config = uvicorn.Config(app, host="127.0.0.1", port=port,
                            log_level="error")
    server = uvicorn.Server(config)
    t = threading.Thread(target=server.run, daemon=True)
    t.start()
    import httpx

    for _ in range(100):
        try:
            if httpx.get(f"http://127.0.0.1:{port}/").status_code == 200:
                return server
        except Exception:
            time.sleep(0.05)
    raise RuntimeError("server did not come up")


async def run_level(port: int, qps: int, seconds: float, protocol: str):
    import httpx

    v1_body = {"instances": [[5.1, 3.5, 1.4, 0.2]]}
    v2_body = {
        "inputs": [{
            "name": "input-0", "shape": [1, 4], "datatype": "FP64",
            "data": [[5.1, 3.5, 1.4, 0.2]],
        }]
    }
    if protocol == "v1":
        url = f"http://127.0.0.1:{port}/v1/models/iris:predict"
        body = v1_body
    elif protocol == "v2":
        url = f"http://127.0.0.1:{port}/v2/models/iris/infer"
        body = v2_body
    else:  # "root": uvicorn+FastAPI floor, no model work
        url = f"http://127.0.0.1:{port}/"
        body = None
    lat = []
    import aiohttp

    conn = aiohttp.TCPConnector(limit=64)
    timeout = aiohttp.ClientTimeout(total=30)
    async with aiohttp.ClientSession(connector=conn,
                                     timeout=timeout) as client:
        async def call():
            if body is None:
                async with client.get(url) as r:
                    await r.read()
                    return r
            async with client.post(url, json=body) as r:
                await r.read()
                return r

        for _ in range(20):
            r = await call()
            assert r.status == 200

        interval = 1.0 / qps
        n = int(seconds * qps)
        tasks = []

        async def one(delay):
            await asyncio.sleep(delay)
            t0 = time.perf_counter()
            r = await call()
            lat.append(time.perf_counter() - t0)
            assert r.status == 200

        for i in range(n):
            tasks.append(asyncio.ensure_future(one(i * interval)))
        await asyncio.gather(*tasks)
    lat.sort()

    def pct(p):
        return lat[min(len(lat) - 1, int(p * len(lat)))] * 1000

    return {
        "protocol": protocol, "qps": qps, "n": len(lat),
        "mean_ms": round(statistics.mean(lat) * 1000, 3),
        "p50_ms": round(pct(0.50), 3),
        "p95_ms": round(pct(0.95), 3),
        "p99_ms": round(pct(0.99), 3),
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=18080)
    ap.add_argument("--seconds", type=float, default=5.0)
    ap.add_argument("--qps", default="5,50,500")
    ap.add_argument("--serve", action="store_true",
                    help="internal: run the server process only")
    ap.add_argument("--workers", type=int, default=1,
                    help="uvicorn worker processes (server side)")
    args = ap.parse_args()
    if args.serve:
        start_server(args.port)
        while True:
            time.sleep(3600)

    # the server runs in its OWN process: sharing the GIL with the load
    # generator inflates latencies ~3x and caps throughput
    import subprocess

    proc = subprocess.Popen(
        [sys.executable, os.path.abspath(__file__), "--serve",
         "--port", str(args.port)],
    )
    try:
        import httpx

        for _ in range(200):
            try:
                if httpx.get(
                    f"http://127.0.0.1:{args.port}/"
                ).status_code == 200:
                    break
            except Exception:
                time.sleep(0.1)
        else:
            raise RuntimeError("server process did not come up")
        loop = asyncio.new_event_loop()
        for protocol in ("root", "v1", "v2"):
            for qps in [int(q) for q in args.qps.split(",")]:
                res = loop.run_until_complete(
                    run_level(args.port, qps, args.seconds, protocol)
                )
                print(json.dumps(res), flush=True)
    finally:
        proc.terminate()
        proc.wait(timeout=10)


if __name__ == "__main__":
    main()
