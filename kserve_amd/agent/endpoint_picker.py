"""Endpoint picker (EPP): load-aware routing for LLM inference pools.

The reference deploys llm-d's external endpoint-picker image
(config-llm-scheduler.yaml:49, grpc :9002/health :9003) as the Gateway
API Inference Extension's picker for an InferencePool
(llmisvc/scheduler.go:74-388). This is the MI355X-native picker itself:

- scrapes each pool member's Prometheus ``/metrics`` (the engine exports
  llm_num_waiting / llm_num_running / llm_kv_usage, metrics.py) on an
  interval,
- scores endpoints: fewest waiting requests first, then lowest KV-cache
  usage, then fewest running (the llm-d default queue/kv scorer shape),
- honors session affinity (``x-session-id`` hashes to a sticky member
  while it stays healthy), and
- serves the picker API (`POST /pick`), the pool state (`GET /endpoints`)
  and health (:9003 contract → ``/healthz``).

Scoring is deterministic and testable without sockets: ``pick()`` is
pure given the scraped state; the HTTP layer is a thin FastAPI shell.
"""

from __future__ import annotations

import asyncio
import hashlib
import re
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import httpx
from fastapi import FastAPI, Request, Response

from kserve_amd.logging import logger

_METRIC_RE = re.compile(
    r"^(llm_num_waiting_requests|llm_num_running_requests|"
    r"llm_kv_cache_usage_ratio)(?:\{[^}]*\})?\s+"
    r"([0-9.eE+-]+)\s*$",
    re.M,
)


@dataclass
class EndpointState:
    url: str
    healthy: bool = False
    num_waiting: float = 0.0
    num_running: float = 0.0
    kv_usage: float = 0.0
    last_scrape: float = 0.0
    consecutive_failures: int = 0

    def score(self):
        """Lower is better: queue depth dominates, then KV pressure, then
        running batch size (the llm-d queue/kv scorer ordering)."""
        return (self.num_waiting, self.kv_usage, self.num_running)


def parse_engine_metrics(text: str) -> Dict[str, float]:
    out: Dict[str, float] = {}
    for m in _METRIC_RE.finditer(text):
        out[m.group(1)] = float(m.group(2))
    return out


class EndpointPicker:
    def __init__(
        self,
        endpoints: List[str],
        scrape_interval_s: float = 1.0,
        unhealthy_after: int = 3,
        transport: Optional[httpx.AsyncBaseTransport] = None,
    ):
        self.state: Dict[str, EndpointState] = {
            u: EndpointState(url=u) for u in endpoints
        }
        self.scrape_interval_s = scrape_interval_s
        self.unhealthy_after = unhealthy_after
        self._client = httpx.AsyncClient(transport=transport, timeout=2.0)
        self._task: Optional[asyncio.Task] = None

    # -- pool management ---------------------------------------------------
    def set_endpoints(self, endpoints: List[str]) -> None:
        for u in endpoints:
            self.state.setdefault(u, EndpointState(url=u))
        for u in list(self.state):
            if u not in endpoints:
                del self.state[u]

    # -- scraping ----------------------------------------------------------
    async def scrape_once(self) -> None:
        async def one(st: EndpointState):
            try:
                r = await self._client.get(f"{st.url}/metrics")
                if r.status_code != 200:
                    raise RuntimeError(f"status {r.status_code}")
                m = parse_engine_metrics(r.text)
                st.num_waiting = m.get("llm_num_waiting_requests", 0.0)
                st.num_running = m.get("llm_num_running_requests", 0.0)
                st.kv_usage = m.get("llm_kv_cache_usage_ratio", 0.0)
                st.healthy = True
                st.consecutive_failures = 0
                st.last_scrape = time.monotonic()
            except Exception:
                st.consecutive_failures += 1
                if st.consecutive_failures >= self.unhealthy_after:
                    st.healthy = False

        await asyncio.gather(*[one(s) for s in self.state.values()])

    async def run(self) -> None:
        while True:
            try:
                await self.scrape_once()
            except Exception:
                logger.exception("EPP scrape loop error")
            await asyncio.sleep(self.scrape_interval_s)

    # -- picking -----------------------------------------------------------
    def pick(self, session_id: Optional[str] = None) -> Optional[str]:
        healthy = [s for s in self.state.values() if s.healthy]
        if not healthy:
            return None
        if session_id:
            # sticky: hash to a member, fall through when it is unhealthy
            ordered = sorted(self.state)
            idx = int(
                hashlib.sha256(session_id.encode()).hexdigest(), 16
            ) % len(ordered)
            sticky = self.state[ordered[idx]]
            if sticky.healthy:
                return sticky.url
        return min(healthy, key=lambda s: (s.score(), s.url)).url


def create_epp_app(picker: EndpointPicker) -> FastAPI:
    from contextlib import asynccontextmanager

    @asynccontextmanager
    async def lifespan(app):
        picker._task = asyncio.create_task(picker.run())
        yield
        picker._task.cancel()

    app = FastAPI(lifespan=lifespan)

    @app.post("/pick")
    async def pick(request: Request):
        body = {}
        try:
            body = await request.json()
        except Exception:
            pass
        session = request.headers.get("x-session-id") or body.get("session_id")
        url = picker.pick(session_id=session)
        if url is None:
            return Response(status_code=503)
        return {"endpoint": url}

    @app.get("/endpoints")
    async def endpoints():
        return {
            s.url: {
                "healthy": s.healthy,
                "num_waiting": s.num_waiting,
                "num_running": s.num_running,
                "kv_usage": s.kv_usage,
            }
            for s in picker.state.values()
        }

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok", "members": len(picker.state)}

    return app


def main(argv=None):
    """EPP process: Envoy ext-proc gRPC on --port (the :9002 contract the
    llmisvc scheduler Deployment exposes) + the HTTP picker/health app on
    --http-port (:9003)."""
    import argparse
    import asyncio

    import uvicorn

    ap = argparse.ArgumentParser(description="kserve-amd endpoint picker")
    ap.add_argument("--endpoints", required=True,
                    help="comma-separated pool member base URLs")
    ap.add_argument("--port", type=int, default=9002,
                    help="ext-proc gRPC port")
    ap.add_argument("--http-port", type=int, default=9003,
                    help="HTTP /pick /endpoints /healthz port")
    ap.add_argument("--scrape-interval", type=float, default=1.0)
    args = ap.parse_args(argv)
    picker = EndpointPicker(
        args.endpoints.split(","), scrape_interval_s=args.scrape_interval
    )

    async def serve():
        from kserve_amd.agent.ext_proc import create_ext_proc_server

        grpc_server = create_ext_proc_server(picker, args.port)
        await grpc_server.start()
        config = uvicorn.Config(
            create_epp_app(picker), host="0.0.0.0", port=args.http_port,
            log_level="warning",
        )
        await uvicorn.Server(config).serve()
        await grpc_server.stop(grace=2.0)

    asyncio.run(serve())


if __name__ == "__main__":
    main()
