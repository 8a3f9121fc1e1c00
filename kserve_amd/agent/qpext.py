"""Metrics aggregator: one Prometheus endpoint merging several local
scrape targets (Prometheus scrapes one port per pod).

Reference parity: qpext/cmd/qpext/main.go — handleStats (:243), scrape
(:198), sanitizeMetrics (:113); rationale qpext/README.md:1-15.
"""

from __future__ import annotations

import asyncio
from typing import List, Optional

import httpx
from fastapi import FastAPI, Response

from kserve_amd.logging import logger


def sanitize_metrics(text: str, drop_prefixes=("python_", "process_")) -> str:
    """Drop duplicate/noisy series so the merged exposition stays valid."""
    out = []
    for line in text.splitlines():
        name = line.split("{")[0].split(" ")[0].lstrip("# HELPTYE ").strip()
        if any(line.startswith(p) or name.startswith(p) for p in drop_prefixes):
            continue
        out.append(line)
    return "\n".join(out)


def create_qpext_app(
    targets: List[str],
    transport: Optional[httpx.AsyncBaseTransport] = None,
) -> FastAPI:
    app = FastAPI()
    client = httpx.AsyncClient(timeout=5, transport=transport)

    @app.get("/metrics")
    async def metrics():
        async def scrape(url):
            try:
                r = await client.get(url)
                if r.status_code == 200:
                    return sanitize_metrics(r.text)
            except Exception:
                logger.warning("qpext scrape failed for %s", url)
            return ""

        parts = await asyncio.gather(*[scrape(t) for t in targets])
        merged = "\n".join(p for p in parts if p)
        return Response(content=merged + "\n", media_type="text/plain; version=0.0.4")

    return app
