"""Metrics aggregator (queue-proxy extension): one Prometheus endpoint
merging several local scrape targets — Prometheus can only scrape one
port per pod, so the queue-proxy's own metrics and the kserve-container's
metrics must come out of a single exposition.

Reference parity: qpext/cmd/qpext/main.go — ScrapeConfigurations.handleStats
(:243) reads the app's scrape config from the Knative env/annotation
contract, scrape (:198) honors per-target timeouts and formats, and
sanitizeMetrics (:113) rewrites the merged exposition so it stays valid:
one ``# HELP``/``# TYPE`` header per metric family across all targets,
duplicate series deduplicated, and optional source labelling. Rationale
qpext/README.md:1-15.
"""

from __future__ import annotations

import asyncio
import json
import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import httpx
from fastapi import FastAPI, Response

from kserve_amd.logging import logger

# Knative queue-proxy env contract (main.go: the aggregator reads the app
# target from these; annotation-driven in the webhook injector)
ENV_APP_PORT = "AGGREGATE_PROMETHEUS_METRICS_PORT"
ENV_APP_PATH = "APP_METRICS_PATH"
ENV_QP_PORT = "QUEUE_PROXY_METRICS_PORT"
ENV_TIMEOUT = "METRICS_SCRAPE_TIMEOUT_S"

PROM_CONTENT_TYPE = "text/plain; version=0.0.4; charset=utf-8"


@dataclass
class ScrapeTarget:
    url: str
    timeout_s: float = 5.0
    # label added to every series from this target (e.g. source="app");
    # empty = leave series untouched
    source_label: str = ""


@dataclass
class ScrapeConfiguration:
    """Targets assembled from the env contract (handleStats :243)."""

    targets: List[ScrapeTarget] = field(default_factory=list)

    @classmethod
    def from_env(cls, env: Optional[Dict[str, str]] = None) -> "ScrapeConfiguration":
        e = env if env is not None else os.environ
        timeout = float(e.get(ENV_TIMEOUT, "5"))
        targets = []
        qp_port = e.get(ENV_QP_PORT)
        if qp_port:
            targets.append(
                ScrapeTarget(
                    f"http://127.0.0.1:{qp_port}/metrics", timeout,
                    source_label="queue-proxy",
                )
            )
        app_port = e.get(ENV_APP_PORT)
        if app_port:
            path = e.get(ENV_APP_PATH, "/metrics")
            targets.append(
                ScrapeTarget(
                    f"http://127.0.0.1:{app_port}{path}", timeout,
                    source_label="kserve-container",
                )
            )
        return cls(targets=targets)


def _parse_families(text: str) -> Tuple[Dict[str, Dict], List[str]]:
    """Split an exposition into {family: {help, type, series:[...]}} plus
    any free-floating (headerless) series lines, preserving order."""
    families: Dict[str, Dict] = {}
    loose: List[str] = []
    for line in text.splitlines():
        line = line.rstrip()
        if not line:
            continue
        if line.startswith("# HELP ") or line.startswith("# TYPE "):
            parts = line.split(" ", 3)
            if len(parts) < 3:
                continue
            kind, name = parts[1], parts[2]
            fam = families.setdefault(
                name, {"help": None, "type": None, "series": []}
            )
            if line.startswith("# HELP "):
                fam["help"] = line
            else:
                fam["type"] = line
            continue
        if line.startswith("#"):
            continue  # comments dropped
        name = line.split("{", 1)[0].split(" ", 1)[0]
        if name in families:
            families[name]["series"].append(line)
        else:
            # series may precede or lack its header; attach by base name
            # (histogram/summary series belong to the family without the
            # _bucket/_sum/_count suffix)
            base = name
            for suffix in ("_bucket", "_sum", "_count", "_total"):
                if base.endswith(suffix) and base[: -len(suffix)] in families:
                    base = base[: -len(suffix)]
                    break
            if base in families:
                families[base]["series"].append(line)
            else:
                loose.append(line)
    return families, loose


def _add_label(series_line: str, label: str, value: str) -> str:
    name_part, _, rest = series_line.partition(" ")
    if "{" in name_part:
        name, _, labels = name_part.partition("{")
        labels = labels.rstrip("}")
        if f'{label}="' in labels:
            return series_line
        return f'{name}{{{labels},{label}="{value}"}} {rest}'
    return f'{name_part}{{{label}="{value}"}} {rest}'


def sanitize_metrics(
    expositions: List[Tuple[str, str]],
    drop_prefixes: Tuple[str, ...] = ("python_", "process_"),
) -> str:
    """Merge (source, exposition-text) pairs into ONE valid exposition:
    a single # HELP/# TYPE header per family, series from every source
    kept (labelled by source when that avoids duplicate series), noisy
    default-collector families dropped (sanitizeMetrics :113)."""
    merged: Dict[str, Dict] = {}
    loose_all: List[str] = []
    order: List[str] = []
    for source, text in expositions:
        families, loose = _parse_families(text)
        for name, fam in families.items():
            if any(name.startswith(p) for p in drop_prefixes):
                continue
            if name not in merged:
                merged[name] = {
                    "help": fam["help"],
                    "type": fam["type"],
                    "series": [],
                    "seen": set(),
                }
                order.append(name)
            tgt = merged[name]
            for s in fam["series"]:
                key = s.rsplit(" ", 1)[0]  # series identity = name+labels
                if key in tgt["seen"]:
                    # same series from two targets: disambiguate by source
                    if source:
                        s = _add_label(s, "source", source)
                        key = s.rsplit(" ", 1)[0]
                    if key in tgt["seen"]:
                        continue
                tgt["seen"].add(key)
                tgt["series"].append(s)
        for s in loose:
            name = s.split("{", 1)[0].split(" ", 1)[0]
            if not any(name.startswith(p) for p in drop_prefixes):
                loose_all.append(s)
    out: List[str] = []
    for name in order:
        fam = merged[name]
        if fam["help"]:
            out.append(fam["help"])
        if fam["type"]:
            out.append(fam["type"])
        out.extend(fam["series"])
    out.extend(loose_all)
    return "\n".join(out)


def create_qpext_app(
    targets: Optional[List] = None,
    transport: Optional[httpx.AsyncBaseTransport] = None,
    config: Optional[ScrapeConfiguration] = None,
) -> FastAPI:
    """The aggregator app: GET /metrics scrapes every target concurrently
    (per-target timeout; a failed target contributes nothing but never
    fails the scrape) and returns the sanitized merge."""
    if config is None:
        if targets is not None:
            config = ScrapeConfiguration(
                targets=[
                    t if isinstance(t, ScrapeTarget) else ScrapeTarget(t)
                    for t in targets
                ]
            )
        else:
            config = ScrapeConfiguration.from_env()
    app = FastAPI()
    client = httpx.AsyncClient(transport=transport)

    @app.get("/metrics")
    async def metrics():
        async def scrape(t: ScrapeTarget) -> Tuple[str, str]:
            try:
                r = await client.get(t.url, timeout=t.timeout_s)
                if r.status_code == 200:
                    return (t.source_label, r.text)
                logger.warning(
                    "qpext target %s returned %d", t.url, r.status_code
                )
            except Exception:
                logger.warning("qpext scrape failed for %s", t.url)
            return (t.source_label, "")

        parts = await asyncio.gather(*[scrape(t) for t in config.targets])
        merged = sanitize_metrics([p for p in parts if p[1]])
        return Response(
            content=merged + "\n", media_type=PROM_CONTENT_TYPE
        )

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok", "targets": [t.url for t in config.targets]}

    return app
