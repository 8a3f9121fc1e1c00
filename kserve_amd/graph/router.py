"""InferenceGraph router: stateless HTTP service executing a serialized graph.

Reference parity: cmd/router/main.go — routeStep (:242-376), executeStep
recursion (:385-391), callService w/ header allow-list (:94-177),
pickupRoute weighted random (:179-193), pickupRouteByCondition (:195-205),
graphHandler (:405-423), /readyz + graceful drain (:468-476).

Node semantics:
- Sequence: steps in order; ``data: $request`` re-sends the original body,
  ``$response`` chains the previous output; per-step condition gates on the
  PREVIOUS response; Hard dependency failure aborts the chain.
- Splitter: one step chosen by weight.
- Ensemble: all steps in parallel; response = {stepName: result}.
- Switch: first step whose condition matches the request.

Run: python -m kserve_amd.graph.router --graph-json '<spec>' [--port 8080]
"""

from __future__ import annotations

import asyncio
import json
import random
import re
from typing import Any, Dict, List, Optional, Tuple

import httpx
from fastapi import FastAPI, Request, Response

from kserve_amd.graph.types import (
    Dependency,
    InferenceGraphSpec,
    InferenceRouter,
    InferenceStep,
    NodeType,
)
from kserve_amd.logging import configure_logging, logger

# headers propagated to steps (reference: allow-list regex via
# PROPAGATE_HEADERS env, main.go compiledHeaderPattern)
DEFAULT_PROPAGATE_HEADERS = ("authorization", "x-request-id", "x-b3-traceid")


def gjson_get(doc: Any, path: str) -> Tuple[bool, Any]:
    """Minimal gjson-compatible path lookup: dotted keys, integer array
    indices, and the ``#`` array-length / query-less forms used by graph
    conditions."""
    cur = doc
    if path == "":
        return True, cur
    for part in path.split("."):
        if isinstance(cur, list):
            if part == "#":
                cur = len(cur)
                continue
            try:
                idx = int(part)
            except ValueError:
                return False, None
            if idx >= len(cur):
                return False, None
            cur = cur[idx]
        elif isinstance(cur, dict):
            if part not in cur:
                return False, None
            cur = cur[part]
        else:
            return False, None
    return True, cur


_COND_RE = re.compile(r"^(?P<path>[^=<>!]+?)\s*(?P<op>==|!=|>=|<=|>|<)\s*(?P<val>.+)$")


def condition_matches(body: Any, condition: Optional[str]) -> bool:
    """gjson-style condition: either a bare path (existence check) or
    ``path <op> literal`` (reference uses gjson.Get(...).Exists())."""
    if not condition:
        return True
    m = _COND_RE.match(condition.strip())
    if not m:
        ok, _ = gjson_get(body, condition.strip())
        return ok
    ok, val = gjson_get(body, m.group("path").strip())
    if not ok:
        return False
    raw = m.group("val").strip().strip('"').strip("'")
    try:
        lit: Any = json.loads(raw)
    except json.JSONDecodeError:
        lit = raw
    op = m.group("op")
    try:
        if op == "==":
            return val == lit or str(val) == str(lit)
        if op == "!=":
            return val != lit and str(val) != str(lit)
        return {
            ">": val > lit,
            "<": val < lit,
            ">=": val >= lit,
            "<=": val <= lit,
        }[op]
    except TypeError:
        return False


class GraphRouter:
    def __init__(
        self,
        spec: InferenceGraphSpec,
        timeout_s: float = 60.0,
        transport: Optional[httpx.AsyncBaseTransport] = None,
        propagate_headers: Tuple[str, ...] = DEFAULT_PROPAGATE_HEADERS,
    ):
        self.spec = spec
        self.timeout_s = timeout_s
        self.propagate_headers = tuple(h.lower() for h in propagate_headers)
        self._client = httpx.AsyncClient(timeout=timeout_s, transport=transport)
        self._rng = random.SystemRandom()  # reference uses crypto-rand

    async def close(self):
        await self._client.aclose()

    def _filter_headers(self, headers: Dict[str, str]) -> Dict[str, str]:
        return {
            k: v for k, v in headers.items() if k.lower() in self.propagate_headers
        }

    async def call_service(
        self, url: str, body: Any, headers: Dict[str, str]
    ) -> Tuple[int, Any]:
        r = await self._client.post(
            url, json=body, headers=self._filter_headers(headers)
        )
        try:
            return r.status_code, r.json()
        except json.JSONDecodeError:
            return r.status_code, {"raw": r.text}

    async def execute_step(
        self, step: InferenceStep, body: Any, headers: Dict[str, str]
    ) -> Tuple[int, Any]:
        if step.node_name:
            return await self.route_node(step.node_name, body, headers)
        url = step.service_url
        if url is None:
            raise ValueError(f"Step {step.step_name} has no serviceUrl")
        return await self.call_service(url, body, headers)

    async def route_node(
        self, node_name: str, body: Any, headers: Dict[str, str]
    ) -> Tuple[int, Any]:
        node = self.spec.nodes.get(node_name)
        if node is None:
            return 404, {"error": f"node {node_name} not found"}
        t = node.router_type
        if t == NodeType.Sequence:
            return await self._route_sequence(node, body, headers)
        if t == NodeType.Splitter:
            return await self._route_splitter(node, body, headers)
        if t == NodeType.Ensemble:
            return await self._route_ensemble(node, body, headers)
        if t == NodeType.Switch:
            return await self._route_switch(node, body, headers)
        return 400, {"error": f"unknown node type {t}"}

    # -- node types (reference main.go:242-376) ----------------------------
    async def _route_sequence(self, node, request_body, headers):
        response: Any = None
        code = 200
        for step in node.steps:
            # input selection
            if step.data == "$request" or response is None:
                body = request_body
            else:
                body = response
            # condition gates on the previous response (main.go:348-356)
            if step.condition is not None and not condition_matches(
                response if response is not None else request_body, step.condition
            ):
                continue
            code, out = await self.execute_step(step, body, headers)
            if code >= 400:
                if step.dependency == Dependency.Hard:
                    return code, {
                        "error": "hard dependency failed",
                        "step": step.step_name,
                        "response": out,
                    }
                # soft: keep previous response, continue
                continue
            response = out
        return code, response

    async def _route_splitter(self, node, body, headers):
        total = sum(s.weight or 0 for s in node.steps)
        if total <= 0:
            return 500, {"error": "splitter weights sum to zero"}
        point = self._rng.randint(0, total - 1)
        acc = 0
        for step in node.steps:
            acc += step.weight or 0
            if point < acc:
                return await self.execute_step(step, body, headers)
        return 500, {"error": "splitter fell through"}

    async def _route_ensemble(self, node, body, headers):
        results = await asyncio.gather(
            *[self.execute_step(s, body, headers) for s in node.steps],
            return_exceptions=True,
        )
        merged: Dict[str, Any] = {}
        for step, res in zip(node.steps, results):
            if isinstance(res, Exception):
                merged[step.step_name] = {"error": str(res)}
            else:
                merged[step.step_name] = res[1]
        return 200, merged

    async def _route_switch(self, node, body, headers):
        for step in node.steps:
            if condition_matches(body, step.condition):
                return await self.execute_step(step, body, headers)
        # no condition matched: echo the request (reference returns request)
        return 404, {"error": "no switch condition matched"}

    async def handle(self, body: Any, headers: Dict[str, str]) -> Tuple[int, Any]:
        return await self.route_node("root", body, headers)


def create_router_app(router: GraphRouter) -> FastAPI:
    app = FastAPI()
    state = {"draining": False}

    @app.post("/")
    async def graph_handler(request: Request):
        try:
            body = await request.json()
        except json.JSONDecodeError:
            return Response(
                content=json.dumps({"error": "invalid JSON"}), status_code=400
            )
        code, out = await router.handle(body, dict(request.headers))
        return Response(
            content=json.dumps(out),
            status_code=code,
            media_type="application/json",
        )

    @app.get("/readyz")
    async def readyz():
        if state["draining"]:
            return Response(status_code=503)
        return {"status": "ok"}

    @app.on_event("shutdown")
    async def drain():
        state["draining"] = True
        await router.close()

    return app


def main(argv=None):
    import argparse

    import uvicorn

    parser = argparse.ArgumentParser()
    parser.add_argument("--graph-json", required=True)
    parser.add_argument("--port", type=int, default=8080)
    parser.add_argument("--timeout", type=float, default=60.0)
    args = parser.parse_args(argv)
    configure_logging()
    spec = InferenceGraphSpec.from_dict(json.loads(args.graph_json))
    router = GraphRouter(spec, timeout_s=args.timeout)
    uvicorn.run(create_router_app(router), host="0.0.0.0", port=args.port)


if __name__ == "__main__":
    main()
