"""InferenceGraph router tests (all four node types + conditions + deps),
mirroring the coverage of reference cmd/router/main_test.go with httpx
MockTransport fake steps."""

import asyncio
import json

import httpx
import pytest

from kserve_amd.graph.router import GraphRouter, condition_matches, gjson_get
from kserve_amd.graph.types import InferenceGraphSpec


def make_transport(routes):
    """routes: {url_path: callable(body) -> (status, dict)}"""

    calls = []

    def handler(request: httpx.Request) -> httpx.Response:
        body = json.loads(request.content) if request.content else {}
        calls.append((str(request.url), body, dict(request.headers)))
        key = f"{request.url.scheme}://{request.url.host}{request.url.path}"
        fn = routes.get(key) or routes.get(request.url.path)
        if fn is None:
            return httpx.Response(404, json={"error": "no route"})
        status, out = fn(body)
        return httpx.Response(status, json=out)

    return httpx.MockTransport(handler), calls


def run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


class TestGjson:
    def test_get(self):
        doc = {"a": {"b": [1, 2, {"c": "x"}]}}
        assert gjson_get(doc, "a.b.2.c") == (True, "x")
        assert gjson_get(doc, "a.b.#") == (True, 3)
        assert gjson_get(doc, "a.missing")[0] is False

    def test_conditions(self):
        body = {"predictions": [2], "label": "cat"}
        assert condition_matches(body, "predictions.0==2")
        assert not condition_matches(body, "predictions.0==3")
        assert condition_matches(body, 'label=="cat"')
        assert condition_matches(body, "predictions")  # existence
        assert not condition_matches(body, "nope")
        assert condition_matches(body, "predictions.0>1")


def make_router(spec_dict, routes):
    transport, calls = make_transport(routes)
    spec = InferenceGraphSpec.from_dict(spec_dict)
    return GraphRouter(spec, transport=transport), calls


class TestSequence:
    def test_chaining(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"name": "s1", "serviceUrl": "http://a/predict"},
                        {"name": "s2", "serviceUrl": "http://b/predict"},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/predict": lambda b: (200, {"stage": 1, "in": b}),
                "http://b/predict": lambda b: (200, {"stage": 2, "in": b}),
            },
        )
        code, out = run(router.handle({"x": 1}, {}))
        assert code == 200
        assert out["stage"] == 2
        # second step received first step's response
        assert out["in"]["stage"] == 1

    def test_data_request(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"name": "s1", "serviceUrl": "http://a/"},
                        {
                            "name": "s2",
                            "serviceUrl": "http://b/",
                            "data": "$request",
                        },
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/": lambda b: (200, {"first": True}),
                "http://b/": lambda b: (200, {"got": b}),
            },
        )
        code, out = run(router.handle({"orig": 1}, {}))
        assert out["got"] == {"orig": 1}

    def test_hard_dependency_fails_fast(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {
                            "name": "s1",
                            "serviceUrl": "http://bad/",
                            "dependency": "Hard",
                        },
                        {"name": "s2", "serviceUrl": "http://b/"},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://bad/": lambda b: (500, {"error": "boom"}),
                "http://b/": lambda b: (200, {"ok": True}),
            },
        )
        code, out = run(router.handle({}, {}))
        assert code == 500
        assert out["step"] == "s1"
        assert len(calls) == 1  # s2 never called

    def test_soft_dependency_continues(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"name": "s1", "serviceUrl": "http://bad/"},
                        {"name": "s2", "serviceUrl": "http://b/", "data": "$request"},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://bad/": lambda b: (503, {"error": "down"}),
                "http://b/": lambda b: (200, {"ok": True}),
            },
        )
        code, out = run(router.handle({}, {}))
        assert code == 200
        assert out == {"ok": True}

    def test_condition_skips_step(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"name": "s1", "serviceUrl": "http://a/"},
                        {
                            "name": "s2",
                            "serviceUrl": "http://b/",
                            "condition": "predictions.0==1",
                        },
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/": lambda b: (200, {"predictions": [0]}),
                "http://b/": lambda b: (200, {"should": "not run"}),
            },
        )
        code, out = run(router.handle({}, {}))
        assert out == {"predictions": [0]}
        assert len(calls) == 1


class TestSplitter:
    def test_weighted_all_to_one(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Splitter",
                    "steps": [
                        {"name": "a", "serviceUrl": "http://a/", "weight": 100},
                        {"name": "b", "serviceUrl": "http://b/", "weight": 0},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/": lambda b: (200, {"from": "a"}),
                "http://b/": lambda b: (200, {"from": "b"}),
            },
        )
        for _ in range(5):
            code, out = run(router.handle({}, {}))
            assert out == {"from": "a"}

    def test_weighted_distribution(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Splitter",
                    "steps": [
                        {"name": "a", "serviceUrl": "http://a/", "weight": 50},
                        {"name": "b", "serviceUrl": "http://b/", "weight": 50},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/": lambda b: (200, {"from": "a"}),
                "http://b/": lambda b: (200, {"from": "b"}),
            },
        )
        seen = set()
        for _ in range(50):
            _, out = run(router.handle({}, {}))
            seen.add(out["from"])
        assert seen == {"a", "b"}


class TestEnsemble:
    def test_parallel_merge(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Ensemble",
                    "steps": [
                        {"name": "m1", "serviceUrl": "http://a/"},
                        {"name": "m2", "serviceUrl": "http://b/"},
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://a/": lambda b: (200, {"p": 1}),
                "http://b/": lambda b: (200, {"p": 2}),
            },
        )
        code, out = run(router.handle({}, {}))
        assert code == 200
        assert out == {"m1": {"p": 1}, "m2": {"p": 2}}


class TestSwitch:
    def test_first_matching(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Switch",
                    "steps": [
                        {
                            "name": "cat",
                            "serviceUrl": "http://cat/",
                            "condition": 'kind=="cat"',
                        },
                        {
                            "name": "dog",
                            "serviceUrl": "http://dog/",
                            "condition": 'kind=="dog"',
                        },
                    ],
                }
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://cat/": lambda b: (200, {"meow": True}),
                "http://dog/": lambda b: (200, {"woof": True}),
            },
        )
        _, out = run(router.handle({"kind": "dog"}, {}))
        assert out == {"woof": True}
        code, out = run(router.handle({"kind": "bird"}, {}))
        assert code == 404


class TestNested:
    def test_node_recursion(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [
                        {"name": "pre", "serviceUrl": "http://pre/"},
                        {"name": "sub", "nodeName": "child"},
                    ],
                },
                "child": {
                    "routerType": "Ensemble",
                    "steps": [{"name": "m", "serviceUrl": "http://m/"}],
                },
            }
        }
        router, calls = make_router(
            spec,
            {
                "http://pre/": lambda b: (200, {"pre": True}),
                "http://m/": lambda b: (200, {"model": b}),
            },
        )
        _, out = run(router.handle({"x": 1}, {}))
        assert out == {"m": {"model": {"pre": True}}}


class TestHeaders:
    def test_header_allowlist(self):
        spec = {
            "nodes": {
                "root": {
                    "routerType": "Sequence",
                    "steps": [{"name": "s", "serviceUrl": "http://a/"}],
                }
            }
        }
        router, calls = make_router(spec, {"http://a/": lambda b: (200, {})})
        run(
            router.handle(
                {}, {"Authorization": "tok", "X-Secret": "no", "x-request-id": "1"}
            )
        )
        _, _, hdrs = calls[0]
        assert hdrs.get("authorization") == "tok"
        assert hdrs.get("x-request-id") == "1"
        assert "x-secret" not in hdrs
