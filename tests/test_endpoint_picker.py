"""Native endpoint picker (the EPP the reference delegates to llm-d's
external image; llmisvc/scheduler.go:74-388 deploys it). Scoring,
health-based exclusion, session affinity, and the HTTP surface against
fake pool members."""

import asyncio

import httpx
import pytest

from kserve_amd.agent.endpoint_picker import (
    EndpointPicker,
    create_epp_app,
    parse_engine_metrics,
)

METRICS_TMPL = """# HELP llm_num_waiting_requests requests queued
# TYPE llm_num_waiting_requests gauge
llm_num_waiting_requests {waiting}
llm_num_running_requests {running}
llm_kv_cache_usage_ratio {kv}
other_metric 42
"""


def member_transport(states, down=()):
    """httpx transport faking N pool members by host name."""

    async def handler(request):
        host = request.url.host
        if host in down:
            raise httpx.ConnectError("down")
        st = states[host]
        return httpx.Response(200, text=METRICS_TMPL.format(**st))

    return httpx.MockTransport(handler)


def test_parse_engine_metrics():
    m = parse_engine_metrics(METRICS_TMPL.format(waiting=3, running=7, kv=0.5))
    assert m == {
        "llm_num_waiting_requests": 3.0,
        "llm_num_running_requests": 7.0,
        "llm_kv_cache_usage_ratio": 0.5,
    }


def run(coro):
    return asyncio.get_event_loop().run_until_complete(coro)


class TestScoring:
    def _picker(self, states, down=()):
        urls = [f"http://{h}" for h in states]
        p = EndpointPicker(urls, transport=member_transport(states, down))
        asyncio.run(p.scrape_once())
        return p

    def test_least_waiting_wins(self):
        p = self._picker(
            {
                "a": {"waiting": 5, "running": 1, "kv": 0.1},
                "b": {"waiting": 0, "running": 200, "kv": 0.9},
            }
        )
        assert p.pick() == "http://b"

    def test_kv_breaks_waiting_tie(self):
        p = self._picker(
            {
                "a": {"waiting": 1, "running": 10, "kv": 0.8},
                "b": {"waiting": 1, "running": 50, "kv": 0.2},
            }
        )
        assert p.pick() == "http://b"

    def test_unhealthy_excluded_after_threshold(self):
        states = {
            "a": {"waiting": 0, "running": 0, "kv": 0.0},
            "b": {"waiting": 9, "running": 9, "kv": 0.9},
        }
        p = EndpointPicker(
            ["http://a", "http://b"],
            transport=member_transport(states, down=("a",)),
            unhealthy_after=2,
        )
        asyncio.run(p.scrape_once())  # failure 1: not yet unhealthy, but
        # never became healthy either (starts unhealthy until first scrape)
        asyncio.run(p.scrape_once())
        assert p.pick() == "http://b"

    def test_recovery_rejoins_pool(self):
        states = {
            "a": {"waiting": 0, "running": 0, "kv": 0.0},
            "b": {"waiting": 9, "running": 0, "kv": 0.0},
        }
        down = {"a"}

        async def handler(request):
            if request.url.host in down:
                raise httpx.ConnectError("down")
            return httpx.Response(
                200, text=METRICS_TMPL.format(**states[request.url.host])
            )

        p = EndpointPicker(
            ["http://a", "http://b"],
            transport=httpx.MockTransport(handler),
            unhealthy_after=1,
        )
        asyncio.run(p.scrape_once())
        assert p.pick() == "http://b"
        down.clear()
        asyncio.run(p.scrape_once())
        assert p.pick() == "http://a"

    def test_session_affinity_sticky_until_unhealthy(self):
        states = {
            "a": {"waiting": 0, "running": 0, "kv": 0.0},
            "b": {"waiting": 0, "running": 0, "kv": 0.0},
        }
        p = self._picker(states)
        first = p.pick(session_id="user-42")
        for _ in range(5):
            assert p.pick(session_id="user-42") == first
        # a different session may land elsewhere, least-loaded still works
        assert p.pick() in ("http://a", "http://b")
        # kill the sticky member: affinity falls through to healthy ones
        host = first.replace("http://", "")
        p2 = self._picker(states, down=(host,))
        asyncio.run(p2.scrape_once())
        asyncio.run(p2.scrape_once())
        asyncio.run(p2.scrape_once())
        other = p2.pick(session_id="user-42")
        assert other is not None and other != first


class TestHTTPSurface:
    def test_pick_and_endpoints_routes(self):
        from fastapi.testclient import TestClient

        states = {
            "a": {"waiting": 2, "running": 1, "kv": 0.3},
            "b": {"waiting": 0, "running": 1, "kv": 0.1},
        }
        picker = EndpointPicker(
            ["http://a", "http://b"], transport=member_transport(states)
        )
        asyncio.run(picker.scrape_once())
        app = create_epp_app(picker)
        with TestClient(app) as client:
            r = client.post("/pick", json={})
            assert r.status_code == 200
            assert r.json()["endpoint"] == "http://b"
            eps = client.get("/endpoints").json()
            assert eps["http://a"]["num_waiting"] == 2.0
            assert client.get("/healthz").json()["members"] == 2

    def test_pick_503_when_pool_empty(self):
        from fastapi.testclient import TestClient

        picker = EndpointPicker([], transport=httpx.MockTransport(
            lambda r: httpx.Response(500)
        ))
        app = create_epp_app(picker)
        with TestClient(app) as client:
            assert client.post("/pick", json={}).status_code == 503


class TestExtProc:
    """Envoy ext-proc v3 face of the EPP (the GIE contract the reference
    gateway speaks to llm-d's picker): request_headers in ->
    header-mutation response naming the pool member."""

    def _picker(self, states, down=()):
        p = EndpointPicker(
            [f"http://{h}" for h in states],
            transport=member_transport(states, down),
        )
        asyncio.run(p.scrape_once())
        return p

    def test_pick_response_sets_destination_header(self):
        from kserve_amd.agent.ext_proc import (
            DESTINATION_HEADER,
            pick_response,
        )

        p = self._picker({
            "a": {"waiting": 5, "running": 0, "kv": 0.1},
            "b": {"waiting": 0, "running": 0, "kv": 0.1},
        })
        resp = pick_response(p, {})
        assert resp.WhichOneof("response") == "request_headers"
        common = resp.request_headers.response
        assert common.status == 0  # CONTINUE
        muts = {o.header.key: o.header.raw_value
                for o in common.header_mutation.set_headers}
        assert muts[DESTINATION_HEADER] == b"http://b"

    def test_pick_response_empty_pool_continues_unmutated(self):
        from kserve_amd.agent.ext_proc import pick_response

        p = EndpointPicker([], transport=httpx.MockTransport(
            lambda r: httpx.Response(500)))
        resp = pick_response(p, {})
        assert resp.request_headers.response.status == 0
        assert not resp.request_headers.response.header_mutation.set_headers

    def test_grpc_stream_roundtrip(self):
        """Full grpc.aio bidi stream: headers (with session id) then a
        body chunk; both phases answered, destination sticky."""
        import grpc

        from kserve_amd.agent.ext_proc import (
            DESTINATION_HEADER,
            SERVICE_NAME,
            HeaderValue,
            ProcessingRequest,
            ProcessingResponse,
            create_ext_proc_server,
        )

        states = {
            "a": {"waiting": 0, "running": 0, "kv": 0.0},
            "b": {"waiting": 0, "running": 0, "kv": 0.0},
        }
        picker = self._picker(states)

        async def run():
            server = create_ext_proc_server(picker, 0)
            port = server.add_insecure_port("127.0.0.1:0")
            await server.start()
            try:
                async with grpc.aio.insecure_channel(
                    f"127.0.0.1:{port}"
                ) as ch:
                    stream = ch.stream_stream(
                        f"/{SERVICE_NAME}/Process",
                        request_serializer=lambda m: m.SerializeToString(),
                        response_deserializer=ProcessingResponse.FromString,
                    )
                    call = stream()
                    req = ProcessingRequest()
                    req.request_headers.headers.headers.append(
                        HeaderValue(key="x-session-id", value="u1"))
                    await call.write(req)
                    first = await call.read()
                    body = ProcessingRequest()
                    body.request_body.body = b"{}"
                    body.request_body.end_of_stream = True
                    await call.write(body)
                    second = await call.read()
                    await call.done_writing()
                    return first, second
            finally:
                await server.stop(grace=None)

        first, second = asyncio.run(run())
        muts = {o.header.key: o.header.raw_value for o in
                first.request_headers.response.header_mutation.set_headers}
        dest = muts[DESTINATION_HEADER]
        assert dest in (b"http://a", b"http://b")
        assert second.WhichOneof("response") == "request_body"
        # sticky: same session hashes to the same member
        assert picker.pick(session_id="u1") == dest.decode()
