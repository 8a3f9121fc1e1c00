"""LLMInferenceService preset config-merge, template substitution, router
validation, TLS (reference llmisvc/config_merge.go:689-799, router*.go,
workload_tls_self_signed.go — table-driven like config_merge_test.go)."""

import base64

import pytest

from kserve_amd.controlplane.llmisvc_config import (
    ConfigMergeError,
    generate_self_signed_cert,
    merge_base_refs,
    render_config,
    render_tls_secret,
    shutdown_timeout_seconds,
    substitute_variables,
    validate_router,
)


class TestConfigMerge:
    def test_presets_merge_in_order_cr_wins(self):
        merged = merge_base_refs(
            spec={"workload": {"maxNumSeqs": 1536}},
            base_refs=[
                "kserve-config-llm-template",
                "kserve-config-llm-scheduler",
            ],
        )
        # preset default present, CR override wins
        assert merged["workload"]["maxModelLen"] == 8192
        assert merged["workload"]["maxNumSeqs"] == 1536
        assert merged["scheduler"]["grpcPort"] == 9002

    def test_later_base_ref_overrides_earlier(self):
        a = {"workload": {"image": "img-a", "x": 1}}
        b = {"workload": {"image": "img-b"}}
        merged = merge_base_refs(
            spec={}, base_refs=["a", "b"], extra_configs={"a": a, "b": b}
        )
        assert merged["workload"]["image"] == "img-b"
        assert merged["workload"]["x"] == 1

    def test_unknown_base_ref_raises(self):
        with pytest.raises(ConfigMergeError, match="unknown baseRef"):
            merge_base_refs({}, ["nope"])

    def test_lists_replace_not_append(self):
        merged = merge_base_refs(
            spec={"workload": {"args": ["--mine"]}},
            base_refs=["kserve-config-llm-decode-worker-data-parallel"],
        )
        assert merged["workload"]["args"] == ["--mine"]


class TestTemplateSubstitution:
    def test_name_namespace_and_spec_paths(self):
        out = render_config(
            "llama", "prod",
            spec={
                "model": {"name": "meta/llama-3-8b"},
                "workload": {
                    "serviceName": "{{ .Name }}-svc.{{ .Namespace }}",
                    "modelArg": "--model={{ .Spec.Model.Name }}",
                },
            },
        )
        assert out["workload"]["serviceName"] == "llama-svc.prod"
        assert out["workload"]["modelArg"] == "--model=meta/llama-3-8b"

    def test_global_config_and_preset_vars(self):
        out = render_config(
            "m", "ns",
            spec={},
            base_refs=["kserve-config-llm-tracing"],
            global_config={"otelCollector": "otel.observability"},
        )
        assert (
            out["tracing"]["otlpEndpoint"]
            == "http://otel.observability:4317"
        )

    def test_data_parallel_preset_substitutes_own_spec(self):
        out = render_config(
            "m", "ns",
            spec={"workload": {"parallelism": {"data": 4}}},
            base_refs=["kserve-config-llm-decode-worker-data-parallel"],
        )
        assert "--data-parallel-size=4" in out["workload"]["args"]
        assert "--data-parallel-rpc-port=5555" in out["workload"]["args"]

    def test_child_name_truncates_to_dns_label(self):
        long = "x" * 70
        out = substitute_variables(
            '{{ ChildName .Name "-decode" }}', {"Name": long}
        )
        assert len(out) <= 63
        assert out.startswith("x" * 50)

    def test_unknown_path_raises(self):
        with pytest.raises(ConfigMergeError, match="not found"):
            render_config("m", "ns", spec={"a": "{{ .Spec.Nope }}"})


class TestShutdownTimeout:
    def test_derived_from_grace_period(self):
        # config_merge.go:766-788: grace minus teardown headroom, floor 5
        assert shutdown_timeout_seconds(300) == 285
        assert shutdown_timeout_seconds(20) == 5
        assert shutdown_timeout_seconds(0) == 5


class TestRouterValidation:
    def test_valid_managed_route(self):
        assert validate_router({"route": {"http": {"spec": {}}}}) == []

    def test_spec_and_refs_exclusive(self):
        errs = validate_router(
            {"route": {"http": {"spec": {}, "refs": [{"name": "r"}]}}}
        )
        assert any("mutually exclusive" in e for e in errs)

    def test_route_refs_require_gateway_refs(self):
        errs = validate_router(
            {"route": {"http": {"refs": [{"name": "r"}]}}}
        )
        assert any("gateway.refs" in e for e in errs)
        ok = validate_router(
            {
                "route": {"http": {"refs": [{"name": "r"}]}},
                "gateway": {"refs": [{"name": "gw"}]},
            }
        )
        assert ok == []

    def test_scheduler_pool_vs_spec(self):
        errs = validate_router(
            {"scheduler": {"pool": {"name": "p"}, "spec": {}}}
        )
        assert any("scheduler" in e for e in errs)


class TestSelfSignedTLS:
    def test_cert_and_secret(self):
        pair = generate_self_signed_cert("svc", ["svc", "svc.ns"])
        assert pair["tls.key"].startswith(b"-----BEGIN PRIVATE KEY-----")
        assert pair["tls.crt"].startswith(b"-----BEGIN CERTIFICATE-----")
        secret = render_tls_secret("m-tls", "ns", "m-decode")
        assert secret["type"] == "kubernetes.io/tls"
        crt = base64.b64decode(secret["data"]["tls.crt"])
        assert b"BEGIN CERTIFICATE" in crt

    def test_cert_covers_cluster_dns_names(self):
        import subprocess
        import tempfile

        secret = render_tls_secret("m-tls", "ns", "m-decode")
        crt = base64.b64decode(secret["data"]["tls.crt"])
        with tempfile.NamedTemporaryFile(suffix=".crt") as f:
            f.write(crt)
            f.flush()
            out = subprocess.run(
                ["openssl", "x509", "-in", f.name, "-noout", "-text"],
                capture_output=True, text=True, check=True,
            ).stdout
        assert "m-decode.ns.svc.cluster.local" in out
