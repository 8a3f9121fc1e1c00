"""Admission webhook server: AdmissionReview envelope + JSONPatch
mutations + validators (reference pod/mutator.go:47-152 handling,
cmd/manager/main.go:238-282 registrations)."""

import base64
import json

import pytest
from fastapi.testclient import TestClient

from kserve_amd.controlplane.admission import create_admission_app, json_patch


def apply_patch(doc, ops):
    """Tiny RFC-6902 applier for round-trip verification."""
    import copy

    doc = copy.deepcopy(doc)

    def resolve(path):
        parts = [
            p.replace("~1", "/").replace("~0", "~")
            for p in path.split("/")[1:]
        ]
        cur = doc
        for p in parts[:-1]:
            cur = cur[int(p)] if isinstance(cur, list) else cur[p]
        return cur, parts[-1]

    for op in ops:
        parent, last = resolve(op["path"])
        if op["op"] == "add":
            if isinstance(parent, list):
                if last == "-":
                    parent.append(op["value"])
                else:
                    parent.insert(int(last), op["value"])
            else:
                parent[last] = op["value"]
        elif op["op"] == "replace":
            if isinstance(parent, list):
                parent[int(last)] = op["value"]
            else:
                parent[last] = op["value"]
        elif op["op"] == "remove":
            if isinstance(parent, list):
                parent.pop(int(last))
            else:
                del parent[last]
    return doc


@pytest.fixture
def client():
    return TestClient(create_admission_app())


def review(obj, uid="u1"):
    return {
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "request": {"uid": uid, "object": obj},
    }


def make_pod():
    return {
        "metadata": {
            "annotations": {
                "internal.serving.kserve.io/storage-initializer-sourceuri":
                    "s3://models/m",
            }
        },
        "spec": {"containers": [{"name": "kserve-container", "image": "i"}]},
    }


class TestMutatePods:
    def test_patch_injects_storage_initializer(self, client):
        pod = make_pod()
        r = client.post("/mutate-pods", json=review(pod))
        assert r.status_code == 200
        resp = r.json()["response"]
        assert resp["allowed"] and resp["uid"] == "u1"
        ops = json.loads(base64.b64decode(resp["patch"]))
        mutated = apply_patch(pod, ops)
        init = mutated["spec"]["initContainers"][0]
        assert init["name"] == "storage-initializer"
        assert init["args"][0] == "s3://models/m"
        # kserve-container got the shared volume mount
        assert any(
            v["name"] == "kserve-provision-location"
            for v in mutated["spec"]["containers"][0]["volumeMounts"]
        )

    def test_plain_pod_untouched(self, client):
        pod = {"metadata": {}, "spec": {"containers": [{"name": "x"}]}}
        r = client.post("/mutate-pods", json=review(pod))
        resp = r.json()["response"]
        assert resp["allowed"]
        assert "patch" not in resp  # no-op mutation => no patch


class TestValidators:
    def test_isvc_valid(self, client):
        obj = {
            "metadata": {"name": "ok"},
            "spec": {
                "predictor": {
                    "model": {
                        "modelFormat": {"name": "sklearn"},
                        "storageUri": "s3://b/m",
                    }
                }
            },
        }
        r = client.post("/validate-inferenceservices", json=review(obj))
        assert r.json()["response"]["allowed"]

    def test_isvc_bad_scheme_denied(self, client):
        obj = {
            "metadata": {"name": "bad"},
            "spec": {
                "predictor": {
                    "model": {
                        "modelFormat": {"name": "sklearn"},
                        "storageUri": "ftp://nope",
                    }
                }
            },
        }
        resp = client.post(
            "/validate-inferenceservices", json=review(obj)
        ).json()["response"]
        assert not resp["allowed"]
        assert "unsupported storage scheme" in resp["status"]["message"]

    def test_llm_router_denied(self, client):
        obj = {
            "metadata": {"name": "llm"},
            "spec": {"router": {"route": {"http": {"spec": {},
                                                   "refs": [{"name": "r"}]}}}},
        }
        resp = client.post(
            "/validate-llminferenceservices", json=review(obj)
        ).json()["response"]
        assert not resp["allowed"]

    def test_servingruntime_duplicate_priority_denied(self, client):
        obj = {
            "metadata": {"name": "rt"},
            "spec": {
                "supportedModelFormats": [
                    {"name": "sklearn", "priority": 1, "autoSelect": True},
                    {"name": "sklearn", "priority": 1, "autoSelect": True},
                ]
            },
        }
        resp = client.post(
            "/validate-servingruntimes", json=review(obj)
        ).json()["response"]
        assert not resp["allowed"]


def test_json_patch_roundtrip_nested():
    before = {"a": {"b": [1, 2]}, "keep": True, "gone": 1}
    after = {"a": {"b": [1, 2, 3], "new": "x"}, "keep": True}
    ops = json_patch(before, after)
    assert apply_patch(before, ops) == after
