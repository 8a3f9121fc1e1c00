"""Credentials builder: ServiceAccount/Secret → storage-initializer env and
volumes (reference pkg/credentials/service_account_credentials.go:102-272,
table-driven like its *_test.go suites)."""

import base64
import json

import pytest

from kserve_amd.controlplane.apiserver import FakeAPIServer
from kserve_amd.controlplane.credentials import (
    CredentialsBuilder,
    build_secret_env_and_volumes,
    inject_credentials,
)
from kserve_amd.controlplane.isvc_controller import (
    FakeDeploymentController,
    InferenceServiceController,
)


def b64(s: str) -> str:
    return base64.b64encode(s.encode()).decode()


def make_secret(name, data=None, annotations=None):
    return {
        "apiVersion": "v1",
        "kind": "Secret",
        "metadata": {"name": name, "namespace": "default",
                     "annotations": annotations or {}},
        "data": {k: b64(v) for k, v in (data or {}).items()},
    }


class TestSecretToEnv:
    def test_s3_secret_with_annotations(self):
        secret = make_secret(
            "s3-creds",
            data={"AWS_ACCESS_KEY_ID": "AKID", "AWS_SECRET_ACCESS_KEY": "SK"},
            annotations={
                "serving.kserve.io/s3-endpoint": "minio:9000",
                "serving.kserve.io/s3-usehttps": "0",
                "serving.kserve.io/s3-region": "us-east-1",
            },
        )
        env, vols, mounts = build_secret_env_and_volumes(secret)
        names = {e["name"]: e for e in env}
        assert names["AWS_ACCESS_KEY_ID"]["valueFrom"]["secretKeyRef"] == {
            "name": "s3-creds",
            "key": "AWS_ACCESS_KEY_ID",
        }
        assert names["S3_ENDPOINT"]["value"] == "minio:9000"
        assert names["AWS_ENDPOINT_URL"]["value"] == "http://minio:9000"
        assert names["AWS_DEFAULT_REGION"]["value"] == "us-east-1"
        assert vols == [] and mounts == []

    def test_gcs_secret_mounts_json(self):
        secret = make_secret(
            "gcs-creds",
            data={"gcloud-application-credentials.json": "{}"},
        )
        env, vols, mounts = build_secret_env_and_volumes(secret)
        assert env[0]["name"] == "GOOGLE_APPLICATION_CREDENTIALS"
        assert vols[0]["secret"]["secretName"] == "gcs-creds"
        assert mounts[0]["mountPath"] == "/var/run/kserve/credentials"

    def test_azure_and_hf(self):
        secret = make_secret(
            "az",
            data={
                "AZURE_CLIENT_ID": "cid",
                "AZURE_CLIENT_SECRET": "cs",
                "AZURE_TENANT_ID": "tid",
                "HF_TOKEN": "tok",
            },
        )
        env, _, _ = build_secret_env_and_volumes(secret)
        names = {e["name"] for e in env}
        assert {"AZURE_CLIENT_ID", "AZURE_CLIENT_SECRET", "AZURE_TENANT_ID",
                "HF_TOKEN"} <= names

    def test_https_headers(self):
        secret = make_secret(
            "web",
            data={
                "https-host": "example.com",
                "headers": json.dumps({"X-Token": "abc"}),
            },
        )
        env, _, _ = build_secret_env_and_volumes(secret)
        assert {"name": "HEADERS_X_Token", "value": "abc"} in env


class TestServiceAccountFlow:
    def _server_with_sa(self):
        server = FakeAPIServer()
        server.create(
            make_secret(
                "s3-creds",
                data={"AWS_ACCESS_KEY_ID": "AKID", "AWS_SECRET_ACCESS_KEY": "SK"},
            )
        )
        server.create(
            {
                "apiVersion": "v1",
                "kind": "ServiceAccount",
                "metadata": {"name": "models-sa", "namespace": "default"},
                "secrets": [{"name": "s3-creds"}],
            }
        )
        return server

    def test_builder_walks_sa_secrets(self):
        server = self._server_with_sa()
        env, vols, mounts = CredentialsBuilder(server, "default").for_service_account(
            "models-sa"
        )
        assert any(e["name"] == "AWS_ACCESS_KEY_ID" for e in env)

    def test_live_controller_injects_into_init_container(self):
        server = self._server_with_sa()
        server.create(
            {
                "apiVersion": "serving.kserve.io/v1beta1",
                "kind": "InferenceService",
                "metadata": {"name": "iris", "namespace": "default"},
                "spec": {
                    "predictor": {
                        "serviceAccountName": "models-sa",
                        "model": {
                            "modelFormat": {"name": "sklearn"},
                            "storageUri": "s3://models/iris",
                            "protocolVersion": "v2",
                        },
                    }
                },
            }
        )
        ctrl = InferenceServiceController(server).build()
        dep_ctrl = FakeDeploymentController(server).build()
        from tests.test_live_controller import converge

        converge(ctrl, dep_ctrl)
        dep = server.get("apps/v1/Deployment", "default", "iris-predictor")
        init = dep["spec"]["template"]["spec"]["initContainers"][0]
        assert init["name"] == "storage-initializer"
        env_names = {e["name"] for e in init.get("env", [])}
        assert "AWS_ACCESS_KEY_ID" in env_names
        assert "AWS_SECRET_ACCESS_KEY" in env_names

    def test_storage_spec_envs(self):
        server = FakeAPIServer()
        server.create(
            make_secret(
                "storage-config",
                data={"my-s3": json.dumps({"type": "s3", "bucket": "b"})},
            )
        )
        env = CredentialsBuilder(server, "default").storage_spec_envs(
            "my-s3", storage_params={"path": "x"}
        )
        assert env[0]["name"] == "STORAGE_CONFIG"
        assert json.loads(env[0]["value"])["bucket"] == "b"
        assert env[1]["name"] == "STORAGE_OVERRIDE_CONFIG"

    def test_storage_spec_missing_key_raises(self):
        server = FakeAPIServer()
        server.create(make_secret("storage-config", data={}))
        with pytest.raises(LookupError):
            CredentialsBuilder(server, "default").storage_spec_envs("nope")


def test_inject_credentials_idempotent():
    pod = {
        "spec": {
            "initContainers": [
                {"name": "storage-initializer", "image": "x",
                 "env": [{"name": "AWS_ACCESS_KEY_ID", "value": "keep"}]}
            ],
            "containers": [],
        }
    }
    env = [
        {"name": "AWS_ACCESS_KEY_ID", "value": "new"},
        {"name": "S3_ENDPOINT", "value": "e"},
    ]
    inject_credentials(pod, env, [], [])
    init_env = pod["spec"]["initContainers"][0]["env"]
    # existing var kept, new var added
    assert {"name": "AWS_ACCESS_KEY_ID", "value": "keep"} in init_env
    assert {"name": "S3_ENDPOINT", "value": "e"} in init_env
