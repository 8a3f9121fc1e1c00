"""Native HTTP cloud providers against in-process fake servers — the moto
role of the reference's storage tests (python/storage/test), bytes
end-to-end and offline."""

import base64
import gzip
import hashlib
import hmac
import io
import json
import os
import tarfile
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, unquote, urlparse

import pytest

from kserve_amd.storage.http_providers import (
    AzureBlobClient,
    GCSClient,
    OCIRegistryClient,
    S3Client,
    WebHDFSClient,
    sigv4_headers,
)
from kserve_amd.storage.storage import Storage


@pytest.fixture
def http_server():
    """Factory: start a ThreadingHTTPServer around a handler class; yields
    (make, cleanup) and tears all servers down."""
    servers = []

    def make(handler_cls):
        srv = ThreadingHTTPServer(("127.0.0.1", 0), handler_cls)
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        servers.append(srv)
        return f"http://127.0.0.1:{srv.server_address[1]}"

    yield make
    for s in servers:
        s.shutdown()
        s.server_close()


# ---------------------------------------------------------------------------
# S3
# ---------------------------------------------------------------------------

S3_OBJECTS = {
    "models/iris/model.joblib": b"JOBLIB-BYTES",
    "models/iris/metadata.json": b'{"name": "iris"}',
    "models/iris/sub/weights.bin": b"\x00\x01\x02",
}
S3_KEY_ID = "AKIDTEST"
S3_SECRET = "sekrit"


class FakeS3Handler(BaseHTTPRequestHandler):
    require_auth = True

    def log_message(self, *a):
        pass

    def _verify_sigv4(self) -> bool:
        auth = self.headers.get("Authorization", "")
        if not auth.startswith("AWS4-HMAC-SHA256"):
            return False
        # recompute the signature with the shared secret over the same
        # canonical request the client claims to have signed
        parsed = urlparse(self.path)
        query = []
        for k, vs in parse_qs(parsed.query, keep_blank_values=True).items():
            for v in vs:
                query.append((k, v))
        import datetime

        amz_date = self.headers["x-amz-date"]
        now = datetime.datetime.strptime(amz_date, "%Y%m%dT%H%M%SZ")
        expect = sigv4_headers(
            self.command,
            self.headers["Host"],
            unquote(parsed.path),
            query,
            "us-east-1",
            S3_KEY_ID,
            S3_SECRET,
            payload_hash=self.headers.get("x-amz-content-sha256", ""),
            now=now,
        )
        return hmac.compare_digest(expect["Authorization"], auth)

    def do_GET(self):
        if self.require_auth and not self._verify_sigv4():
            self.send_response(403)
            self.end_headers()
            self.wfile.write(b"SignatureDoesNotMatch")
            return
        parsed = urlparse(self.path)
        parts = unquote(parsed.path).lstrip("/").split("/", 1)
        bucket, rest = parts[0], parts[1] if len(parts) > 1 else ""
        qs = parse_qs(parsed.query)
        if qs.get("list-type") == ["2"]:
            prefix = qs.get("prefix", [""])[0]
            keys = [k for k in S3_OBJECTS if k.startswith(prefix)]
            body = "<ListBucketResult>"
            for k in keys:
                body += f"<Contents><Key>{k}</Key></Contents>"
            body += "<IsTruncated>false</IsTruncated></ListBucketResult>"
            self.send_response(200)
            self.end_headers()
            self.wfile.write(body.encode())
            return
        if rest in S3_OBJECTS:
            self.send_response(200)
            self.end_headers()
            self.wfile.write(S3_OBJECTS[rest])
            return
        self.send_response(404)
        self.end_headers()


class TestS3:
    def test_signed_download_prefix(self, http_server, tmp_path):
        endpoint = http_server(FakeS3Handler)
        client = S3Client(
            endpoint=endpoint,
            region="us-east-1",
            access_key=S3_KEY_ID,
            secret_key=S3_SECRET,
            anonymous=False,
        )
        n = client.download_prefix("bkt", "models/iris", str(tmp_path))
        assert n == 3
        assert (tmp_path / "model.joblib").read_bytes() == b"JOBLIB-BYTES"
        assert (tmp_path / "sub" / "weights.bin").read_bytes() == b"\x00\x01\x02"

    def test_bad_signature_rejected(self, http_server, tmp_path):
        endpoint = http_server(FakeS3Handler)
        client = S3Client(
            endpoint=endpoint, access_key=S3_KEY_ID, secret_key="WRONG",
            anonymous=False,
        )
        with pytest.raises(RuntimeError, match="403"):
            client.list_objects("bkt", "models/iris")

    def test_storage_download_s3_uri(self, http_server, tmp_path, monkeypatch):
        endpoint = http_server(FakeS3Handler)
        monkeypatch.setenv("AWS_ENDPOINT_URL", endpoint)
        monkeypatch.setenv("AWS_ACCESS_KEY_ID", S3_KEY_ID)
        monkeypatch.setenv("AWS_SECRET_ACCESS_KEY", S3_SECRET)
        out = Storage.download("s3://bkt/models/iris", str(tmp_path / "out"))
        assert sorted(os.listdir(out)) == ["metadata.json", "model.joblib", "sub"]

    def test_missing_prefix_raises(self, http_server, tmp_path):
        endpoint = http_server(FakeS3Handler)
        client = S3Client(
            endpoint=endpoint, access_key=S3_KEY_ID, secret_key=S3_SECRET,
            anonymous=False,
        )
        with pytest.raises(FileNotFoundError):
            client.download_prefix("bkt", "no/such", str(tmp_path))


# ---------------------------------------------------------------------------
# Azure Blob
# ---------------------------------------------------------------------------

AZ_BLOBS = {"mdl/config.json": b"{}", "mdl/weights.safetensors": b"W" * 64}
AZ_KEY = base64.b64encode(b"azure-account-key").decode()


class FakeAzureHandler(BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_GET(self):
        parsed = urlparse(self.path)
        qs = parse_qs(parsed.query)
        auth = self.headers.get("Authorization", "")
        if not auth.startswith("SharedKey testacct:"):
            self.send_response(403)
            self.end_headers()
            return
        parts = unquote(parsed.path).lstrip("/").split("/", 1)
        container, rest = parts[0], parts[1] if len(parts) > 1 else ""
        if qs.get("comp") == ["list"]:
            prefix = qs.get("prefix", [""])[0]
            body = "<EnumerationResults><Blobs>"
            for n in AZ_BLOBS:
                if n.startswith(prefix):
                    body += f"<Blob><Name>{n}</Name></Blob>"
            body += "</Blobs></EnumerationResults>"
            self.send_response(200)
            self.end_headers()
            self.wfile.write(body.encode())
            return
        if rest in AZ_BLOBS:
            self.send_response(200)
            self.end_headers()
            self.wfile.write(AZ_BLOBS[rest])
            return
        self.send_response(404)
        self.end_headers()


class TestAzure:
    def test_sharedkey_download(self, http_server, tmp_path):
        endpoint = http_server(FakeAzureHandler)
        client = AzureBlobClient(
            "testacct", endpoint=endpoint, account_key=AZ_KEY
        )
        n = client.download_prefix("container", "mdl", str(tmp_path))
        assert n == 2
        assert (tmp_path / "weights.safetensors").read_bytes() == b"W" * 64

    def test_storage_wasbs_uri(self, http_server, tmp_path, monkeypatch):
        endpoint = http_server(FakeAzureHandler)
        monkeypatch.setenv("AZURE_BLOB_ENDPOINT", endpoint)
        monkeypatch.setenv("AZURE_STORAGE_ACCESS_KEY", AZ_KEY)
        out = Storage.download(
            "wasbs://container@testacct.blob.core.windows.net/mdl",
            str(tmp_path / "out"),
        )
        assert sorted(os.listdir(out)) == ["config.json", "weights.safetensors"]


# ---------------------------------------------------------------------------
# GCS
# ---------------------------------------------------------------------------

GCS_OBJECTS = {"m/vocab.txt": b"hello\nworld\n", "m/model.bin": b"B" * 32}


class FakeGCSHandler(BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_GET(self):
        parsed = urlparse(self.path)
        qs = parse_qs(parsed.query)
        path = unquote(parsed.path)
        if path.startswith("/storage/v1/b/") and path.endswith("/o"):
            prefix = qs.get("prefix", [""])[0]
            items = [
                {"name": n} for n in GCS_OBJECTS if n.startswith(prefix)
            ]
            self.send_response(200)
            self.end_headers()
            self.wfile.write(json.dumps({"items": items}).encode())
            return
        if "/o/" in path and qs.get("alt") == ["media"]:
            name = unquote(path.split("/o/", 1)[1])
            if name in GCS_OBJECTS:
                self.send_response(200)
                self.end_headers()
                self.wfile.write(GCS_OBJECTS[name])
                return
        self.send_response(404)
        self.end_headers()


class TestGCS:
    def test_download_prefix(self, http_server, tmp_path):
        endpoint = http_server(FakeGCSHandler)
        client = GCSClient(endpoint=endpoint)
        n = client.download_prefix("bkt", "m", str(tmp_path))
        assert n == 2
        assert (tmp_path / "vocab.txt").read_bytes().startswith(b"hello")

    def test_storage_gs_uri(self, http_server, tmp_path, monkeypatch):
        endpoint = http_server(FakeGCSHandler)
        monkeypatch.setenv("GCS_API_ENDPOINT", endpoint)
        out = Storage.download("gs://bkt/m", str(tmp_path / "out"))
        assert sorted(os.listdir(out)) == ["model.bin", "vocab.txt"]


# ---------------------------------------------------------------------------
# WebHDFS
# ---------------------------------------------------------------------------

HDFS_TREE = {
    "/models/demo": [
        {"pathSuffix": "a.txt", "type": "FILE"},
        {"pathSuffix": "nested", "type": "DIRECTORY"},
    ],
    "/models/demo/nested": [{"pathSuffix": "b.txt", "type": "FILE"}],
}
HDFS_FILES = {
    "/models/demo/a.txt": b"AAA",
    "/models/demo/nested/b.txt": b"BBB",
}


class FakeHDFSHandler(BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_GET(self):
        parsed = urlparse(self.path)
        qs = parse_qs(parsed.query)
        path = unquote(parsed.path)[len("/webhdfs/v1"):]
        op = qs.get("op", [""])[0]
        if op == "LISTSTATUS" and path in HDFS_TREE:
            self.send_response(200)
            self.end_headers()
            self.wfile.write(
                json.dumps(
                    {"FileStatuses": {"FileStatus": HDFS_TREE[path]}}
                ).encode()
            )
            return
        if op == "OPEN" and path in HDFS_FILES:
            self.send_response(200)
            self.end_headers()
            self.wfile.write(HDFS_FILES[path])
            return
        self.send_response(404)
        self.end_headers()


class TestWebHDFS:
    def test_download_tree(self, http_server, tmp_path):
        nn = http_server(FakeHDFSHandler)
        client = WebHDFSClient(namenode=nn)
        n = client.download_tree("/models/demo", str(tmp_path))
        assert n == 2
        assert (tmp_path / "a.txt").read_bytes() == b"AAA"
        assert (tmp_path / "nested" / "b.txt").read_bytes() == b"BBB"

    def test_storage_hdfs_uri(self, http_server, tmp_path, monkeypatch):
        nn = http_server(FakeHDFSHandler)
        monkeypatch.setenv("HDFS_NAMENODE", nn)
        out = Storage.download("hdfs://models/demo", str(tmp_path / "out"))
        assert (tmp_path / "out" / "a.txt").exists()


# ---------------------------------------------------------------------------
# OCI registry
# ---------------------------------------------------------------------------

def _make_layer() -> bytes:
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as t:
        data = b"MODEL-WEIGHTS"
        info = tarfile.TarInfo("models/weights.bin")
        info.size = len(data)
        t.addfile(info, io.BytesIO(data))
    return buf.getvalue()


OCI_LAYER = _make_layer()
OCI_DIGEST = "sha256:" + hashlib.sha256(OCI_LAYER).hexdigest()


class FakeRegistryHandler(BaseHTTPRequestHandler):
    def log_message(self, *a):
        pass

    def do_GET(self):
        path = unquote(self.path)
        if path == "/v2/acme/models/manifests/v1":
            man = {
                "schemaVersion": 2,
                "layers": [
                    {
                        "mediaType": "application/vnd.oci.image.layer.v1.tar+gzip",
                        "digest": OCI_DIGEST,
                        "size": len(OCI_LAYER),
                    }
                ],
            }
            self.send_response(200)
            self.end_headers()
            self.wfile.write(json.dumps(man).encode())
            return
        if path == f"/v2/acme/models/blobs/{OCI_DIGEST}":
            self.send_response(200)
            self.end_headers()
            self.wfile.write(OCI_LAYER)
            return
        self.send_response(404)
        self.end_headers()


class TestOCIRegistry:
    def test_pull_and_extract(self, http_server, tmp_path):
        reg = http_server(FakeRegistryHandler)
        client = OCIRegistryClient(reg)
        n = client.pull_model("acme/models", "v1", str(tmp_path))
        assert n == 1
        assert (
            tmp_path / "models" / "weights.bin"
        ).read_bytes() == b"MODEL-WEIGHTS"

    def test_storage_oci_uri(self, http_server, tmp_path, monkeypatch):
        reg = http_server(FakeRegistryHandler)
        host = reg.split("://", 1)[1]
        monkeypatch.setenv("OCI_INSECURE", "1")
        out = Storage.download(
            f"oci://{host}/acme/models:v1", str(tmp_path / "out")
        )
        assert (tmp_path / "out" / "models" / "weights.bin").exists()


def test_storage_config_env_drives_s3(http_server, tmp_path, monkeypatch):
    """storage-spec mode: STORAGE_CONFIG (the mounted storage-config
    Secret key) supplies endpoint + credentials without individual env
    vars (reference _update_with_storage_spec)."""
    endpoint = http_server(FakeS3Handler)
    monkeypatch.delenv("AWS_ENDPOINT_URL", raising=False)
    monkeypatch.delenv("AWS_ACCESS_KEY_ID", raising=False)
    monkeypatch.setenv(
        "STORAGE_CONFIG",
        json.dumps(
            {
                "type": "s3",
                "endpoint_url": endpoint,
                "access_key_id": S3_KEY_ID,
                "secret_access_key": S3_SECRET,
            }
        ),
    )
    out = Storage.download("s3://bkt/models/iris", str(tmp_path / "o"))
    assert sorted(os.listdir(out)) == ["metadata.json", "model.joblib", "sub"]


def test_storage_override_config_wins(http_server, tmp_path, monkeypatch):
    endpoint = http_server(FakeS3Handler)
    monkeypatch.setenv(
        "STORAGE_CONFIG",
        json.dumps({"endpoint_url": "http://127.0.0.1:1",
                    "access_key_id": S3_KEY_ID,
                    "secret_access_key": S3_SECRET}),
    )
    monkeypatch.setenv(
        "STORAGE_OVERRIDE_CONFIG", json.dumps({"endpoint_url": endpoint})
    )
    out = Storage.download("s3://bkt/models/iris", str(tmp_path / "o2"))
    assert (tmp_path / "o2" / "model.joblib").exists()
