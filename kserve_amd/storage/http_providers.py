"""Native HTTP cloud-storage providers — no vendor SDKs.

The reference shells out to boto3 / google-cloud-storage / azure-storage-blob
(python/storage/kserve_storage/kserve_storage.py:62-103) — none of which
exist in this offline image. Every one of those services is an HTTP API, so
this module implements the protocols directly over ``requests``:

- S3: ListObjectsV2 + GetObject with hand-rolled AWS Signature V4 (hmac/
  hashlib stdlib), anonymous mode, custom endpoints (MinIO et al.), and the
  reference's parallel-worker download pool (kserve_storage.py:637-681)
- Azure Blob: List Blobs + Get Blob with SharedKey signing or SAS/anonymous
- GCS: JSON API (storage/v1) with bearer-token or anonymous access
- WebHDFS: LISTSTATUS + OPEN
- OCI registry: v2 manifest fetch + layer blob download with tar extraction
  (modelcar-less path of kserve_storage.py oci:// mode :189-245)

Everything is exercised offline against in-process fake servers in
tests/test_http_providers.py.
"""

from __future__ import annotations

import base64
import datetime
import hashlib
import hmac
import json
import os
import tarfile
import tempfile
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, List, Optional, Tuple
from urllib.parse import quote, urlparse
from xml.etree import ElementTree

import requests

DOWNLOAD_WORKERS = int(os.environ.get("KSERVE_DOWNLOAD_WORKERS", "8"))
EMPTY_SHA256 = hashlib.sha256(b"").hexdigest()
# (connect, read) timeouts: a wedged object store must fail the download,
# not hang the storage-initializer forever
REQUEST_TIMEOUT = (
    float(os.environ.get("KSERVE_STORAGE_CONNECT_TIMEOUT_S", "10")),
    float(os.environ.get("KSERVE_STORAGE_READ_TIMEOUT_S", "600")),
)


# ---------------------------------------------------------------------------
# AWS Signature V4 (S3)
# ---------------------------------------------------------------------------

def _hmac_sha256(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(
    method: str,
    host: str,
    path: str,
    query: List[Tuple[str, str]],
    region: str,
    access_key: str,
    secret_key: str,
    payload_hash: str = EMPTY_SHA256,
    service: str = "s3",
    now: Optional[datetime.datetime] = None,
) -> Dict[str, str]:
    """AWS SigV4 request headers (Authorization + x-amz-date +
    x-amz-content-sha256). Stdlib-only."""
    t = now or datetime.datetime.utcnow()
    amz_date = t.strftime("%Y%m%dT%H%M%SZ")
    datestamp = t.strftime("%Y%m%d")
    canonical_uri = quote(path, safe="/")
    q = sorted((quote(k, safe="~"), quote(v, safe="~")) for k, v in query)
    canonical_query = "&".join(f"{k}={v}" for k, v in q)
    headers = {
        "host": host,
        "x-amz-content-sha256": payload_hash,
        "x-amz-date": amz_date,
    }
    signed = ";".join(sorted(headers))
    canonical_headers = "".join(
        f"{k}:{headers[k]}\n" for k in sorted(headers)
    )
    canonical_request = "\n".join(
        [method, canonical_uri, canonical_query, canonical_headers, signed,
         payload_hash]
    )
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join(
        [
            "AWS4-HMAC-SHA256",
            amz_date,
            scope,
            hashlib.sha256(canonical_request.encode()).hexdigest(),
        ]
    )
    k = _hmac_sha256(("AWS4" + secret_key).encode(), datestamp)
    k = _hmac_sha256(k, region)
    k = _hmac_sha256(k, service)
    k = _hmac_sha256(k, "aws4_request")
    signature = hmac.new(k, string_to_sign.encode(), hashlib.sha256).hexdigest()
    return {
        "Authorization": (
            f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
            f"SignedHeaders={signed}, Signature={signature}"
        ),
        "x-amz-date": amz_date,
        "x-amz-content-sha256": payload_hash,
    }


class S3Client:
    """Minimal S3 over HTTP: path-style addressing against a configurable
    endpoint (AWS_ENDPOINT_URL / S3_ENDPOINT env, default AWS)."""

    def __init__(
        self,
        endpoint: Optional[str] = None,
        region: Optional[str] = None,
        access_key: Optional[str] = None,
        secret_key: Optional[str] = None,
        anonymous: Optional[bool] = None,
        verify_ssl: Optional[bool] = None,
        session: Optional[requests.Session] = None,
    ):
        self.endpoint = (
            endpoint
            or os.environ.get("AWS_ENDPOINT_URL")
            or (
                ("https://" + os.environ["S3_ENDPOINT"])
                if os.environ.get("S3_ENDPOINT")
                and "://" not in os.environ["S3_ENDPOINT"]
                else os.environ.get("S3_ENDPOINT")
            )
            or "https://s3.amazonaws.com"
        )
        self.region = region or os.environ.get("AWS_DEFAULT_REGION", "us-east-1")
        self.access_key = access_key or os.environ.get("AWS_ACCESS_KEY_ID", "")
        self.secret_key = secret_key or os.environ.get("AWS_SECRET_ACCESS_KEY", "")
        if anonymous is None:
            anonymous = (
                os.environ.get("AWS_ANONYMOUS_CREDENTIAL", "").lower()
                in ("1", "true")
                or not self.access_key
            )
        self.anonymous = anonymous
        if verify_ssl is None:
            verify_ssl = os.environ.get("S3_VERIFY_SSL", "1").lower() not in (
                "0", "false",
            )
        self.verify_ssl = verify_ssl
        self.http = session or requests.Session()

    def _request(self, method: str, path: str,
                 query: List[Tuple[str, str]],
                 body: Optional[bytes] = None) -> requests.Response:
        parsed = urlparse(self.endpoint)
        host = parsed.netloc
        headers = {}
        if not self.anonymous:
            payload_hash = (
                hashlib.sha256(body).hexdigest() if body is not None
                else EMPTY_SHA256
            )
            headers = sigv4_headers(
                method, host, path, query, self.region,
                self.access_key, self.secret_key,
                payload_hash=payload_hash,
            )
        qs = "&".join(
            f"{quote(k, safe='~')}={quote(v, safe='~')}" for k, v in sorted(query)
        )
        url = f"{self.endpoint}{quote(path, safe='/')}" + (f"?{qs}" if qs else "")
        r = self.http.request(
            method, url, headers=headers, data=body,
            verify=self.verify_ssl, stream=True, timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(
                f"S3 {method} {path} failed: {r.status_code} {r.text[:300]}"
            )
        return r

    def put_object(self, bucket: str, key: str, body: bytes) -> None:
        """PutObject (used by the payload-logger blob sink)."""
        self._request("PUT", f"/{bucket}/{key}", [], body=body)

    def list_objects(self, bucket: str, prefix: str) -> List[str]:
        """ListObjectsV2 with continuation."""
        keys: List[str] = []
        token: Optional[str] = None
        while True:
            query = [("list-type", "2"), ("prefix", prefix)]
            if token:
                query.append(("continuation-token", token))
            r = self._request("GET", f"/{bucket}", query)
            root = ElementTree.fromstring(r.content)
            ns = ""
            if root.tag.startswith("{"):
                ns = root.tag[: root.tag.index("}") + 1]
            for c in root.findall(f"{ns}Contents"):
                keys.append(c.find(f"{ns}Key").text)
            trunc = root.find(f"{ns}IsTruncated")
            if trunc is not None and trunc.text == "true":
                token = root.find(f"{ns}NextContinuationToken").text
            else:
                return keys

    def download_file(self, bucket: str, key: str, target: str) -> None:
        r = self._request("GET", f"/{bucket}/{key}", [])
        os.makedirs(os.path.dirname(target) or ".", exist_ok=True)
        with open(target, "wb") as f:
            for chunk in r.iter_content(chunk_size=4 << 20):
                f.write(chunk)

    def download_prefix(self, bucket: str, prefix: str, out_dir: str) -> int:
        """Parallel-worker pool download of every object under prefix
        (reference kserve_storage.py:637-681)."""
        keys = self.list_objects(bucket, prefix)
        keys = [k for k in keys if not k.endswith("/")]
        if not keys:
            raise FileNotFoundError(f"No objects under s3://{bucket}/{prefix}")

        def rel(key: str) -> str:
            r = key[len(prefix):].lstrip("/") if key != prefix else ""
            return r or os.path.basename(key)

        with ThreadPoolExecutor(max_workers=DOWNLOAD_WORKERS) as pool:
            futs = [
                pool.submit(
                    self.download_file, bucket, k, os.path.join(out_dir, rel(k))
                )
                for k in keys
            ]
            for f in futs:
                f.result()
        return len(keys)


# ---------------------------------------------------------------------------
# Azure Blob
# ---------------------------------------------------------------------------

class AzureBlobClient:
    """Azure Blob REST: SharedKey (account key), SAS token, or anonymous."""

    def __init__(
        self,
        account: str,
        endpoint: Optional[str] = None,
        account_key: Optional[str] = None,
        sas_token: Optional[str] = None,
        session: Optional[requests.Session] = None,
    ):
        self.account = account
        self.endpoint = (
            endpoint
            or os.environ.get("AZURE_BLOB_ENDPOINT")
            or f"https://{account}.blob.core.windows.net"
        )
        self.account_key = account_key or os.environ.get("AZURE_STORAGE_ACCESS_KEY")
        self.sas_token = sas_token or os.environ.get("AZURE_STORAGE_SAS_TOKEN")
        self.http = session or requests.Session()

    def _auth_headers(self, method: str, path: str,
                      query: List[Tuple[str, str]]) -> Dict[str, str]:
        if not self.account_key:
            return {}
        now = datetime.datetime.utcnow().strftime("%a, %d %b %Y %H:%M:%S GMT")
        headers = {"x-ms-date": now, "x-ms-version": "2021-08-06"}
        canon_headers = "".join(
            f"{k}:{headers[k]}\n" for k in sorted(headers)
        )
        canon_resource = f"/{self.account}{path}"
        for k, v in sorted(query):
            canon_resource += f"\n{k.lower()}:{v}"
        string_to_sign = "\n".join(
            [method, "", "", "", "", "", "", "", "", "", "", "",
             canon_headers + canon_resource]
        )
        key = base64.b64decode(self.account_key)
        sig = base64.b64encode(
            hmac.new(key, string_to_sign.encode(), hashlib.sha256).digest()
        ).decode()
        headers["Authorization"] = f"SharedKey {self.account}:{sig}"
        return headers

    def _request(self, method: str, path: str,
                 query: List[Tuple[str, str]]) -> requests.Response:
        headers = self._auth_headers(method, path, query)
        q = list(query)
        if self.sas_token:
            q += [
                tuple(kv.split("=", 1))
                for kv in self.sas_token.lstrip("?").split("&")
                if "=" in kv
            ]
        qs = "&".join(f"{quote(k, safe='~')}={quote(v, safe='~')}" for k, v in q)
        url = f"{self.endpoint}{quote(path, safe='/')}" + (f"?{qs}" if qs else "")
        r = self.http.request(
            method, url, headers=headers, stream=True,
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(
                f"Azure {method} {path} failed: {r.status_code} {r.text[:300]}"
            )
        return r

    def list_blobs(self, container: str, prefix: str) -> List[str]:
        r = self._request(
            "GET",
            f"/{container}",
            [("restype", "container"), ("comp", "list"), ("prefix", prefix)],
        )
        root = ElementTree.fromstring(r.content)
        return [
            b.find("Name").text
            for b in root.iter("Blob")
            if b.find("Name") is not None
        ]

    def download_blob(self, container: str, name: str, target: str) -> None:
        r = self._request("GET", f"/{container}/{name}", [])
        os.makedirs(os.path.dirname(target) or ".", exist_ok=True)
        with open(target, "wb") as f:
            for chunk in r.iter_content(chunk_size=4 << 20):
                f.write(chunk)

    def download_prefix(self, container: str, prefix: str, out_dir: str) -> int:
        names = self.list_blobs(container, prefix)
        if not names:
            raise FileNotFoundError(
                f"No blobs under {container}/{prefix} in {self.account}"
            )

        def rel(n: str) -> str:
            r = n[len(prefix):].lstrip("/") if n != prefix else ""
            return r or os.path.basename(n)

        with ThreadPoolExecutor(max_workers=DOWNLOAD_WORKERS) as pool:
            futs = [
                pool.submit(
                    self.download_blob, container, n,
                    os.path.join(out_dir, rel(n)),
                )
                for n in names
            ]
            for f in futs:
                f.result()
        return len(names)


# ---------------------------------------------------------------------------
# GCS (JSON API)
# ---------------------------------------------------------------------------

class GCSClient:
    """GCS JSON API: bearer token (GCS_OAUTH_TOKEN env) or anonymous
    (public buckets)."""

    def __init__(self, endpoint: Optional[str] = None,
                 token: Optional[str] = None,
                 session: Optional[requests.Session] = None):
        self.endpoint = (
            endpoint
            or os.environ.get("GCS_API_ENDPOINT")
            or "https://storage.googleapis.com"
        )
        self.token = token or os.environ.get("GCS_OAUTH_TOKEN")
        self.http = session or requests.Session()

    def _headers(self) -> Dict[str, str]:
        return {"Authorization": f"Bearer {self.token}"} if self.token else {}

    def list_objects(self, bucket: str, prefix: str) -> List[str]:
        names: List[str] = []
        page: Optional[str] = None
        while True:
            params = {"prefix": prefix}
            if page:
                params["pageToken"] = page
            r = self.http.get(
                f"{self.endpoint}/storage/v1/b/{bucket}/o",
                params=params,
                headers=self._headers(),
                timeout=REQUEST_TIMEOUT,
            )
            if r.status_code >= 400:
                raise RuntimeError(f"GCS list failed: {r.status_code}")
            data = r.json()
            names += [i["name"] for i in data.get("items", [])]
            page = data.get("nextPageToken")
            if not page:
                return names

    def download_object(self, bucket: str, name: str, target: str) -> None:
        r = self.http.get(
            f"{self.endpoint}/storage/v1/b/{bucket}/o/{quote(name, safe='')}",
            params={"alt": "media"},
            headers=self._headers(),
            stream=True,
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(f"GCS get {name} failed: {r.status_code}")
        os.makedirs(os.path.dirname(target) or ".", exist_ok=True)
        with open(target, "wb") as f:
            for chunk in r.iter_content(chunk_size=4 << 20):
                f.write(chunk)

    def download_prefix(self, bucket: str, prefix: str, out_dir: str) -> int:
        names = self.list_objects(bucket, prefix)
        if not names:
            raise FileNotFoundError(f"No objects under gs://{bucket}/{prefix}")

        def rel(n: str) -> str:
            r = n[len(prefix):].lstrip("/") if n != prefix else ""
            return r or os.path.basename(n)

        with ThreadPoolExecutor(max_workers=DOWNLOAD_WORKERS) as pool:
            futs = [
                pool.submit(
                    self.download_object, bucket, n,
                    os.path.join(out_dir, rel(n)),
                )
                for n in names
            ]
            for f in futs:
                f.result()
        return len(names)


# ---------------------------------------------------------------------------
# WebHDFS
# ---------------------------------------------------------------------------

class WebHDFSClient:
    """WebHDFS REST (LISTSTATUS / OPEN). hdfs:// URIs resolve through the
    HDFS_NAMENODE env (http address of the namenode)."""

    def __init__(self, namenode: Optional[str] = None,
                 user: Optional[str] = None,
                 session: Optional[requests.Session] = None):
        self.namenode = (namenode or os.environ.get("HDFS_NAMENODE", "")).rstrip("/")
        if not self.namenode:
            raise RuntimeError("webhdfs requires HDFS_NAMENODE")
        self.user = user or os.environ.get("HDFS_USER")
        self.http = session or requests.Session()

    def _params(self, op: str) -> Dict[str, str]:
        p = {"op": op}
        if self.user:
            p["user.name"] = self.user
        return p

    def list_status(self, path: str) -> List[Dict]:
        r = self.http.get(
            f"{self.namenode}/webhdfs/v1{path}",
            params=self._params("LISTSTATUS"),
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(f"webhdfs LISTSTATUS {path}: {r.status_code}")
        return r.json()["FileStatuses"]["FileStatus"]

    def open(self, path: str, target: str) -> None:
        r = self.http.get(
            f"{self.namenode}/webhdfs/v1{path}",
            params=self._params("OPEN"),
            stream=True,
            allow_redirects=True,
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(f"webhdfs OPEN {path}: {r.status_code}")
        os.makedirs(os.path.dirname(target) or ".", exist_ok=True)
        with open(target, "wb") as f:
            for chunk in r.iter_content(chunk_size=4 << 20):
                f.write(chunk)

    def download_tree(self, path: str, out_dir: str) -> int:
        count = 0
        for st in self.list_status(path):
            name = st["pathSuffix"]
            sub = f"{path.rstrip('/')}/{name}" if name else path
            if st["type"] == "DIRECTORY":
                count += self.download_tree(
                    sub, os.path.join(out_dir, name)
                )
            else:
                self.open(sub, os.path.join(out_dir, name or os.path.basename(path)))
                count += 1
        return count


# ---------------------------------------------------------------------------
# OCI registry (distribution v2)
# ---------------------------------------------------------------------------

class OCIRegistryClient:
    """Pull model layers from an OCI registry: GET manifest, download layer
    blobs, extract tars under out_dir (reference kserve_storage.py oci
    mode :189-245 — layer modes: every non-config layer is extracted)."""

    MANIFEST_TYPES = (
        "application/vnd.oci.image.manifest.v1+json, "
        "application/vnd.docker.distribution.manifest.v2+json"
    )

    def __init__(self, registry: str, insecure: bool = False,
                 token: Optional[str] = None,
                 session: Optional[requests.Session] = None):
        scheme = "http" if insecure else "https"
        self.base = f"{scheme}://{registry}" if "://" not in registry else registry
        self.token = token
        self.http = session or requests.Session()

    def _headers(self, accept: Optional[str] = None) -> Dict[str, str]:
        h: Dict[str, str] = {}
        if accept:
            h["Accept"] = accept
        if self.token:
            h["Authorization"] = f"Bearer {self.token}"
        return h

    def manifest(self, name: str, reference: str) -> Dict:
        r = self.http.get(
            f"{self.base}/v2/{name}/manifests/{reference}",
            headers=self._headers(self.MANIFEST_TYPES),
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(
                f"OCI manifest {name}:{reference}: {r.status_code}"
            )
        return r.json()

    def blob(self, name: str, digest: str, target: str) -> None:
        r = self.http.get(
            f"{self.base}/v2/{name}/blobs/{digest}",
            headers=self._headers(),
            stream=True,
            timeout=REQUEST_TIMEOUT,
        )
        if r.status_code >= 400:
            raise RuntimeError(f"OCI blob {digest}: {r.status_code}")
        os.makedirs(os.path.dirname(target) or ".", exist_ok=True)
        with open(target, "wb") as f:
            for chunk in r.iter_content(chunk_size=4 << 20):
                f.write(chunk)

    def pull_model(self, name: str, reference: str, out_dir: str) -> int:
        man = self.manifest(name, reference)
        layers = man.get("layers", [])
        n = 0
        for layer in layers:
            digest = layer["digest"]
            with tempfile.NamedTemporaryFile(suffix=".tar", delete=False) as tmp:
                tmp_path = tmp.name
            try:
                self.blob(name, digest, tmp_path)
                # "r:*" auto-detects gzip vs plain tar from the payload
                with tarfile.open(tmp_path, "r:*") as t:
                    t.extractall(out_dir, filter="data")
                n += 1
            finally:
                os.unlink(tmp_path)
        if n == 0:
            raise FileNotFoundError(f"OCI image {name}:{reference} has no layers")
        return n
