"""Model artifact download by URI scheme.

Reference parity: python/storage/kserve_storage/kserve_storage.py:62-103
(scheme dispatch), :317-423 (Storage.download), :295 (download_files).
Fully implemented offline-capable providers: file://, pvc://, http(s)://
(with tar/zip unpack), hf:// (hub snapshot via huggingface_hub). Cloud
providers (s3/gs/azure/hdfs/oci) implement the same interface and raise a
clear error when their SDK or network is unavailable in this image.
"""

from __future__ import annotations

import glob
import json
import os
import re
import shutil
import tarfile
import tempfile
import zipfile
from typing import List, Optional
from urllib.parse import urlparse

from kserve_amd.constants import MODEL_MOUNT_PATH, PVC_MOUNT_PATH
from kserve_amd.logging import logger

_LOCAL_PREFIX = "file://"


class Storage:
    @staticmethod
    def download(uri: str, out_dir: Optional[str] = None) -> str:
        """Download the artifact(s) at ``uri`` into ``out_dir`` (defaults to
        a temp dir; the pod contract uses /mnt/models)."""
        logger.info("Copying contents of %s to local", uri)
        if out_dir is None:
            out_dir = tempfile.mkdtemp()
        os.makedirs(out_dir, exist_ok=True)

        scheme = urlparse(uri).scheme
        if uri.startswith(_LOCAL_PREFIX) or os.path.exists(uri):
            return Storage._download_local(uri.replace(_LOCAL_PREFIX, "", 1), out_dir)
        if scheme == "pvc":
            return Storage._download_pvc(uri, out_dir)
        if scheme in ("http", "https"):
            return Storage._download_http(uri, out_dir)
        if scheme == "hf":
            return Storage._download_hf(uri, out_dir)
        if scheme == "s3":
            return Storage._download_s3(uri, out_dir)
        if scheme == "gs":
            return Storage._download_gcs(uri, out_dir)
        if scheme in ("wasb", "wasbs", "https+azure"):
            return Storage._download_azure(uri, out_dir)
        if scheme in ("hdfs", "webhdfs"):
            return Storage._download_hdfs(uri, out_dir)
        if scheme in ("oci", "oci+native"):
            return Storage._download_oci(uri, out_dir)
        raise ValueError(
            f"Cannot recognize storage type for {uri}; "
            "supported: file, pvc, http(s), hf, s3, gs, wasb, hdfs, oci"
        )

    @staticmethod
    def download_files(pairs: List[str]) -> None:
        """storage-initializer entry: alternating (src_uri, dest) args
        (reference initializer-entrypoint:1-50)."""
        if len(pairs) % 2 != 0:
            raise ValueError("download_files expects (src, dest) pairs")
        for i in range(0, len(pairs), 2):
            Storage.download(pairs[i], pairs[i + 1])

    # -- providers ----------------------------------------------------------
    @staticmethod
    def _download_local(path: str, out_dir: str) -> str:
        if not os.path.exists(path):
            raise FileNotFoundError(path)
        if os.path.isdir(path):
            for f in glob.glob(os.path.join(path, "*")):
                dest = os.path.join(out_dir, os.path.basename(f))
                if os.path.isdir(f):
                    shutil.copytree(f, dest, dirs_exist_ok=True)
                else:
                    shutil.copy2(f, dest)
        else:
            shutil.copy2(path, os.path.join(out_dir, os.path.basename(path)))
        return out_dir

    @staticmethod
    def _download_pvc(uri: str, out_dir: str) -> str:
        # pvc://{pvc-name}/{path} -> mounted at /mnt/pvc/{pvc-name}/{path}
        parsed = urlparse(uri)
        path = os.path.join(PVC_MOUNT_PATH, parsed.netloc, parsed.path.lstrip("/"))
        return Storage._download_local(path, out_dir)

    @staticmethod
    def _download_http(uri: str, out_dir: str) -> str:
        import requests

        name = os.path.basename(urlparse(uri).path) or "model"
        headers = {}
        for k, v in os.environ.items():
            m = re.match(r"^HEADERS_(.+)$", k)
            if m:
                headers[m.group(1).replace("_", "-")] = v
        with requests.get(uri, stream=True, headers=headers) as r:
            r.raise_for_status()
            target = os.path.join(out_dir, name)
            with open(target, "wb") as f:
                for chunk in r.iter_content(chunk_size=4 << 20):
                    f.write(chunk)
        Storage._maybe_unpack(target, out_dir)
        return out_dir

    @staticmethod
    def _maybe_unpack(path: str, out_dir: str) -> None:
        if path.endswith((".tar.gz", ".tgz")):
            with tarfile.open(path) as t:
                # filter="data" blocks tar-slip path traversal (absolute
                # paths, "..", symlink escapes) from untrusted archives
                # fetched over http(s) — mirrors the reference's
                # kserve_storage extraction hardening.
                t.extractall(out_dir, filter="data")
            os.remove(path)
        elif path.endswith(".zip"):
            with zipfile.ZipFile(path) as z:
                for member in z.namelist():
                    dest = os.path.realpath(os.path.join(out_dir, member))
                    if not dest.startswith(os.path.realpath(out_dir) + os.sep):
                        raise ValueError(
                            f"zip member escapes extraction dir: {member!r}"
                        )
                z.extractall(out_dir)
            os.remove(path)

    @staticmethod
    def _download_hf(uri: str, out_dir: str) -> str:
        # hf://{repo-id}[:revision]
        from huggingface_hub import snapshot_download

        parsed = urlparse(uri)
        repo = parsed.netloc + parsed.path
        revision = None
        if ":" in repo:
            repo, revision = repo.rsplit(":", 1)
        snapshot_download(
            repo_id=repo,
            revision=revision,
            local_dir=out_dir,
            token=os.environ.get("HF_TOKEN"),
        )
        return out_dir

    @staticmethod
    def _download_s3(uri: str, out_dir: str) -> str:
        try:
            import boto3  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "s3:// download requires boto3, which is not installed in "
                "this offline image"
            ) from e
        import boto3

        parsed = urlparse(uri)
        bucket = parsed.netloc
        prefix = parsed.path.lstrip("/")
        kwargs = {}
        if os.environ.get("AWS_ENDPOINT_URL") or os.environ.get("S3_ENDPOINT"):
            kwargs["endpoint_url"] = os.environ.get("AWS_ENDPOINT_URL") or os.environ.get("S3_ENDPOINT")
        s3 = boto3.client("s3", **kwargs)
        paginator = s3.get_paginator("list_objects_v2")
        count = 0
        for page in paginator.paginate(Bucket=bucket, Prefix=prefix):
            for obj in page.get("Contents", []):
                key = obj["Key"]
                rel = key[len(prefix):].lstrip("/") if key != prefix else os.path.basename(key)
                target = os.path.join(out_dir, rel or os.path.basename(key))
                os.makedirs(os.path.dirname(target) or out_dir, exist_ok=True)
                s3.download_file(bucket, key, target)
                count += 1
        if count == 0:
            raise FileNotFoundError(f"No objects under {uri}")
        return out_dir

    @staticmethod
    def _download_gcs(uri: str, out_dir: str) -> str:
        try:
            from google.cloud import storage as gcs  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "gs:// download requires google-cloud-storage, not installed "
                "in this offline image"
            ) from e
        from google.cloud import storage as gcs

        parsed = urlparse(uri)
        client = gcs.Client()
        bucket = client.bucket(parsed.netloc)
        prefix = parsed.path.lstrip("/")
        count = 0
        for blob in bucket.list_blobs(prefix=prefix):
            rel = blob.name[len(prefix):].lstrip("/") or os.path.basename(blob.name)
            target = os.path.join(out_dir, rel)
            os.makedirs(os.path.dirname(target) or out_dir, exist_ok=True)
            blob.download_to_filename(target)
            count += 1
        if count == 0:
            raise FileNotFoundError(f"No objects under {uri}")
        return out_dir

    @staticmethod
    def _download_azure(uri: str, out_dir: str) -> str:
        try:
            from azure.storage.blob import BlobServiceClient  # noqa: F401
        except ImportError as e:
            raise RuntimeError(
                "azure blob download requires azure-storage-blob, not "
                "installed in this offline image"
            ) from e
        raise NotImplementedError("azure blob provider: SDK present but no network in this image")

    @staticmethod
    def _download_hdfs(uri: str, out_dir: str) -> str:
        raise RuntimeError(
            "hdfs:// download requires the hdfs client, not installed in "
            "this offline image"
        )

    @staticmethod
    def _download_oci(uri: str, out_dir: str) -> str:
        raise RuntimeError(
            "oci:// model images are delivered by the modelcar sidecar / "
            "ImageVolume in-cluster (SURVEY.md §7.1); direct registry pull "
            "requires network access"
        )
