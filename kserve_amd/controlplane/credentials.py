"""Credentials builder: storage Secrets / ServiceAccounts → env vars and
volume mounts on the storage-initializer init container.

Reference parity: pkg/credentials — CredentialBuilder.CreateSecretVolumeAndEnv
(service_account_credentials.go:212-272) walks the pod ServiceAccount's
secrets and, per provider, turns them into env/volumes on the init
container: S3 (pkg/credentials/s3: access keys from secret data + endpoint/
region/SSL from secret annotations), GCS (service-account JSON mounted +
GOOGLE_APPLICATION_CREDENTIALS), Azure (client/tenant/secret envs or storage
access key), HDFS (hdfs-site.xml/core-site.xml volume), HF token, HTTPS
headers. The storage-spec mode CreateStorageSpecSecretEnvs (:102-199) reads
one key of the common ``storage-config`` secret and emits STORAGE_CONFIG /
STORAGE_OVERRIDE_CONFIG envs consumed by the initializer
(kserve_storage.py _update_with_storage_spec :425).

Env names match the python storage layer (storage/storage.py) so the
injected init container picks them up without translation.
"""

from __future__ import annotations

import base64
import json
from typing import Dict, List, Optional, Tuple

# S3 secret annotations (reference pkg/credentials/s3/s3_secret.go)
S3_ANN_ENDPOINT = "serving.kserve.io/s3-endpoint"
S3_ANN_USE_HTTPS = "serving.kserve.io/s3-usehttps"
S3_ANN_REGION = "serving.kserve.io/s3-region"
S3_ANN_VERIFY_SSL = "serving.kserve.io/s3-verifyssl"
S3_ANN_USE_ANON = "serving.kserve.io/s3-useanoncredential"
S3_ANN_CABUNDLE = "serving.kserve.io/s3-cabundle"

GCS_CRED_FILE = "gcloud-application-credentials.json"
CRED_VOLUME = "kserve-credentials"
CRED_MOUNT = "/var/run/kserve/credentials"


def _b64get(data: Dict[str, str], key: str) -> Optional[str]:
    v = data.get(key)
    if v is None:
        return None
    try:
        return base64.b64decode(v).decode()
    except Exception:
        return v  # tolerate stringData-style plain values


def build_secret_env_and_volumes(
    secret: Dict,
) -> Tuple[List[Dict], List[Dict], List[Dict]]:
    """One Secret → (env, volumes, volume_mounts). Provider detected from
    the secret's data keys, mirroring the reference's per-provider
    builders."""
    env: List[Dict] = []
    volumes: List[Dict] = []
    mounts: List[Dict] = []
    data = secret.get("data", {}) or {}
    string_data = secret.get("stringData", {}) or {}
    merged = {**data, **{k: base64.b64encode(v.encode()).decode()
                         for k, v in string_data.items()}}
    ann = secret.get("metadata", {}).get("annotations", {}) or {}
    name = secret["metadata"]["name"]

    def secret_env(var: str, key: str) -> Dict:
        return {
            "name": var,
            "valueFrom": {
                "secretKeyRef": {"name": name, "key": key}
            },
        }

    # ---- S3 -------------------------------------------------------------
    if "AWS_ACCESS_KEY_ID" in merged or "awsAccessKeyID" in merged:
        id_key = "AWS_ACCESS_KEY_ID" if "AWS_ACCESS_KEY_ID" in merged else "awsAccessKeyID"
        sk_key = (
            "AWS_SECRET_ACCESS_KEY"
            if "AWS_SECRET_ACCESS_KEY" in merged
            else "awsSecretAccessKey"
        )
        env.append(secret_env("AWS_ACCESS_KEY_ID", id_key))
        env.append(secret_env("AWS_SECRET_ACCESS_KEY", sk_key))
        if ann.get(S3_ANN_ENDPOINT):
            use_https = ann.get(S3_ANN_USE_HTTPS, "1") != "0"
            scheme = "https" if use_https else "http"
            env.append({"name": "S3_ENDPOINT", "value": ann[S3_ANN_ENDPOINT]})
            env.append(
                {
                    "name": "AWS_ENDPOINT_URL",
                    "value": f"{scheme}://{ann[S3_ANN_ENDPOINT]}",
                }
            )
        if ann.get(S3_ANN_REGION):
            env.append({"name": "AWS_DEFAULT_REGION", "value": ann[S3_ANN_REGION]})
        if ann.get(S3_ANN_VERIFY_SSL) is not None:
            env.append(
                {"name": "S3_VERIFY_SSL", "value": ann[S3_ANN_VERIFY_SSL]}
            )
        if ann.get(S3_ANN_USE_ANON):
            env.append(
                {"name": "AWS_ANONYMOUS_CREDENTIAL", "value": ann[S3_ANN_USE_ANON]}
            )
        if ann.get(S3_ANN_CABUNDLE):
            env.append({"name": "AWS_CA_BUNDLE", "value": ann[S3_ANN_CABUNDLE]})

    # ---- GCS ------------------------------------------------------------
    if GCS_CRED_FILE in merged:
        volumes.append(
            {"name": CRED_VOLUME, "secret": {"secretName": name}}
        )
        mounts.append(
            {"name": CRED_VOLUME, "mountPath": CRED_MOUNT, "readOnly": True}
        )
        env.append(
            {
                "name": "GOOGLE_APPLICATION_CREDENTIALS",
                "value": f"{CRED_MOUNT}/{GCS_CRED_FILE}",
            }
        )

    # ---- Azure ----------------------------------------------------------
    if "AZURE_CLIENT_ID" in merged or "AZ_CLIENT_ID" in merged:
        for std, legacy in (
            ("AZURE_CLIENT_ID", "AZ_CLIENT_ID"),
            ("AZURE_CLIENT_SECRET", "AZ_CLIENT_SECRET"),
            ("AZURE_TENANT_ID", "AZ_TENANT_ID"),
            ("AZURE_SUBSCRIPTION_ID", "AZ_SUBSCRIPTION_ID"),
        ):
            key = std if std in merged else legacy
            if key in merged:
                env.append(secret_env(std, key))
    if "AZURE_STORAGE_ACCESS_KEY" in merged:
        env.append(secret_env("AZURE_STORAGE_ACCESS_KEY", "AZURE_STORAGE_ACCESS_KEY"))

    # ---- HDFS -----------------------------------------------------------
    if "HDFS_NAMENODE" in merged or "hdfs-site.xml" in merged:
        volumes.append(
            {"name": "kserve-hdfs-config", "secret": {"secretName": name}}
        )
        mounts.append(
            {
                "name": "kserve-hdfs-config",
                "mountPath": "/var/run/kserve/hdfs",
                "readOnly": True,
            }
        )
        if "HDFS_NAMENODE" in merged:
            env.append(secret_env("HDFS_NAMENODE", "HDFS_NAMENODE"))

    # ---- HF token --------------------------------------------------------
    if "HF_TOKEN" in merged:
        env.append(secret_env("HF_TOKEN", "HF_TOKEN"))

    # ---- HTTPS headers ---------------------------------------------------
    if "https-host" in merged and "headers" in merged:
        raw = _b64get(merged, "headers")
        try:
            headers = json.loads(raw) if raw else {}
        except json.JSONDecodeError:
            headers = {}
        for hk in headers:
            env.append(
                {
                    "name": f"HEADERS_{hk.replace('-', '_')}",
                    "value": str(headers[hk]),
                }
            )
    return env, volumes, mounts


class CredentialsBuilder:
    """Resolves a pod's ServiceAccount → Secrets → init-container env/volumes
    (reference CredentialBuilder, service_account_credentials.go:212-272)."""

    def __init__(self, server, namespace: str):
        self.server = server
        self.namespace = namespace

    def for_service_account(
        self, sa_name: str = "default"
    ) -> Tuple[List[Dict], List[Dict], List[Dict]]:
        sa = self.server.try_get("v1/ServiceAccount", self.namespace, sa_name)
        if sa is None:
            return [], [], []
        env: List[Dict] = []
        volumes: List[Dict] = []
        mounts: List[Dict] = []
        for ref in sa.get("secrets", []) or []:
            secret = self.server.try_get(
                "v1/Secret", self.namespace, ref.get("name", "")
            )
            if secret is None:
                continue
            e, v, m = build_secret_env_and_volumes(secret)
            env.extend(e)
            volumes.extend(v)
            mounts.extend(m)
        return env, volumes, mounts

    def storage_spec_envs(
        self, storage_key: str, storage_params: Optional[Dict] = None,
        secret_name: str = "storage-config",
    ) -> List[Dict]:
        """CreateStorageSpecSecretEnvs (:102-199): one key of the common
        storage-config Secret becomes STORAGE_CONFIG, spec parameters become
        STORAGE_OVERRIDE_CONFIG."""
        secret = self.server.try_get("v1/Secret", self.namespace, secret_name)
        if secret is None:
            raise LookupError(f"secret {secret_name} not found")
        raw = _b64get(secret.get("data", {}) or {}, storage_key)
        if raw is None:
            raise LookupError(
                f"storage key {storage_key!r} not in secret {secret_name}"
            )
        env = [{"name": "STORAGE_CONFIG", "value": raw}]
        if storage_params:
            env.append(
                {
                    "name": "STORAGE_OVERRIDE_CONFIG",
                    "value": json.dumps(storage_params, sort_keys=True),
                }
            )
        return env


def inject_credentials(pod: Dict, env: List[Dict], volumes: List[Dict],
                       mounts: List[Dict]) -> Dict:
    """Attach resolved credentials to the storage-initializer init
    container (and modelcar-less pods' kserve-container as fallback)."""
    spec = pod.get("spec", {})
    targets = [
        c for c in spec.get("initContainers", [])
        if c.get("name") == "storage-initializer"
    ]
    if not targets:
        return pod
    for c in targets:
        have = {e["name"] for e in c.setdefault("env", [])}
        c["env"].extend(e for e in env if e["name"] not in have)
        have_m = {m["name"] for m in c.setdefault("volumeMounts", [])}
        c["volumeMounts"].extend(m for m in mounts if m["name"] not in have_m)
    vols = spec.setdefault("volumes", [])
    have_v = {v["name"] for v in vols}
    vols.extend(v for v in volumes if v["name"] not in have_v)
    return pod
