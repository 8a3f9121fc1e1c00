"""Pod mutating webhook logic: sidecar/init-container injection.

Reference parity: pkg/webhook/admission/pod — mutator chain
(mutator.go:131-143): InjectStorageInitializer
(storage_initializer_injector.go:729), InjectAgent (agent_injector.go:177),
InjectBatcher (batcher_injector.go:79), InjectModelcar (:201) — as a pure
function on pod manifests driven by the internal annotations.
"""

from __future__ import annotations

import copy
from typing import Dict

from kserve_amd.constants import (
    AGENT_PORT,
    MODEL_MOUNT_PATH,
)

ANN_STORAGE_URI = "internal.serving.kserve.io/storage-initializer-sourceuri"
ANN_LOGGER = "internal.serving.kserve.io/logger"
ANN_LOGGER_URL = "internal.serving.kserve.io/logger-sink-url"
ANN_LOGGER_MODE = "internal.serving.kserve.io/logger-mode"
ANN_BATCHER = "internal.serving.kserve.io/batcher"
ANN_BATCHER_MAX_SIZE = "internal.serving.kserve.io/batcher-max-batchsize"
ANN_BATCHER_MAX_LATENCY = "internal.serving.kserve.io/batcher-max-latency"
ANN_AGENT = "internal.serving.kserve.io/agent"
ANN_METRICS_AGG = "serving.kserve.io/enable-metric-aggregation"
ANN_PROM_PORT = "serving.kserve.io/metrics-port"
ANN_PROM_PATH = "serving.kserve.io/metrics-path"
ANN_ACCELERATOR = "serving.kserve.io/accelerator"
ANN_ISTIO_CNI = "sidecar.istio.io/interceptionMode"
ANN_CA_BUNDLE = "serving.kserve.io/ca-bundle-configmap"
GPU_RESOURCE = "amd.com/gpu"


def inject_storage_initializer(pod: Dict, image: str) -> Dict:
    ann = pod["metadata"].get("annotations", {})
    uri = ann.get(ANN_STORAGE_URI)
    if not uri:
        return pod
    if uri.startswith(("oci://", "oci+native://")):
        return inject_modelcar(pod, uri)
    spec = pod["spec"]
    volumes = spec.setdefault("volumes", [])
    if not any(v["name"] == "kserve-provision-location" for v in volumes):
        volumes.append({"name": "kserve-provision-location", "emptyDir": {}})
    init = {
        "name": "storage-initializer",
        "image": image,
        "args": [uri, MODEL_MOUNT_PATH],
        "volumeMounts": [
            {"name": "kserve-provision-location", "mountPath": MODEL_MOUNT_PATH}
        ],
    }
    # pvc fast-path: mount the PVC directly (reference :44)
    if uri.startswith("pvc://"):
        pvc_name = uri[len("pvc://"):].split("/", 1)[0]
        volumes.append(
            {
                "name": "kserve-pvc-source",
                "persistentVolumeClaim": {"claimName": pvc_name},
            }
        )
        init["volumeMounts"].append(
            {"name": "kserve-pvc-source", "mountPath": "/mnt/pvc", "readOnly": True}
        )
    spec.setdefault("initContainers", []).insert(0, init)
    for c in spec["containers"]:
        if c.get("name") == "kserve-container":
            c.setdefault("volumeMounts", []).append(
                {
                    "name": "kserve-provision-location",
                    "mountPath": MODEL_MOUNT_PATH,
                    "readOnly": True,
                }
            )
    return pod


def inject_modelcar(pod: Dict, uri: str) -> Dict:
    """OCI model image as a sidecar sharing /mnt/models (reference
    storage_initializer_injector.go:201-256)."""
    image = uri.split("://", 1)[1]
    spec = pod["spec"]
    spec.setdefault("volumes", []).append(
        {"name": "kserve-provision-location", "emptyDir": {}}
    )
    spec["containers"].append(
        {
            "name": "modelcar",
            "image": image,
            "command": ["sh", "-c", f"ln -sf /models/* {MODEL_MOUNT_PATH}/ && sleep infinity"],
            "volumeMounts": [
                {"name": "kserve-provision-location", "mountPath": MODEL_MOUNT_PATH}
            ],
        }
    )
    for c in spec["containers"]:
        if c.get("name") == "kserve-container":
            c.setdefault("volumeMounts", []).append(
                {
                    "name": "kserve-provision-location",
                    "mountPath": MODEL_MOUNT_PATH,
                    "readOnly": True,
                }
            )
    return pod


def inject_agent(pod: Dict, image: str) -> Dict:
    """Agent sidecar proxy on :9081 for logger/batcher roles (reference
    agent_injector.go:177; flag surface cmd/agent/main.go:55-99)."""
    ann = pod["metadata"].get("annotations", {})
    if ann.get(ANN_AGENT) != "true":
        return pod
    args = ["--port", str(AGENT_PORT), "--component-port", "8080"]
    if ann.get(ANN_LOGGER) == "true":
        args += ["--log-mode", ann.get(ANN_LOGGER_MODE, "all")]
        if ann.get(ANN_LOGGER_URL):
            args += ["--log-url", ann[ANN_LOGGER_URL]]
    if ann.get(ANN_BATCHER) == "true":
        args += [
            "--enable-batcher",
            "--max-batchsize", ann.get(ANN_BATCHER_MAX_SIZE, "32"),
            "--max-latency", ann.get(ANN_BATCHER_MAX_LATENCY, "5000"),
        ]
    pod["spec"]["containers"].append(
        {
            "name": "agent",
            "image": image,
            "args": args,
            "ports": [{"containerPort": AGENT_PORT, "name": "agent-port"}],
        }
    )
    return pod


def inject_metrics_aggregator(pod: Dict) -> Dict:
    """qpext wiring (reference metrics_aggregate_injector.go:94, env
    :52): when aggregation is on, tell the queue-proxy extension where
    the kserve container's Prometheus endpoint lives."""
    ann = pod["metadata"].get("annotations", {})
    if ann.get(ANN_METRICS_AGG) != "true":
        return pod
    env = [
        {"name": "KSERVE_CONTAINER_PROMETHEUS_METRICS_PORT",
         "value": ann.get(ANN_PROM_PORT, "8080")},
        {"name": "KSERVE_CONTAINER_PROMETHEUS_METRICS_PATH",
         "value": ann.get(ANN_PROM_PATH, "/metrics")},
    ]
    for c in pod["spec"]["containers"]:
        if c.get("name") == "queue-proxy":
            c.setdefault("env", []).extend(env)
    # Prometheus scrapes one port per pod: point it at the aggregator
    ann.setdefault("prometheus.kserve.io/port", "9088")
    ann.setdefault("prometheus.kserve.io/path", "/metrics")
    pod["metadata"]["annotations"] = ann
    return pod


def inject_accelerator_selector(pod: Dict) -> Dict:
    """Node-pool pinning for GPU pods (reference mutator.go GKE
    accelerator injector): a pod with amd.com/gpu limits and an
    accelerator annotation gets the matching nodeSelector."""
    ann = pod["metadata"].get("annotations", {})
    acc = ann.get(ANN_ACCELERATOR)
    if not acc:
        return pod
    wants_gpu = any(
        GPU_RESOURCE in (c.get("resources", {}).get("limits") or {})
        for c in pod["spec"].get("containers", [])
    )
    if wants_gpu:
        pod["spec"].setdefault("nodeSelector", {})[
            "kserve.amd.com/accelerator"] = acc
    return pod


def inject_ca_bundle(pod: Dict) -> Dict:
    """Custom CA bundle for the storage-initializer (reference
    storage_initializer_injector.go:918): mount the named ConfigMap and
    point AWS_CA_BUNDLE/CA_BUNDLE_CONFIGMAP_NAME at it."""
    ann = pod["metadata"].get("annotations", {})
    cm = ann.get(ANN_CA_BUNDLE)
    if not cm:
        return pod
    inits = pod["spec"].get("initContainers", [])
    target = next((c for c in inits if c["name"] == "storage-initializer"),
                  None)
    if target is None:
        return pod
    pod["spec"].setdefault("volumes", []).append(
        {"name": "cabundle-cert", "configMap": {"name": cm}}
    )
    target.setdefault("volumeMounts", []).append(
        {"name": "cabundle-cert", "mountPath": "/etc/ssl/custom-certs"}
    )
    target.setdefault("env", []).extend([
        {"name": "CA_BUNDLE_CONFIGMAP_NAME", "value": cm},
        {"name": "AWS_CA_BUNDLE",
         "value": "/etc/ssl/custom-certs/cabundle.crt"},
    ])
    return pod


def inject_istio_cni_security_context(pod: Dict) -> Dict:
    """Istio CNI compatibility (reference :802): init containers must
    not run as UID 1337 (the proxy's UID) or traffic bypasses the mesh;
    pin the storage-initializer to a distinct non-root UID."""
    ann = pod["metadata"].get("annotations", {})
    if ANN_ISTIO_CNI not in ann:
        return pod
    for c in pod["spec"].get("initContainers", []):
        if c["name"] == "storage-initializer":
            sc = c.setdefault("securityContext", {})
            sc.setdefault("runAsUser", 1000)
            sc.setdefault("runAsNonRoot", True)
    return pod


def mutate_pod(
    pod: Dict,
    storage_init_image: str = "kserve-amd/storage-initializer:latest",
    agent_image: str = "kserve-amd/agent:latest",
) -> Dict:
    """Ordered mutator chain (reference mutator.go:131-143)."""
    pod = copy.deepcopy(pod)
    pod = inject_storage_initializer(pod, storage_init_image)
    pod = inject_agent(pod, agent_image)
    pod = inject_metrics_aggregator(pod)
    pod = inject_accelerator_selector(pod)
    pod = inject_ca_bundle(pod)
    pod = inject_istio_cni_security_context(pod)
    return pod


# -- ClusterStorageContainer resolution (reference
# storage_initializer_injector.go:123-199; CR types
# storage_container_types.go:29-73) ------------------------------------------

def resolve_storage_container(uri: str, containers) -> Dict:
    """First ClusterStorageContainer CR whose supportedUriFormats matches
    the storage URI (prefix or regex), or None. The matching CR's container
    spec overrides the default storage-initializer image/env/resources."""
    import re as _re

    for csc in containers or []:
        spec = csc.get("spec", {}) or {}
        for fmt in spec.get("supportedUriFormats", []) or []:
            prefix = fmt.get("prefix")
            if prefix and uri.startswith(prefix):
                return spec.get("container")
            regex = fmt.get("regex")
            if regex and _re.match(regex, uri):
                return spec.get("container")
    return None


def apply_storage_container(pod: Dict, container_spec: Dict) -> Dict:
    """Overlay a resolved CSC container spec onto the injected
    storage-initializer init container (image/env/resources; args keep the
    (uri, dest) contract)."""
    if not container_spec:
        return pod
    for init in pod.get("spec", {}).get("initContainers", []):
        if init.get("name") == "storage-initializer":
            if container_spec.get("image"):
                init["image"] = container_spec["image"]
            if container_spec.get("env"):
                have = {e["name"] for e in init.setdefault("env", [])}
                init["env"].extend(
                    e for e in container_spec["env"] if e["name"] not in have
                )
            if container_spec.get("resources"):
                init["resources"] = container_spec["resources"]
    return pod
