"""Cluster config system: the ``inferenceservice-config`` ConfigMap parsed
into typed sections.

Reference parity: pkg/apis/serving/v1beta1/configmap.go — keys (:39-52):
ingress, deploy, localModel, storageInitializer, autoscaler, service,
resource, security, explainers, multiNode, opentelemetryCollector; structs
IngressConfig (:116-136), DeployConfig (:139-142), LocalModelConfig
(:162-170); loader GetInferenceServiceConfigMap (:192-199) reads the
ConfigMap from the controller namespace.

Each section is stored as a JSON string value in the ConfigMap ``data``;
missing sections fall back to defaults, so a cluster with no ConfigMap at
all still reconciles (same as the reference's NewInferenceServicesConfig).
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, Optional

CONFIGMAP_NAME = "inferenceservice-config"
CONFIGMAP_GVK = "v1/ConfigMap"


@dataclass
class IngressConfig:
    """configmap.go IngressConfig (:116-136)."""

    ingress_gateway: str = "kserve/kserve-ingress-gateway"
    ingress_class_name: str = "istio"
    ingress_domain: str = "example.com"
    domain_template: str = "{{ .Name }}-{{ .Namespace }}.{{ .IngressDomain }}"
    path_template: str = ""
    url_scheme: str = "http"
    disable_istio_virtual_host: bool = False
    disable_ingress_creation: bool = False
    enable_gateway_api: bool = False


@dataclass
class DeployConfig:
    """configmap.go DeployConfig (:139-142)."""

    default_deployment_mode: str = "RawDeployment"  # or Serverless
    # cluster supports native OCI ImageVolumes (drives the advisory
    # condition, reference controller.go:774-831)
    image_volume_available: bool = False


@dataclass
class StorageInitializerConfig:
    image: str = "kserve-amd/storage-initializer:latest"
    cpu_request: str = "100m"
    cpu_limit: str = "1"
    memory_request: str = "100Mi"
    memory_limit: str = "1Gi"
    enable_modelcar: bool = True
    uid_modelcar: Optional[int] = None


@dataclass
class AgentConfig:
    image: str = "kserve-amd/agent:latest"
    cpu_request: str = "100m"
    cpu_limit: str = "1"
    memory_request: str = "100Mi"
    memory_limit: str = "1Gi"


@dataclass
class LocalModelConfig:
    """configmap.go LocalModelConfig (:162-170)."""

    enabled: bool = False
    job_namespace: str = "kserve-localmodel-jobs"
    default_job_image: str = "kserve-amd/storage-initializer:latest"
    fs_group: Optional[int] = None
    job_ttl_seconds: int = 3600
    reconcile_interval_s: int = 60


@dataclass
class AutoscalerConfig:
    autoscaler_class: str = "hpa"  # hpa | keda | external


@dataclass
class ServiceConfig:
    service_cluster_ip_none: bool = False


@dataclass
class InferenceServicesConfig:
    ingress: IngressConfig = field(default_factory=IngressConfig)
    deploy: DeployConfig = field(default_factory=DeployConfig)
    storage_initializer: StorageInitializerConfig = field(
        default_factory=StorageInitializerConfig
    )
    agent: AgentConfig = field(default_factory=AgentConfig)
    local_model: LocalModelConfig = field(default_factory=LocalModelConfig)
    autoscaler: AutoscalerConfig = field(default_factory=AutoscalerConfig)
    service: ServiceConfig = field(default_factory=ServiceConfig)


_SECTIONS = {
    "ingress": IngressConfig,
    "deploy": DeployConfig,
    "storageInitializer": StorageInitializerConfig,
    "agent": AgentConfig,
    "localModel": LocalModelConfig,
    "autoscaler": AutoscalerConfig,
    "service": ServiceConfig,
}

_ATTR = {
    "ingress": "ingress",
    "deploy": "deploy",
    "storageInitializer": "storage_initializer",
    "agent": "agent",
    "localModel": "local_model",
    "autoscaler": "autoscaler",
    "service": "service",
}

# camelCase JSON keys <-> snake_case dataclass fields
def _snake(k: str) -> str:
    out = []
    for ch in k:
        if ch.isupper():
            out.append("_")
            out.append(ch.lower())
        else:
            out.append(ch)
    return "".join(out)


def _parse_section(cls, raw: str):
    try:
        data = json.loads(raw)
    except json.JSONDecodeError as e:
        raise ValueError(f"invalid JSON in config section: {e}")
    obj = cls()
    for k, v in data.items():
        attr = _snake(k)
        if hasattr(obj, attr):
            setattr(obj, attr, v)
    return obj


def parse_config(data: Dict[str, str]) -> InferenceServicesConfig:
    """Parse ConfigMap ``data`` (section name -> JSON string)."""
    cfg = InferenceServicesConfig()
    for section, cls in _SECTIONS.items():
        raw = data.get(section)
        if raw:
            setattr(cfg, _ATTR[section], _parse_section(cls, raw))
    return cfg


def load_config(server, namespace: str = "kserve") -> InferenceServicesConfig:
    """GetInferenceServiceConfigMap equivalent: read + parse, defaults when
    absent (so controllers work on a bare cluster)."""
    cm = server.try_get(CONFIGMAP_GVK, namespace, CONFIGMAP_NAME)
    if cm is None:
        return InferenceServicesConfig()
    return parse_config(cm.get("data", {}) or {})


def render_domain(cfg: IngressConfig, name: str, namespace: str) -> str:
    """domain templating (reference reconcilers/ingress/domain.go)."""
    return (
        cfg.domain_template.replace("{{ .Name }}", name)
        .replace("{{ .Namespace }}", namespace)
        .replace("{{ .IngressDomain }}", cfg.ingress_domain)
        .replace("{{.Name}}", name)
        .replace("{{.Namespace}}", namespace)
        .replace("{{.IngressDomain}}", cfg.ingress_domain)
    )
