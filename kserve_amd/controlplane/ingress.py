"""Ingress reconcilers: the three routing backends + autoscaler breadth.

Reference parity: pkg/controller/v1beta1/inferenceservice/reconcilers/ingress —
Istio VirtualService (ingress_reconciler.go, 759 LoC), Gateway API HTTPRoute
(httproute_reconciler.go), plain k8s Ingress (kube_ingress_reconciler.go:360),
domain templating (domain.go); factory.go:CreateIngressReconciler picks the
backend from IngressConfig. KEDA ScaledObject per
reconcilers/keda/keda_reconciler.go:345.

The backend choice mirrors the factory: Serverless mode → Istio
VirtualService (unless disabled); RawDeployment + enableGatewayApi →
HTTPRoute; RawDeployment otherwise → k8s Ingress with ingressClassName.
"""

from __future__ import annotations

from typing import Dict, List, Optional

from kserve_amd.controlplane.configmap import IngressConfig, render_domain


def _top_component(has_transformer: bool) -> str:
    # requests enter through the transformer when one exists
    # (reference: ingress routes to transformer service, which forwards
    # to the predictor via --predictor_host)
    return "transformer" if has_transformer else "predictor"


def render_virtual_service(
    name: str,
    namespace: str,
    cfg: IngressConfig,
    has_transformer: bool = False,
    has_explainer: bool = False,
) -> Optional[Dict]:
    """Istio VirtualService (reference ingress_reconciler.go
    createVirtualService): external host via domain template + internal
    cluster-local host; :explain routes to the explainer service."""
    if cfg.disable_istio_virtual_host or cfg.disable_ingress_creation:
        return None
    host = render_domain(cfg, name, namespace)
    top = f"{name}-{_top_component(has_transformer)}"
    internal_host = f"{name}.{namespace}.svc.cluster.local"
    routes: List[Dict] = []
    if has_explainer:
        routes.append(
            {
                "match": [
                    {"uri": {"regex": r"^/v1/models/[\w.\-]+:explain$"}},
                    {"uri": {"regex": r"^/v2/models/[\w.\-]+/explain$"}},
                ],
                "route": [
                    {
                        "destination": {
                            "host": f"{name}-explainer.{namespace}.svc.cluster.local",
                            "port": {"number": 80},
                        },
                        "weight": 100,
                    }
                ],
            }
        )
    routes.append(
        {
            "route": [
                {
                    "destination": {
                        "host": f"{top}.{namespace}.svc.cluster.local",
                        "port": {"number": 80},
                    },
                    "weight": 100,
                }
            ]
        }
    )
    gw_ns, gw_name = (
        cfg.ingress_gateway.split("/", 1)
        if "/" in cfg.ingress_gateway
        else ("kserve", cfg.ingress_gateway)
    )
    return {
        "apiVersion": "networking.istio.io/v1beta1",
        "kind": "VirtualService",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "hosts": [host, internal_host],
            "gateways": [f"{gw_ns}/{gw_name}", "mesh"],
            "http": routes,
        },
    }


def render_kube_ingress(
    name: str,
    namespace: str,
    cfg: IngressConfig,
    has_transformer: bool = False,
) -> Optional[Dict]:
    """Plain networking.k8s.io Ingress (kube_ingress_reconciler.go:360)."""
    if cfg.disable_ingress_creation:
        return None
    host = render_domain(cfg, name, namespace)
    top = f"{name}-{_top_component(has_transformer)}"
    return {
        "apiVersion": "networking.k8s.io/v1",
        "kind": "Ingress",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "ingressClassName": cfg.ingress_class_name,
            "rules": [
                {
                    "host": host,
                    "http": {
                        "paths": [
                            {
                                "path": "/",
                                "pathType": "Prefix",
                                "backend": {
                                    "service": {
                                        "name": top,
                                        "port": {"number": 80},
                                    }
                                },
                            }
                        ]
                    },
                }
            ],
        },
    }


def render_http_route(
    name: str,
    namespace: str,
    cfg: IngressConfig,
    has_transformer: bool = False,
    traffic_split: Optional[Dict[str, int]] = None,
) -> Optional[Dict]:
    """Gateway API HTTPRoute (httproute_reconciler.go), with optional
    stable/canary weighted backends."""
    if cfg.disable_ingress_creation:
        return None
    host = render_domain(cfg, name, namespace)
    top = f"{name}-{_top_component(has_transformer)}"
    if traffic_split:
        backends = [
            {"name": top, "port": 80, "weight": traffic_split["stable"]},
            {
                "name": f"{top}-canary",
                "port": 80,
                "weight": traffic_split["canary"],
            },
        ]
    else:
        backends = [{"name": top, "port": 80}]
    gw_ns, gw_name = (
        cfg.ingress_gateway.split("/", 1)
        if "/" in cfg.ingress_gateway
        else ("kserve", cfg.ingress_gateway)
    )
    return {
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "parentRefs": [{"name": gw_name, "namespace": gw_ns}],
            "hostnames": [host],
            "rules": [
                {
                    "matches": [
                        {"path": {"type": "PathPrefix", "value": "/"}}
                    ],
                    "backendRefs": backends,
                }
            ],
        },
    }


def select_ingress(
    name: str,
    namespace: str,
    cfg: IngressConfig,
    deployment_mode: str,
    has_transformer: bool = False,
    has_explainer: bool = False,
    traffic_split: Optional[Dict[str, int]] = None,
) -> Optional[Dict]:
    """factory.go:CreateIngressReconciler backend choice."""
    if deployment_mode == "Serverless":
        return render_virtual_service(
            name, namespace, cfg, has_transformer, has_explainer
        )
    if cfg.enable_gateway_api:
        return render_http_route(
            name, namespace, cfg, has_transformer, traffic_split
        )
    return render_kube_ingress(name, namespace, cfg, has_transformer)


# -- KEDA (keda_reconciler.go:345) ------------------------------------------

_SCALE_METRIC_TO_TRIGGER = {
    # ScaleMetric -> KEDA trigger type + metadata template
    "cpu": lambda target: {
        "type": "cpu",
        "metricType": "Utilization",
        "metadata": {"value": str(target)},
    },
    "memory": lambda target: {
        "type": "memory",
        "metricType": "Utilization",
        "metadata": {"value": str(target)},
    },
}


def render_keda_scaled_object(
    name: str,
    namespace: str,
    deployment_name: str,
    min_replicas: int,
    max_replicas: int,
    scale_metric: str = "cpu",
    scale_target: int = 80,
    prometheus_server: str = "http://prometheus:9090",
    custom_query: Optional[str] = None,
) -> Dict:
    """KEDA ScaledObject mapping ScaleMetric → trigger. Resource metrics map
    to cpu/memory triggers; anything else becomes a prometheus trigger
    (reference keda_reconciler.go getTriggers)."""
    maker = _SCALE_METRIC_TO_TRIGGER.get(scale_metric)
    if maker is not None:
        trigger = maker(scale_target)
    else:
        query = custom_query or (
            f'avg({scale_metric}{{deployment="{deployment_name}"}})'
        )
        trigger = {
            "type": "prometheus",
            "metadata": {
                "serverAddress": prometheus_server,
                "query": query,
                "threshold": str(scale_target),
            },
        }
    return {
        "apiVersion": "keda.sh/v1alpha1",
        "kind": "ScaledObject",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "scaleTargetRef": {
                "apiVersion": "apps/v1",
                "kind": "Deployment",
                "name": deployment_name,
            },
            "minReplicaCount": min_replicas,
            "maxReplicaCount": max_replicas,
            "triggers": [trigger],
        },
    }


# -- OTel Collector sidecar CR (reconcilers/otel/otel_reconciler.go:302) ----

def render_otel_collector(
    name: str,
    namespace: str,
    metric_names: Optional[List[str]] = None,
    otlp_endpoint: str = "",
    scrape_interval_s: int = 15,
) -> Dict:
    """OpenTelemetryCollector sidecar CR created per-ISVC when PodMetrics
    autoscaling is used: scrapes the pod's Prometheus endpoint, filters to
    the metrics the autoscaler consumes, ships OTLP (reference
    otel_reconciler.go; wired by raw_kube_reconciler.go:80-101)."""
    filter_cfg = {}
    processors = ["batch"]
    if metric_names:
        filter_cfg = {
            "filter/metrics": {
                "metrics": {
                    "include": {
                        "match_type": "strict",
                        "metric_names": list(metric_names),
                    }
                }
            }
        }
        processors = ["filter/metrics", "batch"]
    return {
        "apiVersion": "opentelemetry.io/v1beta1",
        "kind": "OpenTelemetryCollector",
        "metadata": {"name": name, "namespace": namespace},
        "spec": {
            "mode": "sidecar",
            "config": {
                "receivers": {
                    "prometheus": {
                        "config": {
                            "scrape_configs": [
                                {
                                    "job_name": "kserve-container",
                                    "scrape_interval": f"{scrape_interval_s}s",
                                    "static_configs": [
                                        {"targets": ["localhost:8080"]}
                                    ],
                                }
                            ]
                        }
                    }
                },
                "processors": {"batch": {}, **filter_cfg},
                "exporters": {
                    "otlp": {"endpoint": otlp_endpoint or "keda-otel-scaler:4317"}
                },
                "service": {
                    "pipelines": {
                        "metrics": {
                            "receivers": ["prometheus"],
                            "processors": processors,
                            "exporters": ["otlp"],
                        }
                    }
                },
            },
        },
    }
