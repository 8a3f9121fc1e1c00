"""InferenceService reconciler: render Kubernetes manifests (Standard /
RawDeployment mode).

Reference parity: pkg/controller/v1beta1/inferenceservice —
Predictor.Reconcile / buildPodSpec (components/predictor.go:184-422),
RawKubeReconciler (reconcilers/raw/raw_kube_reconciler.go:48-150),
deployment/service reconcilers, HPA mapping
(reconcilers/hpa/hpa_reconciler.go), modelconfig
(pkg/modelconfig/configmap.go:37-161) — as pure functions emitting k8s
manifest dicts, unit-testable without a cluster (the reference's envtest
asserts translate to dict asserts).
"""

from __future__ import annotations

import copy
import json
from typing import Dict, List, Optional

from kserve_amd.controlplane.servingruntime import ServingRuntime, select_runtime
from kserve_amd.controlplane.v1beta1 import (
    InferenceService,
    PredictorSpec,
    default_inference_service,
    validate_inference_service,
)
from kserve_amd.controlplane.webhook import mutate_pod

# internal annotations carried from controller to webhook
# (reference constants.go:158-171)
ANN_STORAGE_URI = "internal.serving.kserve.io/storage-initializer-sourceuri"
ANN_LOGGER = "internal.serving.kserve.io/logger"
ANN_LOGGER_URL = "internal.serving.kserve.io/logger-sink-url"
ANN_LOGGER_MODE = "internal.serving.kserve.io/logger-mode"
ANN_BATCHER = "internal.serving.kserve.io/batcher"
ANN_BATCHER_MAX_SIZE = "internal.serving.kserve.io/batcher-max-batchsize"
ANN_BATCHER_MAX_LATENCY = "internal.serving.kserve.io/batcher-max-latency"
ANN_AGENT = "internal.serving.kserve.io/agent"


def component_annotations(spec) -> Dict[str, str]:
    """reference components/component.go:40-92 (any component spec:
    predictor has storage_uri, transformer/explainer usually do not)."""
    ann: Dict[str, str] = {}
    if getattr(spec, "storage_uri", None):
        ann[ANN_STORAGE_URI] = spec.storage_uri
    if spec.logger is not None:
        ann[ANN_LOGGER] = "true"
        if spec.logger.url:
            ann[ANN_LOGGER_URL] = spec.logger.url
        ann[ANN_LOGGER_MODE] = spec.logger.mode
        ann[ANN_AGENT] = "true"
    if spec.batcher is not None:
        ann[ANN_BATCHER] = "true"
        ann[ANN_BATCHER_MAX_SIZE] = str(spec.batcher.max_batch_size)
        ann[ANN_BATCHER_MAX_LATENCY] = str(spec.batcher.max_latency_ms)
        ann[ANN_AGENT] = "true"
    return ann


def _render_placeholders(container: Dict, isvc: InferenceService) -> Dict:
    """reference isvc utils ReplacePlaceholders ({{.Name}} templating)."""
    raw = json.dumps(container)
    raw = raw.replace("{{.Name}}", isvc.name).replace(
        "{{.Namespace}}", isvc.namespace
    )
    return json.loads(raw)


def predictor_service_name(isvc: InferenceService) -> str:
    return f"{isvc.name}-predictor"


def render_predictor_pod_spec(
    isvc: InferenceService,
    runtimes: List[ServingRuntime],
    storage_init_image: str = "kserve-amd/storage-initializer:latest",
    agent_image: str = "kserve-amd/agent:latest",
) -> Dict:
    """buildPodSpec + pod webhook mutation chain."""
    p = isvc.spec.predictor
    if p.containers:
        container = copy.deepcopy(p.containers[0])
        container.setdefault("name", "kserve-container")
    else:
        rt = select_runtime(
            p.model.model_format.name,
            p.model.protocol_version,
            [],
            runtimes,
            explicit_runtime=p.model.runtime,
            multinode=p.worker is not None,
        )
        container = _render_placeholders(copy.deepcopy(rt.container), isvc)
        if p.model.args:
            container.setdefault("args", []).extend(p.model.args)
        if p.model.resources:
            container["resources"] = p.model.resources
        if p.model.image:
            container["image"] = p.model.image
    pod = {
        "metadata": {
            "labels": {
                "serving.kserve.io/inferenceservice": isvc.name,
                "component": "predictor",
                **isvc.labels,
            },
            "annotations": {
                **isvc.annotations,
                **component_annotations(p),
            },
        },
        "spec": {"containers": [container]},
    }
    return mutate_pod(
        pod, storage_init_image=storage_init_image, agent_image=agent_image
    )


def render_deployment(
    isvc: InferenceService,
    runtimes: List[ServingRuntime],
    canary: bool = False,
    storage_init_image: str = "kserve-amd/storage-initializer:latest",
    agent_image: str = "kserve-amd/agent:latest",
) -> Dict:
    """Deployment manifest (raw mode; reference deployment_reconciler.go)."""
    p = isvc.spec.predictor
    name = predictor_service_name(isvc) + ("-canary" if canary else "")
    pod = render_predictor_pod_spec(
        isvc, runtimes, storage_init_image=storage_init_image,
        agent_image=agent_image,
    )
    pod["metadata"]["labels"]["app"] = name
    return {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {
            "name": name,
            "namespace": isvc.namespace,
            "labels": pod["metadata"]["labels"],
        },
        "spec": {
            "replicas": max(p.min_replicas, 1),
            "selector": {"matchLabels": {"app": name}},
            "template": pod,
        },
    }


def render_service(isvc: InferenceService) -> Dict:
    name = predictor_service_name(isvc)
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {"name": name, "namespace": isvc.namespace},
        "spec": {
            "selector": {"app": name},
            "ports": [
                {"name": "http", "port": 80, "targetPort": 8080},
                {"name": "grpc", "port": 81, "targetPort": 8081},
            ],
        },
    }


def render_hpa(isvc: InferenceService) -> Optional[Dict]:
    """HPA from ScaleMetric (reference hpa_reconciler.go)."""
    p = isvc.spec.predictor
    if not p.max_replicas or p.max_replicas <= p.min_replicas:
        return None
    metric = p.scale_metric or "cpu"
    target = p.scale_target or 80
    if metric in ("cpu", "memory"):
        metrics = [
            {
                "type": "Resource",
                "resource": {
                    "name": metric,
                    "target": {
                        "type": "Utilization",
                        "averageUtilization": target,
                    },
                },
            }
        ]
    else:
        metrics = [
            {
                "type": "Pods",
                "pods": {
                    "metric": {"name": metric},
                    "target": {"type": "AverageValue", "averageValue": str(target)},
                },
            }
        ]
    return {
        "apiVersion": "autoscaling/v2",
        "kind": "HorizontalPodAutoscaler",
        "metadata": {
            "name": predictor_service_name(isvc),
            "namespace": isvc.namespace,
        },
        "spec": {
            "scaleTargetRef": {
                "apiVersion": "apps/v1",
                "kind": "Deployment",
                "name": predictor_service_name(isvc),
            },
            "minReplicas": p.min_replicas,
            "maxReplicas": p.max_replicas,
            "metrics": metrics,
        },
    }


def render_http_route(isvc: InferenceService, ingress_domain: str = "example.com") -> Dict:
    """Gateway API HTTPRoute (reference httproute_reconciler.go)."""
    host = f"{isvc.name}.{isvc.namespace}.{ingress_domain}"
    return {
        "apiVersion": "gateway.networking.k8s.io/v1",
        "kind": "HTTPRoute",
        "metadata": {"name": isvc.name, "namespace": isvc.namespace},
        "spec": {
            "hostnames": [host],
            "rules": [
                {
                    "matches": [{"path": {"type": "PathPrefix", "value": "/"}}],
                    "backendRefs": [
                        {
                            "name": predictor_service_name(isvc),
                            "port": 80,
                        }
                    ],
                }
            ],
        },
    }


def reconcile(
    isvc: InferenceService, runtimes: List[ServingRuntime]
) -> Dict[str, object]:
    """Full reconcile: default -> validate -> manifests (+ canary pair when
    canaryTrafficPercent set; reference reconcileCanaryDeployments
    predictor.go:918)."""
    default_inference_service(isvc)
    validate_inference_service(isvc)
    if isvc.deployment_mode == "Serverless":
        return {
            "knative_service": render_knative_service(isvc, runtimes),
            "httproute": render_http_route(isvc),
        }
    out: Dict[str, object] = {
        "deployment": render_deployment(isvc, runtimes),
        "service": render_service(isvc),
        "httproute": render_http_route(isvc),
    }
    hpa = render_hpa(isvc)
    if hpa:
        out["hpa"] = hpa
    pct = isvc.spec.predictor.canary_traffic_percent
    if pct is not None and 0 < pct < 100:
        out["canary_deployment"] = render_deployment(isvc, runtimes, canary=True)
        out["traffic_split"] = {"stable": 100 - pct, "canary": pct}
    return out


def build_model_config(trained_models: List[Dict]) -> str:
    """modelconfig ConfigMap payload for the agent (reference
    pkg/modelconfig/configmap.go:37-161; MemoryStrategy shard 0)."""
    entries = []
    for tm in trained_models:
        entries.append(
            {
                "modelName": tm["name"],
                "modelSpec": {
                    "storageUri": tm["storageUri"],
                    "framework": tm.get("framework", ""),
                    "memory": tm.get("memory", ""),
                },
            }
        )
    return json.dumps(entries, sort_keys=True)


def render_knative_service(isvc: InferenceService, runtimes: List[ServingRuntime]) -> Dict:
    """Knative Service manifest (Serverless mode; reference
    ksvc_reconciler.go:159 with canaryTrafficPercent traffic split)."""
    p = isvc.spec.predictor
    pod = render_predictor_pod_spec(isvc, runtimes)
    ann = {
        "autoscaling.knative.dev/min-scale": str(p.min_replicas),
    }
    if p.max_replicas:
        ann["autoscaling.knative.dev/max-scale"] = str(p.max_replicas)
    if p.scale_target:
        ann["autoscaling.knative.dev/target"] = str(p.scale_target)
    traffic = [{"latestRevision": True, "percent": 100}]
    pct = p.canary_traffic_percent
    if pct is not None and 0 <= pct < 100:
        traffic = [
            {"latestRevision": True, "percent": pct},
            {
                "latestRevision": False,
                "revisionName": f"{predictor_service_name(isvc)}-prev",
                "percent": 100 - pct,
            },
        ]
    return {
        "apiVersion": "serving.knative.dev/v1",
        "kind": "Service",
        "metadata": {
            "name": predictor_service_name(isvc),
            "namespace": isvc.namespace,
        },
        "spec": {
            "template": {
                "metadata": {
                    "labels": pod["metadata"]["labels"],
                    "annotations": {**pod["metadata"]["annotations"], **ann},
                },
                "spec": pod["spec"],
            },
            "traffic": traffic,
        },
    }


# ---- InferenceGraph controller (reference v1alpha1/inferencegraph) ---------

GRAPH_ROUTER_IMAGE = "kserve-amd/graph-router:latest"


def render_graph_deployment(
    name: str,
    namespace: str,
    graph_spec: Dict,
    router_image: str = GRAPH_ROUTER_IMAGE,
    min_replicas: int = 1,
    max_replicas: int = 1,
) -> Dict:
    """Deployment running the graph router with --graph-json <spec>
    (reference raw_ig.go:50-104 createInferenceGraphPodSpec)."""
    import json as _json

    labels = {"serving.kserve.io/inferencegraph": name}
    return {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {"name": name, "namespace": namespace, "labels": labels},
        "spec": {
            "replicas": min_replicas,
            "selector": {"matchLabels": labels},
            "template": {
                "metadata": {"labels": labels},
                "spec": {
                    "containers": [
                        {
                            "name": "kserve-router",
                            "image": router_image,
                            "args": [
                                "--graph-json",
                                _json.dumps(graph_spec, separators=(",", ":")),
                            ],
                            "ports": [{"containerPort": 8080}],
                            "readinessProbe": {
                                "httpGet": {"path": "/readyz", "port": 8080}
                            },
                        }
                    ],
                    "automountServiceAccountToken": False,
                },
            },
        },
    }


def render_graph_service(name: str, namespace: str) -> Dict:
    labels = {"serving.kserve.io/inferencegraph": name}
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {"name": name, "namespace": namespace, "labels": labels},
        "spec": {
            "selector": labels,
            "ports": [{"name": "http", "port": 80, "targetPort": 8080}],
        },
    }


def reconcile_graph(
    name: str,
    namespace: str,
    graph_spec: Dict,
    min_replicas: int = 1,
    max_replicas: int = 1,
) -> Dict[str, Dict]:
    """Desired state for an InferenceGraph CR (raw-deployment mode):
    Deployment + Service (+ HPA when max_replicas > min_replicas)."""
    out = {
        "deployment": render_graph_deployment(
            name, namespace, graph_spec, min_replicas=min_replicas,
            max_replicas=max_replicas,
        ),
        "service": render_graph_service(name, namespace),
    }
    if max_replicas > min_replicas:
        out["hpa"] = {
            "apiVersion": "autoscaling/v2",
            "kind": "HorizontalPodAutoscaler",
            "metadata": {"name": name, "namespace": namespace},
            "spec": {
                "scaleTargetRef": {
                    "apiVersion": "apps/v1",
                    "kind": "Deployment",
                    "name": name,
                },
                "minReplicas": min_replicas,
                "maxReplicas": max_replicas,
                "metrics": [
                    {
                        "type": "Resource",
                        "resource": {
                            "name": "cpu",
                            "target": {
                                "type": "Utilization",
                                "averageUtilization": 80,
                            },
                        },
                    }
                ],
            },
        }
    return out


# ---- Transformer / Explainer component deployments -------------------------
# (reference components/transformer.go, explainer.go; --predictor_host arg
# injection per v1beta1/transformer_custom.go:99-105, explainer_custom.go:80-84)

def component_service_name(isvc: InferenceService, component: str) -> str:
    return f"{isvc.name}-{component}"


def render_component_deployment(
    isvc: InferenceService,
    component: str,  # "transformer" | "explainer"
    storage_init_image: str = "kserve-amd/storage-initializer:latest",
    agent_image: str = "kserve-amd/agent:latest",
) -> Dict:
    """Deployment for a transformer or explainer component. The container
    comes from the component spec; --model_name and --predictor_host are
    injected so the data plane forwards predict/explain calls to the
    predictor service (reference TransformerSpec.GetContainer)."""
    spec = getattr(isvc.spec, component)
    if spec is None:
        raise ValueError(f"isvc has no {component} spec")
    containers = getattr(spec, "containers", None) or []
    if containers:
        container = copy.deepcopy(containers[0])
    elif component == "explainer" and getattr(spec, "art", None) is not None:
        container = {
            "image": "kserve-amd/artexplainer:latest",
            "command": ["python", "-m", "kserve_amd.runtimes.aifserver"],
        }
    else:
        raise ValueError(f"{component} must specify a container")
    container.setdefault("name", "kserve-container")
    args = container.setdefault("args", [])
    predictor_host = (
        f"{predictor_service_name(isvc)}.{isvc.namespace}"
    )
    if not any(a.startswith("--predictor_host") for a in args):
        args.extend(["--predictor_host", predictor_host])
    if not any(a.startswith("--model_name") for a in args):
        args.extend(["--model_name", isvc.name])
    name = component_service_name(isvc, component)
    pod = {
        "metadata": {
            "labels": {
                "serving.kserve.io/inferenceservice": isvc.name,
                "component": component,
                "app": name,
                **isvc.labels,
            },
            "annotations": {
                **isvc.annotations,
                **component_annotations(spec),
            },
        },
        "spec": {"containers": [container]},
    }
    pod = mutate_pod(
        pod, storage_init_image=storage_init_image, agent_image=agent_image
    )
    return {
        "apiVersion": "apps/v1",
        "kind": "Deployment",
        "metadata": {
            "name": name,
            "namespace": isvc.namespace,
            "labels": pod["metadata"]["labels"],
        },
        "spec": {
            "replicas": max(getattr(spec, "min_replicas", 1) or 1, 1),
            "selector": {"matchLabels": {"app": name}},
            "template": pod,
        },
    }


def render_component_service(isvc: InferenceService, component: str) -> Dict:
    name = component_service_name(isvc, component)
    return {
        "apiVersion": "v1",
        "kind": "Service",
        "metadata": {"name": name, "namespace": isvc.namespace},
        "spec": {
            "selector": {"app": name},
            "ports": [
                {"name": "http", "port": 80, "targetPort": 8080},
                {"name": "grpc", "port": 81, "targetPort": 8081},
            ],
        },
    }


def desired_state(
    isvc: InferenceService,
    runtimes: List[ServingRuntime],
    config=None,
) -> List[Dict]:
    """Complete desired manifest list for the LIVE controller
    (isvc_controller.py): predictor (+canary) + transformer + explainer
    deployments/services, autoscaler (HPA or KEDA per AutoscalerConfig /
    serving.kserve.io/autoscalerClass annotation), and the ingress backend
    chosen by IngressConfig (Istio VS / HTTPRoute / k8s Ingress).

    reference: the per-component fan-out of controller.go:281-305 plus
    factory.go:CreateIngressReconciler."""
    from kserve_amd.controlplane.configmap import InferenceServicesConfig
    from kserve_amd.controlplane.ingress import (
        render_keda_scaled_object,
        select_ingress,
    )

    cfg = config or InferenceServicesConfig()
    default_inference_service(isvc)
    validate_inference_service(isvc)
    p = isvc.spec.predictor
    manifests: List[Dict] = []

    mode = isvc.annotations.get(
        "serving.kserve.io/deploymentMode", cfg.deploy.default_deployment_mode
    )
    has_transformer = isvc.spec.transformer is not None
    has_explainer = isvc.spec.explainer is not None
    traffic_split = None

    if mode == "Serverless":
        manifests.append(render_knative_service(isvc, runtimes))
    else:
        manifests.append(
            render_deployment(
                isvc, runtimes,
                storage_init_image=cfg.storage_initializer.image,
                agent_image=cfg.agent.image,
            )
        )
        manifests.append(render_service(isvc))
        pct = p.canary_traffic_percent
        if pct is not None and 0 < pct < 100:
            manifests.append(
                render_deployment(
                    isvc, runtimes, canary=True,
                    storage_init_image=cfg.storage_initializer.image,
                    agent_image=cfg.agent.image,
                )
            )
            canary_svc = render_service(isvc)
            canary_svc["metadata"]["name"] += "-canary"
            canary_svc["spec"]["selector"] = {
                "app": predictor_service_name(isvc) + "-canary"
            }
            manifests.append(canary_svc)
            traffic_split = {"stable": 100 - pct, "canary": pct}
        # autoscaler: keda class -> ScaledObject, else HPA; PodMetrics
        # scale metrics additionally deploy the OTel collector sidecar CR
        # feeding the scaler (reference otel_reconciler.go)
        autoscaler_class = isvc.annotations.get(
            "serving.kserve.io/autoscalerClass", cfg.autoscaler.autoscaler_class
        )
        if p.scale_metric and p.scale_metric not in (
            "cpu", "memory", "concurrency", "rps"
        ):
            from kserve_amd.controlplane.ingress import render_otel_collector

            manifests.append(
                render_otel_collector(
                    predictor_service_name(isvc),
                    isvc.namespace,
                    metric_names=[p.scale_metric],
                )
            )
        if p.max_replicas and p.max_replicas > p.min_replicas:
            if autoscaler_class == "keda":
                manifests.append(
                    render_keda_scaled_object(
                        predictor_service_name(isvc),
                        isvc.namespace,
                        predictor_service_name(isvc),
                        p.min_replicas,
                        p.max_replicas,
                        p.scale_metric or "cpu",
                        p.scale_target or 80,
                    )
                )
            else:
                hpa = render_hpa(isvc)
                if hpa:
                    manifests.append(hpa)

    for component in ("transformer", "explainer"):
        if getattr(isvc.spec, component) is None:
            continue
        manifests.append(
            render_component_deployment(
                isvc,
                component,
                storage_init_image=cfg.storage_initializer.image,
                agent_image=cfg.agent.image,
            )
        )
        manifests.append(render_component_service(isvc, component))

    ingress = select_ingress(
        isvc.name,
        isvc.namespace,
        cfg.ingress,
        mode,
        has_transformer=has_transformer,
        has_explainer=has_explainer,
        traffic_split=traffic_split,
    )
    if ingress is not None:
        manifests.append(ingress)
    return manifests
