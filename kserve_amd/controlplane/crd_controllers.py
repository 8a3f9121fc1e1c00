"""Live controllers for the remaining CRDs: InferenceGraph + TrainedModel.

Reference parity:
- InferenceGraph controller (pkg/controller/v1alpha1/inferencegraph/
  controller.go:133): deploys the graph-router image with the serialized
  ``--graph-json`` spec as a Deployment + Service (+ HPA), propagates
  readiness into the CR status (raw_ig.go:50, PropagateRawStatus :177).
- TrainedModel controller (pkg/controller/v1alpha1/trainedmodel/): each
  TrainedModel CR targeting a multi-model InferenceService upserts its
  entry into the ``modelconfig-<isvc>-<shard>`` ConfigMap the agent
  watcher consumes (pkg/modelconfig/configmap.go:37-161 ConfigsDelta,
  MemoryStrategy shard 0); delete removes the entry; a finalizer
  guarantees removal before the CR goes away.
"""

from __future__ import annotations

import copy
import json
from typing import Dict, Optional, Tuple

from kserve_amd.controlplane.controller import (
    Controller,
    Result,
    create_or_update,
    set_condition,
)
from kserve_amd.controlplane.reconciler import reconcile_graph

IG_GVK = "serving.kserve.io/v1alpha1/InferenceGraph"
TM_GVK = "serving.kserve.io/v1alpha1/TrainedModel"
TM_FINALIZER = "trainedmodel.finalizers"


class InferenceGraphController:
    def __init__(self, server):
        self.server = server

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        namespace, name = key
        obj = self.server.try_get(IG_GVK, namespace, name)
        if obj is None:
            return None  # owned objects GC'd via ownerReferences
        spec = obj.get("spec", {}) or {}
        manifests = reconcile_graph(
            name,
            namespace,
            graph_spec=spec,
            min_replicas=spec.get("minReplicas", 1),
            max_replicas=spec.get("maxReplicas", 1),
        )
        for m in manifests.values():
            m["metadata"].setdefault("labels", {})[
                "serving.kserve.io/inferencegraph"
            ] = name
            create_or_update(self.server, m, owner=obj)
        # status from the router Deployment
        dep = self.server.try_get("apps/v1/Deployment", namespace, name)
        want = (dep or {}).get("spec", {}).get("replicas", 1)
        have = (dep or {}).get("status", {}).get("availableReplicas", 0)
        ready = dep is not None and have >= max(1, want)
        status = copy.deepcopy(obj.get("status", {}) or {})
        set_condition(
            status, "Ready", "True" if ready else "False",
            reason="" if ready else "RouterNotReady",
        )
        if ready:
            status["url"] = f"http://{name}.{namespace}.svc.cluster.local"
        if status != obj.get("status", {}):
            newobj = copy.deepcopy(obj)
            newobj["status"] = status
            self.server.update_status(newobj)
        if not ready:
            return Result(requeue_after=0.05)
        return None

    def build(self) -> Controller:
        c = Controller(
            self.server,
            IG_GVK,
            self.reconcile,
            owned_gvks=("apps/v1/Deployment",),
            owner_label="serving.kserve.io/inferencegraph",
        )
        c.start_watches()
        return c


def _modelconfig_name(isvc: str, shard: int = 0) -> str:
    # MemoryStrategy always places models on shard 0
    # (reference sharding/memory/strategy.go:25-34)
    return f"modelconfig-{isvc}-{shard}"


class TrainedModelController:
    def __init__(self, server):
        self.server = server

    def _upsert_entry(self, namespace: str, isvc: str, entry: Dict) -> None:
        cm_name = _modelconfig_name(isvc)
        cm = self.server.try_get("v1/ConfigMap", namespace, cm_name)
        if cm is None:
            cm = {
                "apiVersion": "v1",
                "kind": "ConfigMap",
                "metadata": {
                    "name": cm_name,
                    "namespace": namespace,
                    "labels": {"serving.kserve.io/inferenceservice": isvc},
                },
                "data": {"models.json": "[]"},
            }
            cm = self.server.create(cm)
        models = json.loads(cm.get("data", {}).get("models.json", "[]"))
        models = [m for m in models if m["modelName"] != entry["modelName"]]
        models.append(entry)
        models.sort(key=lambda m: m["modelName"])
        cm["data"]["models.json"] = json.dumps(models, sort_keys=True)
        self.server.update(cm)

    def _remove_entry(self, namespace: str, isvc: str, model_name: str) -> None:
        cm_name = _modelconfig_name(isvc)
        cm = self.server.try_get("v1/ConfigMap", namespace, cm_name)
        if cm is None:
            return
        models = json.loads(cm.get("data", {}).get("models.json", "[]"))
        models = [m for m in models if m["modelName"] != model_name]
        cm["data"]["models.json"] = json.dumps(models, sort_keys=True)
        self.server.update(cm)

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        namespace, name = key
        obj = self.server.try_get(TM_GVK, namespace, name)
        if obj is None:
            return None
        md = obj["metadata"]
        spec = obj.get("spec", {}) or {}
        isvc = spec.get("inferenceService", "")

        if md.get("deletionTimestamp"):
            if TM_FINALIZER in (md.get("finalizers") or []):
                # delete-external-resources: drop the modelconfig entry so
                # the agent unloads the model (controller.go:738-758 shape)
                if isvc:
                    self._remove_entry(namespace, isvc, name)
                newobj = copy.deepcopy(obj)
                newobj["metadata"]["finalizers"] = [
                    f for f in md["finalizers"] if f != TM_FINALIZER
                ]
                self.server.update(newobj)
            return None

        if TM_FINALIZER not in (md.get("finalizers") or []):
            newobj = copy.deepcopy(obj)
            newobj["metadata"].setdefault("finalizers", []).append(TM_FINALIZER)
            obj = self.server.update(newobj)
            md = obj["metadata"]

        status = copy.deepcopy(obj.get("status", {}) or {})
        model = spec.get("model", {}) or {}
        if not isvc or not model.get("storageUri"):
            set_condition(
                status, "Ready", "False", reason="InvalidSpec",
                message="inferenceService and model.storageUri are required",
            )
        else:
            self._upsert_entry(
                namespace,
                isvc,
                {
                    "modelName": name,
                    "modelSpec": {
                        "storageUri": model.get("storageUri", ""),
                        "framework": model.get("framework", ""),
                        "memory": model.get("memory", ""),
                    },
                },
            )
            set_condition(status, "Ready", "True")
        if status != obj.get("status", {}):
            newobj = copy.deepcopy(obj)
            newobj["status"] = status
            self.server.update_status(newobj)
        return None

    def build(self) -> Controller:
        c = Controller(self.server, TM_GVK, self.reconcile)
        c.start_watches()
        return c
