"""LocalModel cache cluster orchestration: CacheCR → per-node CRs → PV/PVC +
download Jobs.

Reference parity: pkg/controller/v1alpha1/localmodel — LocalModelReconciler
(localmodelcache_reconciler.go:59) fans a cluster-scoped LocalModelCache CR
out to per-node LocalModelNode CRs for every node matching the NodeGroup
selector, and creates a PersistentVolume + PersistentVolumeClaim pair per
(cache, node-group) from the group's storage template
(reconcilers/utils.go, 774 LoC). pkg/controller/v1alpha1/localmodelnode —
LocalModelNodeReconciler (controller.go:527) runs on each node, launching a
download Job with the storage-initializer image per missing model
(launchJob :117-121) and reporting per-model status; reconcile every minute
(:85).

The in-process LocalModelNodeAgent (localmodel.py) remains the
filesystem-convergence engine; these controllers are the cluster-side
orchestration against the APIServer interface.
"""

from __future__ import annotations

import copy
from typing import Dict, List, Optional, Tuple

from kserve_amd.controlplane.controller import (
    Controller,
    Result,
    create_or_update,
    delete_if_exists,
)

CACHE_GVK = "serving.kserve.io/v1alpha1/LocalModelCache"
NODEGROUP_GVK = "serving.kserve.io/v1alpha1/LocalModelNodeGroup"
NODE_CR_GVK = "serving.kserve.io/v1alpha1/LocalModelNode"
NODE_GVK = "v1/Node"
PV_GVK = "v1/PersistentVolume"
PVC_GVK = "v1/PersistentVolumeClaim"
JOB_GVK = "batch/v1/Job"


class LocalModelCacheController:
    """Cluster side: LocalModelCache + NodeGroups + Nodes → per-node
    LocalModelNode CRs and PV/PVC pairs; aggregates node statuses back."""

    def __init__(self, server, jobs_namespace: str = "kserve-localmodel-jobs"):
        self.server = server
        self.jobs_namespace = jobs_namespace

    def _nodes_for_group(self, group: Dict) -> List[Dict]:
        selector = (
            group.get("spec", {}).get("nodeSelector", {}) or {}
        )
        return self.server.list(NODE_GVK, namespace="", label_selector=selector or None)

    def _pv_name(self, cache_name: str, group_name: str) -> str:
        return f"{cache_name}-{group_name}-pv"

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        _, name = key
        cache = self.server.try_get(CACHE_GVK, "", name)
        if cache is None:
            # cascade: remove node-CR entries for this cache
            for node_cr in self.server.list(NODE_CR_GVK, ""):
                models = [
                    m
                    for m in node_cr.get("spec", {}).get("localModels", [])
                    if m.get("modelName") != name
                ]
                if len(models) != len(node_cr["spec"].get("localModels", [])):
                    node_cr["spec"]["localModels"] = models
                    self.server.update(node_cr)
            return None
        spec = cache.get("spec", {}) or {}
        uri = spec.get("sourceModelUri", "")
        groups = spec.get("nodeGroups", []) or ["default"]
        node_names: List[str] = []
        status = copy.deepcopy(cache.get("status", {}) or {})
        node_status = status.setdefault("nodeStatus", {})

        for group_name in groups:
            group = self.server.try_get(NODEGROUP_GVK, "", group_name) or {
                "spec": {
                    "persistentVolumeSpec": {
                        "hostPath": {"path": f"/models/{group_name}"},
                        "capacity": {"storage": spec.get("modelSize", "10Gi")},
                    }
                }
            }
            # PV + PVC pair per (cache, group) — reconcilers/utils.go
            pv_spec = copy.deepcopy(
                group.get("spec", {}).get("persistentVolumeSpec", {}) or {}
            )
            capacity = pv_spec.pop("capacity", None) or {
                "storage": spec.get("modelSize", "10Gi")
            }
            pv_name = self._pv_name(name, group_name)
            create_or_update(
                self.server,
                {
                    "apiVersion": "v1",
                    "kind": "PersistentVolume",
                    "metadata": {
                        "name": pv_name,
                        "labels": {"serving.kserve.io/localmodel": name},
                    },
                    "spec": {
                        **pv_spec,
                        "capacity": capacity,
                        "accessModes": ["ReadOnlyMany"],
                        "persistentVolumeReclaimPolicy": "Delete",
                    },
                },
                owner=cache,
            )
            create_or_update(
                self.server,
                {
                    "apiVersion": "v1",
                    "kind": "PersistentVolumeClaim",
                    "metadata": {
                        "name": pv_name,
                        "namespace": self.jobs_namespace,
                        "labels": {"serving.kserve.io/localmodel": name},
                    },
                    "spec": {
                        "volumeName": pv_name,
                        "accessModes": ["ReadOnlyMany"],
                        "resources": {"requests": capacity},
                    },
                },
                owner=cache,
            )
            # fan out to per-node CRs
            for node in self._nodes_for_group(group):
                node_name = node["metadata"]["name"]
                node_names.append(node_name)
                cr = self.server.try_get(NODE_CR_GVK, "", node_name)
                entry = {"modelName": name, "sourceModelUri": uri}
                if cr is None:
                    self.server.create(
                        {
                            "apiVersion": "serving.kserve.io/v1alpha1",
                            "kind": "LocalModelNode",
                            "metadata": {"name": node_name},
                            "spec": {"localModels": [entry]},
                        }
                    )
                else:
                    models = cr["spec"].setdefault("localModels", [])
                    if not any(m.get("modelName") == name for m in models):
                        models.append(entry)
                        self.server.update(cr)
                # pull node-side status up into the cache CR
                cr = self.server.try_get(NODE_CR_GVK, "", node_name)
                st = (cr or {}).get("status", {}).get("modelStatus", {})
                node_status[node_name] = st.get(name, "NodeNotReady")

        ready = sum(1 for v in node_status.values() if v == "NodeDownloaded")
        status["copies"] = {"total": len(node_names), "available": ready}
        newobj = copy.deepcopy(cache)
        newobj["status"] = status
        if cache.get("status") != status:
            self.server.update_status(newobj)
        if ready < len(node_names):
            return Result(requeue_after=0.05)
        return None

    def build(self) -> Controller:
        c = Controller(
            self.server,
            CACHE_GVK,
            self.reconcile,
            owned_gvks=(NODE_CR_GVK,),
        )
        # any LocalModelNode change re-triggers every cache it names
        orig = c._enqueue_from_event

        def enqueue(ev, primary):
            if primary:
                orig(ev, True)
                return
            for m in ev.object.get("spec", {}).get("localModels", []) or []:
                c.queue.add(("", m.get("modelName", "")))

        c._enqueue_from_event = enqueue
        c.start_watches()
        return c


class LocalModelNodeController:
    """Node side: converge the node's LocalModelNode CR by launching download
    Jobs (storage-initializer image) and reporting per-model status
    (localmodelnode/controller.go:527, launchJob :117)."""

    def __init__(
        self,
        server,
        node_name: str,
        jobs_namespace: str = "kserve-localmodel-jobs",
        job_image: str = "kserve-amd/storage-initializer:latest",
    ):
        self.server = server
        self.node_name = node_name
        self.jobs_namespace = jobs_namespace
        self.job_image = job_image

    def _job_name(self, model: str) -> str:
        return f"{model}-{self.node_name}-download"

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        _, name = key
        if name != self.node_name:
            return None
        cr = self.server.try_get(NODE_CR_GVK, "", name)
        if cr is None:
            return None
        declared = cr.get("spec", {}).get("localModels", []) or []
        status = copy.deepcopy(cr.get("status", {}) or {})
        model_status = status.setdefault("modelStatus", {})
        pending = False
        for m in declared:
            model = m.get("modelName", "")
            job = self.server.try_get(
                JOB_GVK, self.jobs_namespace, self._job_name(model)
            )
            if job is None:
                self.server.create(self._render_job(m))
                model_status[model] = "NodeDownloadPending"
                pending = True
            elif job.get("status", {}).get("succeeded"):
                model_status[model] = "NodeDownloaded"
            elif job.get("status", {}).get("failed"):
                model_status[model] = "NodeDownloadError"
            else:
                model_status[model] = "NodeDownloadPending"
                pending = True
        # GC: models removed from the CR → drop their Jobs + status
        names = {m.get("modelName") for m in declared}
        for model in list(model_status):
            if model not in names:
                delete_if_exists(
                    self.server, JOB_GVK, self.jobs_namespace,
                    self._job_name(model),
                )
                model_status.pop(model)
        if cr.get("status") != status:
            newobj = copy.deepcopy(cr)
            newobj["status"] = status
            self.server.update_status(newobj)
        if pending:
            return Result(requeue_after=0.05)
        return None

    def _render_job(self, model: Dict) -> Dict:
        model_name = model.get("modelName", "")
        return {
            "apiVersion": "batch/v1",
            "kind": "Job",
            "metadata": {
                "name": self._job_name(model_name),
                "namespace": self.jobs_namespace,
                "labels": {"serving.kserve.io/localmodel": model_name},
            },
            "spec": {
                "ttlSecondsAfterFinished": 3600,
                "template": {
                    "spec": {
                        "nodeName": self.node_name,
                        "restartPolicy": "Never",
                        "containers": [
                            {
                                "name": "storage-initializer",
                                "image": self.job_image,
                                "args": [
                                    model.get("sourceModelUri", ""),
                                    f"/mnt/models/{model_name}",
                                ],
                                "volumeMounts": [
                                    {
                                        "name": "model-cache",
                                        "mountPath": "/mnt/models",
                                    }
                                ],
                            }
                        ],
                        "volumes": [
                            {
                                "name": "model-cache",
                                "hostPath": {"path": "/models"},
                            }
                        ],
                    }
                },
            },
        }

    def build(self) -> Controller:
        c = Controller(
            self.server,
            NODE_CR_GVK,
            self.reconcile,
        )
        # Job completion events re-trigger this node's CR
        jw = self.server.watch(JOB_GVK)

        def pump_jobs():
            while True:
                ev = jw.next(timeout=0.0)
                if ev is None:
                    break
                c.queue.add(("", self.node_name))

        orig_pump = c.pump_events

        def pump(budget=0.0):
            n = orig_pump(budget)
            pump_jobs()
            return n

        c.pump_events = pump
        c.start_watches()
        return c


class FakeJobController:
    """Test stand-in for the batch Job controller: marks every Job
    succeeded (the envtest role — no kubelet runs the pod)."""

    def __init__(self, server, fail_names: Optional[set] = None):
        self.server = server
        self.fail_names = fail_names or set()

    def reconcile(self, key: Tuple[str, str]) -> Optional[Result]:
        ns, name = key
        job = self.server.try_get(JOB_GVK, ns, name)
        if job is None or job.get("status"):
            return None
        job["status"] = (
            {"failed": 1} if name in self.fail_names else {"succeeded": 1}
        )
        self.server.update_status(job)
        return None

    def build(self) -> Controller:
        c = Controller(self.server, JOB_GVK, self.reconcile)
        c.start_watches()
        return c
