#!/usr/bin/env python3
"""Generate the deployable YAML catalog (reference config/crd + config/
runtimes): CRD definitions for every CRD the controllers serve, and
ClusterServingRuntime manifests for the native runtime images."""
import os
import sys

import yaml

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from kserve_amd.controlplane.servingruntime import default_cluster_runtimes

OUT = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "config")


def crd(group, version, kind, plural, scope="Namespaced", extra_versions=()):
    versions = [{
        "name": version,
        "served": True,
        "storage": True,
        "schema": {
            "openAPIV3Schema": {
                "type": "object",
                "x-kubernetes-preserve-unknown-fields": True,
            }
        },
        "subresources": {"status": {}},
    }]
    for v in extra_versions:
        versions.append({
            "name": v,
            "served": True,
            "storage": False,
            "schema": {
                "openAPIV3Schema": {
                    "type": "object",
                    "x-kubernetes-preserve-unknown-fields": True,
                }
            },
            "subresources": {"status": {}},
        })
    return {
        "apiVersion": "apiextensions.k8s.io/v1",
        "kind": "CustomResourceDefinition",
        "metadata": {"name": f"{plural}.{group}"},
        "spec": {
            "group": group,
            "names": {
                "kind": kind,
                "plural": plural,
                "singular": kind.lower(),
            },
            "scope": scope,
            "versions": versions,
        },
    }


CRDS = [
    crd("serving.kserve.io", "v1beta1", "InferenceService",
        "inferenceservices"),
    crd("serving.kserve.io", "v1alpha1", "InferenceGraph",
        "inferencegraphs"),
    crd("serving.kserve.io", "v1alpha1", "ServingRuntime",
        "servingruntimes"),
    crd("serving.kserve.io", "v1alpha1", "ClusterServingRuntime",
        "clusterservingruntimes", scope="Cluster"),
    crd("serving.kserve.io", "v1alpha1", "TrainedModel", "trainedmodels"),
    crd("serving.kserve.io", "v1alpha1", "LocalModelCache",
        "localmodelcaches", scope="Cluster"),
    crd("serving.kserve.io", "v1alpha1", "LocalModelNode",
        "localmodelnodes", scope="Cluster"),
    crd("serving.kserve.io", "v1alpha1", "LocalModelNodeGroup",
        "localmodelnodegroups", scope="Cluster"),
    crd("serving.kserve.io", "v1alpha1", "ClusterStorageContainer",
        "clusterstoragecontainers", scope="Cluster"),
    crd("serving.kserve.io", "v1alpha2", "LLMInferenceService",
        "llminferenceservices"),
    crd("serving.kserve.io", "v1alpha2", "LLMInferenceServiceConfig",
        "llminferenceserviceconfigs"),
]


def main():
    crd_dir = os.path.join(OUT, "crds")
    os.makedirs(crd_dir, exist_ok=True)
    for c in CRDS:
        path = os.path.join(crd_dir, c["metadata"]["name"] + ".yaml")
        with open(path, "w") as f:
            yaml.safe_dump(c, f, sort_keys=False)
    rt_dir = os.path.join(OUT, "runtimes")
    os.makedirs(rt_dir, exist_ok=True)
    for rt in default_cluster_runtimes():
        manifest = {
            "apiVersion": "serving.kserve.io/v1alpha1",
            "kind": "ClusterServingRuntime",
            "metadata": {"name": rt.name},
            "spec": {
                "supportedModelFormats": [
                    {"name": f.name, "autoSelect": f.auto_select,
                     "priority": f.priority}
                    for f in rt.supported_model_formats
                ],
                "protocolVersions": rt.protocol_versions,
                "containers": [rt.container],
                **({"workerSpec": {"pipelineParallelSize": 1,
                                   "tensorParallelSize": 1}}
                   if rt.workers else {}),
            },
        }
        with open(os.path.join(rt_dir, rt.name + ".yaml"), "w") as f:
            yaml.safe_dump(manifest, f, sort_keys=False)
    print(f"wrote {len(CRDS)} CRDs + {len(default_cluster_runtimes())} runtimes")


if __name__ == "__main__":
    main()
