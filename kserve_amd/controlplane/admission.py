"""Admission webhook server: the AdmissionReview HTTP endpoints.

Reference parity: cmd/manager/main.go:238-282 registers the pod mutator at
/mutate-pods and per-CRD validators/defaulters; pkg/webhook/admission/pod/
mutator.go:47-152 handles AdmissionReview and responds with a JSONPatch.

This module wraps the pure functions (webhook.mutate_pod,
v1beta1 defaulter/validator, llmisvc_config.validate_router) in the
admission.k8s.io/v1 envelope: requests carry the object, responses carry
``allowed`` plus an RFC-6902 JSONPatch for mutations. Serve with uvicorn
behind the TLS secret the manager's manifest mounts."""

from __future__ import annotations

import base64
import copy
import json
from typing import Any, Dict, List, Optional

from fastapi import FastAPI, Request

from kserve_amd.controlplane.v1beta1 import ValidationError
from kserve_amd.controlplane.webhook import mutate_pod


def json_patch(before: Any, after: Any, path: str = "") -> List[Dict]:
    """Minimal RFC-6902 diff (add/replace/remove) sufficient for pod
    mutations (containers/volumes appends, annotation adds)."""
    if type(before) is not type(after):
        return [{"op": "replace", "path": path or "/", "value": after}]
    if isinstance(before, dict):
        ops: List[Dict] = []
        for k in before:
            esc = k.replace("~", "~0").replace("/", "~1")
            if k not in after:
                ops.append({"op": "remove", "path": f"{path}/{esc}"})
            elif before[k] != after[k]:
                ops.extend(json_patch(before[k], after[k], f"{path}/{esc}"))
        for k in after:
            if k not in before:
                esc = k.replace("~", "~0").replace("/", "~1")
                ops.append(
                    {"op": "add", "path": f"{path}/{esc}", "value": after[k]}
                )
        return ops
    if isinstance(before, list):
        if before == after:
            return []
        ops = []
        common = min(len(before), len(after))
        for i in range(common):
            if before[i] != after[i]:
                ops.extend(json_patch(before[i], after[i], f"{path}/{i}"))
        for i in range(len(before) - 1, common - 1, -1):
            ops.append({"op": "remove", "path": f"{path}/{i}"})
        for i in range(common, len(after)):
            ops.append({"op": "add", "path": f"{path}/-", "value": after[i]})
        return ops
    if before != after:
        return [{"op": "replace", "path": path, "value": after}]
    return []


def _review_response(req: Dict, allowed: bool, patch: Optional[List] = None,
                     message: str = "") -> Dict:
    resp: Dict = {
        "uid": req.get("uid", ""),
        "allowed": allowed,
    }
    if message:
        resp["status"] = {"message": message}
    if patch:
        resp["patchType"] = "JSONPatch"
        resp["patch"] = base64.b64encode(
            json.dumps(patch).encode()
        ).decode()
    return {
        "apiVersion": "admission.k8s.io/v1",
        "kind": "AdmissionReview",
        "response": resp,
    }


def create_admission_app(
    storage_init_image: str = "kserve-amd/storage-initializer:latest",
    agent_image: str = "kserve-amd/agent:latest",
) -> FastAPI:
    app = FastAPI()

    @app.post("/mutate-pods")
    async def mutate_pods(request: Request):
        review = await request.json()
        req = review.get("request", {}) or {}
        pod = req.get("object", {}) or {}
        mutated = mutate_pod(
            copy.deepcopy(pod),
            storage_init_image=storage_init_image,
            agent_image=agent_image,
        )
        return _review_response(req, True, json_patch(pod, mutated))

    @app.post("/validate-inferenceservices")
    async def validate_isvc(request: Request):
        review = await request.json()
        req = review.get("request", {}) or {}
        obj = req.get("object", {}) or {}
        from kserve_amd.controlplane.isvc_controller import isvc_from_manifest
        from kserve_amd.controlplane.v1beta1 import (
            default_inference_service,
            validate_inference_service,
        )

        try:
            isvc = isvc_from_manifest(obj)
            default_inference_service(isvc)
            validate_inference_service(isvc)
        except (ValidationError, KeyError, TypeError) as e:
            return _review_response(req, False, message=str(e))
        return _review_response(req, True)

    @app.post("/validate-llminferenceservices")
    async def validate_llm(request: Request):
        review = await request.json()
        req = review.get("request", {}) or {}
        obj = req.get("object", {}) or {}
        from kserve_amd.controlplane.llmisvc_config import validate_router

        violations = validate_router(
            (obj.get("spec", {}) or {}).get("router")
        )
        if violations:
            return _review_response(req, False, message="; ".join(violations))
        return _review_response(req, True)

    @app.post("/validate-servingruntimes")
    async def validate_sr(request: Request):
        """Reject duplicate model-format priorities within one runtime
        (reference servingruntime validator)."""
        review = await request.json()
        req = review.get("request", {}) or {}
        obj = req.get("object", {}) or {}
        seen: Dict[str, int] = {}
        for f in (obj.get("spec", {}) or {}).get(
            "supportedModelFormats", []
        ) or []:
            name = f.get("name", "")
            pri = f.get("priority", 1)
            if name in seen and seen[name] == pri and f.get("autoSelect"):
                return _review_response(
                    req, False,
                    message=(
                        f"duplicate priority {pri} for model format "
                        f"{name!r}"
                    ),
                )
            seen[name] = pri
        return _review_response(req, True)

    @app.get("/healthz")
    async def healthz():
        return {"status": "ok"}

    return app


def main(argv=None):
    import argparse

    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=9443)
    ap.add_argument("--tls-cert", default="")
    ap.add_argument("--tls-key", default="")
    args = ap.parse_args(argv)
    kwargs = {}
    if args.tls_cert:
        kwargs = {"ssl_certfile": args.tls_cert, "ssl_keyfile": args.tls_key}
    uvicorn.run(create_admission_app(), host="0.0.0.0", port=args.port,
                **kwargs)


if __name__ == "__main__":
    main()
