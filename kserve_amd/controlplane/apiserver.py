"""Pluggable Kubernetes API-server interface + in-memory implementation.

The reference reconciles against a real API server through controller-runtime
(watches, create-or-update with semantic diff, finalizers, status conditions,
requeue-until-ready — pkg/controller/v1beta1/inferenceservice/controller.go:122-455,
reconcilers/deployment/deployment_reconciler.go). Its envtest suites run the
same controllers against kubebuilder's real-etcd-no-kubelet control plane
(pkg/testing/).

MI355X-native equivalent: the controllers in this package talk to an
``APIServer`` interface. ``FakeAPIServer`` is the in-process implementation —
an object store with resourceVersion/generation bookkeeping, finalizer-aware
deletion, ownerReference garbage collection, and watch streams — playing the
envtest role for our controller tests while ``KubectlAPIServer`` (thin shell
adapter, same interface) serves real clusters.
"""

from __future__ import annotations

import copy
import itertools
import queue
import threading
import uuid
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple


class APIError(Exception):
    def __init__(self, code: int, message: str):
        super().__init__(message)
        self.code = code


class NotFound(APIError):
    def __init__(self, message: str = "not found"):
        super().__init__(404, message)


class Conflict(APIError):
    def __init__(self, message: str = "conflict"):
        super().__init__(409, message)


class AlreadyExists(APIError):
    def __init__(self, message: str = "already exists"):
        super().__init__(409, message)


GVK = str  # "apps/v1/Deployment" — apiVersion + "/" + kind
Key = Tuple[GVK, str, str]  # (gvk, namespace, name)


def gvk_of(obj: Dict) -> GVK:
    return f"{obj['apiVersion']}/{obj['kind']}"


def key_of(obj: Dict) -> Key:
    md = obj.get("metadata", {})
    return (gvk_of(obj), md.get("namespace", ""), md["name"])


@dataclass
class WatchEvent:
    type: str  # ADDED | MODIFIED | DELETED
    object: Dict


class Watch:
    """A live subscription: iterate or poll ``events``; ``stop()`` to end."""

    def __init__(self, server: "FakeAPIServer", gvk: GVK):
        self.events: "queue.Queue[WatchEvent]" = queue.Queue()
        self._server = server
        self._gvk = gvk
        self._stopped = False

    def stop(self) -> None:
        self._stopped = True
        self._server._unsubscribe(self)

    def next(self, timeout: Optional[float] = None) -> Optional[WatchEvent]:
        try:
            return self.events.get(timeout=timeout)
        except queue.Empty:
            return None


class FakeAPIServer:
    """In-memory API server with the semantics the controllers rely on:

    - resourceVersion bumped on every write; update() with a stale
      resourceVersion raises Conflict (optimistic concurrency)
    - metadata.generation bumped on spec changes only (status writes via
      ``update_status`` leave it untouched, like the /status subresource)
    - delete() with finalizers present sets deletionTimestamp and keeps the
      object; removal of the last finalizer completes the delete
    - ownerReferences: deleting an owner cascades to owned objects
      (foreground-style, synchronous — fine for tests)
    - watch(gvk) streams ADDED/MODIFIED/DELETED events
    """

    def __init__(self):
        self._lock = threading.RLock()
        self._objects: Dict[Key, Dict] = {}
        self._rv = itertools.count(1)
        self._watches: Dict[GVK, List[Watch]] = {}

    # -- helpers -----------------------------------------------------------
    def _emit(self, event_type: str, obj: Dict) -> None:
        for w in self._watches.get(gvk_of(obj), []):
            w.events.put(WatchEvent(event_type, copy.deepcopy(obj)))

    def _unsubscribe(self, w: Watch) -> None:
        with self._lock:
            lst = self._watches.get(w._gvk, [])
            if w in lst:
                lst.remove(w)

    @staticmethod
    def _spec_view(obj: Dict) -> Dict:
        """Everything except status + server-managed metadata (what
        generation tracks)."""
        o = {k: v for k, v in obj.items() if k != "status"}
        md = dict(o.get("metadata", {}))
        for k in ("resourceVersion", "generation", "uid", "creationTimestamp",
                  "managedFields"):
            md.pop(k, None)
        o["metadata"] = md
        return o

    # -- CRUD --------------------------------------------------------------
    def create(self, obj: Dict) -> Dict:
        obj = copy.deepcopy(obj)
        with self._lock:
            k = key_of(obj)
            if k in self._objects:
                raise AlreadyExists(f"{k} already exists")
            md = obj.setdefault("metadata", {})
            md["uid"] = str(uuid.uuid4())
            md["resourceVersion"] = str(next(self._rv))
            md["generation"] = 1
            md.setdefault("creationTimestamp", f"t{md['resourceVersion']}")
            self._objects[k] = obj
            self._emit("ADDED", obj)
            return copy.deepcopy(obj)

    def get(self, gvk: GVK, namespace: str, name: str) -> Dict:
        with self._lock:
            obj = self._objects.get((gvk, namespace, name))
            if obj is None:
                raise NotFound(f"{gvk} {namespace}/{name}")
            return copy.deepcopy(obj)

    def try_get(self, gvk: GVK, namespace: str, name: str) -> Optional[Dict]:
        try:
            return self.get(gvk, namespace, name)
        except NotFound:
            return None

    def list(
        self,
        gvk: GVK,
        namespace: Optional[str] = None,
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict]:
        with self._lock:
            out = []
            for (g, ns, _), obj in self._objects.items():
                if g != gvk:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if label_selector:
                    labels = obj.get("metadata", {}).get("labels", {}) or {}
                    if any(labels.get(k) != v for k, v in label_selector.items()):
                        continue
                out.append(copy.deepcopy(obj))
            return out

    def update(self, obj: Dict) -> Dict:
        """Full replace of everything except status. Honors optimistic
        concurrency when metadata.resourceVersion is supplied."""
        obj = copy.deepcopy(obj)
        with self._lock:
            k = key_of(obj)
            cur = self._objects.get(k)
            if cur is None:
                raise NotFound(f"{k}")
            md = obj.setdefault("metadata", {})
            sent_rv = md.get("resourceVersion")
            if sent_rv is not None and sent_rv != cur["metadata"]["resourceVersion"]:
                raise Conflict(f"resourceVersion mismatch for {k}")
            # preserve server-managed fields + status
            for f in ("uid", "creationTimestamp", "generation"):
                md[f] = cur["metadata"][f]
            obj["status"] = cur.get("status", {})  # status via subresource only
            if self._spec_view(obj) != self._spec_view(cur):
                md["generation"] = cur["metadata"]["generation"] + 1
            md["resourceVersion"] = str(next(self._rv))
            self._objects[k] = obj
            self._emit("MODIFIED", obj)
            # finalizer removal completes a pending delete
            if obj["metadata"].get("deletionTimestamp") and not obj["metadata"].get(
                "finalizers"
            ):
                self._finalize_delete(k)
            return copy.deepcopy(self._objects.get(k, obj))

    def update_status(self, obj: Dict) -> Dict:
        """Status-subresource write: only .status changes; generation kept."""
        with self._lock:
            k = key_of(obj)
            cur = self._objects.get(k)
            if cur is None:
                raise NotFound(f"{k}")
            cur["status"] = copy.deepcopy(obj.get("status", {}))
            cur["metadata"]["resourceVersion"] = str(next(self._rv))
            self._emit("MODIFIED", cur)
            return copy.deepcopy(cur)

    def patch(self, gvk: GVK, namespace: str, name: str,
              patch: Dict) -> Dict:
        """Strategic-merge-ish patch: deep dict merge; None deletes a key."""
        with self._lock:
            cur = self._objects.get((gvk, namespace, name))
            if cur is None:
                raise NotFound(f"{gvk} {namespace}/{name}")
            before_spec = self._spec_view(cur)
            _deep_merge(cur, copy.deepcopy(patch))
            if self._spec_view(cur) != before_spec:
                cur["metadata"]["generation"] += 1
            cur["metadata"]["resourceVersion"] = str(next(self._rv))
            self._emit("MODIFIED", cur)
            if cur["metadata"].get("deletionTimestamp") and not cur[
                "metadata"
            ].get("finalizers"):
                self._finalize_delete((gvk, namespace, name))
            return copy.deepcopy(self._objects.get((gvk, namespace, name), cur))

    def delete(self, gvk: GVK, namespace: str, name: str) -> None:
        with self._lock:
            k = (gvk, namespace, name)
            cur = self._objects.get(k)
            if cur is None:
                raise NotFound(f"{gvk} {namespace}/{name}")
            if cur["metadata"].get("finalizers"):
                # graceful: mark for deletion, wait for finalizers
                if not cur["metadata"].get("deletionTimestamp"):
                    cur["metadata"]["deletionTimestamp"] = "now"
                    cur["metadata"]["resourceVersion"] = str(next(self._rv))
                    self._emit("MODIFIED", cur)
                return
            self._finalize_delete(k)

    def _finalize_delete(self, k: Key) -> None:
        obj = self._objects.pop(k, None)
        if obj is None:
            return
        self._emit("DELETED", obj)
        # ownerReference cascade (uid-matched)
        uid = obj["metadata"].get("uid")
        owned = [
            kk
            for kk, o in list(self._objects.items())
            if any(
                ref.get("uid") == uid
                for ref in o.get("metadata", {}).get("ownerReferences", []) or []
            )
        ]
        for kk in owned:
            g, ns, n = kk
            try:
                self.delete(g, ns, n)
            except NotFound:
                pass

    # -- watch -------------------------------------------------------------
    def watch(self, gvk: GVK, send_initial: bool = True) -> Watch:
        with self._lock:
            w = Watch(self, gvk)
            self._watches.setdefault(gvk, []).append(w)
            if send_initial:
                for (g, _, _), obj in self._objects.items():
                    if g == gvk:
                        w.events.put(WatchEvent("ADDED", copy.deepcopy(obj)))
            return w


def _deep_merge(dst: Dict, src: Dict) -> Dict:
    for k, v in src.items():
        if v is None:
            dst.pop(k, None)
        elif isinstance(v, dict) and isinstance(dst.get(k), dict):
            _deep_merge(dst[k], v)
        else:
            dst[k] = v
    return dst


def owner_reference(owner: Dict, controller: bool = True) -> Dict:
    return {
        "apiVersion": owner["apiVersion"],
        "kind": owner["kind"],
        "name": owner["metadata"]["name"],
        "uid": owner["metadata"]["uid"],
        "controller": controller,
        "blockOwnerDeletion": True,
    }


class KubectlAPIServer:
    """Real-cluster adapter over the kubectl CLI (same interface as
    FakeAPIServer minus watch, which shells out to ``kubectl get -w``).
    Thin by design: every controller behavior is exercised against the fake;
    this class only translates calls."""

    def __init__(self, kubectl: str = "kubectl"):
        self.kubectl = kubectl

    def _run(self, args: List[str], stdin: Optional[str] = None) -> str:
        import subprocess

        proc = subprocess.run(
            [self.kubectl, *args],
            input=stdin,
            capture_output=True,
            text=True,
        )
        if proc.returncode != 0:
            err = proc.stderr.strip()
            if "NotFound" in err or "not found" in err:
                raise NotFound(err)
            if "AlreadyExists" in err or "already exists" in err:
                raise AlreadyExists(err)
            raise APIError(500, err)
        return proc.stdout

    @staticmethod
    def _res(gvk: GVK) -> str:
        api_version, kind = gvk.rsplit("/", 1)
        group = api_version.split("/")[0] if "/" in api_version else ""
        return f"{kind.lower()}.{group}" if group else kind.lower()

    def create(self, obj: Dict) -> Dict:
        import json

        out = self._run(["create", "-o", "json", "-f", "-"], json.dumps(obj))
        return json.loads(out)

    def get(self, gvk: GVK, namespace: str, name: str) -> Dict:
        import json

        out = self._run(
            ["get", self._res(gvk), name, "-n", namespace, "-o", "json"]
        )
        return json.loads(out)

    def try_get(self, gvk: GVK, namespace: str, name: str) -> Optional[Dict]:
        try:
            return self.get(gvk, namespace, name)
        except NotFound:
            return None

    def list(self, gvk: GVK, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None) -> List[Dict]:
        import json

        args = ["get", self._res(gvk), "-o", "json"]
        args += ["-n", namespace] if namespace else ["-A"]
        if label_selector:
            args += ["-l", ",".join(f"{k}={v}" for k, v in label_selector.items())]
        return json.loads(self._run(args)).get("items", [])

    def update(self, obj: Dict) -> Dict:
        import json

        out = self._run(["replace", "-o", "json", "-f", "-"], json.dumps(obj))
        return json.loads(out)

    def update_status(self, obj: Dict) -> Dict:
        import json

        md = obj["metadata"]
        patch = json.dumps({"status": obj.get("status", {})})
        out = self._run(
            ["patch", self._res(gvk_of(obj)), md["name"], "-n",
             md.get("namespace", "default"), "--subresource=status",
             "--type=merge", "-p", patch, "-o", "json"]
        )
        return json.loads(out)

    def patch(self, gvk: GVK, namespace: str, name: str, patch: Dict) -> Dict:
        import json

        out = self._run(
            ["patch", self._res(gvk), name, "-n", namespace,
             "--type=merge", "-p", json.dumps(patch), "-o", "json"]
        )
        return json.loads(out)

    def delete(self, gvk: GVK, namespace: str, name: str) -> None:
        self._run(["delete", self._res(gvk), name, "-n", namespace,
                   "--wait=false"])
