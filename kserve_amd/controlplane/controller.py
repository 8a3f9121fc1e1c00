"""Controller runtime: watch → work queue → reconcile, with requeue/backoff.

The reference uses controller-runtime's manager: each controller owns a
rate-limited work queue fed by watches, workers call Reconcile(ctx, req)
which returns (Result{Requeue, RequeueAfter}, err), and create-or-update
writes go through a semantic diff so no-op updates are elided
(pkg/controller/v1beta1/inferenceservice/controller.go:122-455,
reconcilers/deployment/deployment_reconciler.go:544 create-or-update,
updateStatus equality short-circuit controller.go:420-455).

This module is that runtime in Python against the ``APIServer`` interface
(apiserver.py): ``Controller`` wires a primary watch plus owned-resource
watches (mapped back to the owner key) into a deduplicating ``WorkQueue``;
``run_once``/``run`` drain it; ``create_or_update`` applies desired
manifests with owner references and a subset-aware semantic diff.
"""

from __future__ import annotations

import copy
import threading
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from kserve_amd.controlplane.apiserver import (
    Conflict,
    FakeAPIServer,
    NotFound,
    WatchEvent,
    gvk_of,
    owner_reference,
)

ReconcileKey = Tuple[str, str]  # (namespace, name)


@dataclass
class Result:
    requeue_after: Optional[float] = None  # seconds; None = done


class WorkQueue:
    """Deduplicating queue with per-key exponential backoff on failure
    (controller-runtime's rate-limited queue semantics, simplified)."""

    def __init__(self, base_delay: float = 0.005, max_delay: float = 2.0):
        self._lock = threading.Condition()
        self._pending: List[ReconcileKey] = []
        self._in_queue: set = set()
        self._failures: Dict[ReconcileKey, int] = {}
        self._delayed: List[Tuple[float, ReconcileKey]] = []
        self.base_delay = base_delay
        self.max_delay = max_delay

    def add(self, key: ReconcileKey) -> None:
        with self._lock:
            if key not in self._in_queue:
                self._pending.append(key)
                self._in_queue.add(key)
                self._lock.notify()

    def add_after(self, key: ReconcileKey, delay: float) -> None:
        with self._lock:
            self._delayed.append((time.monotonic() + delay, key))
            self._lock.notify()

    def add_rate_limited(self, key: ReconcileKey) -> None:
        with self._lock:
            n = self._failures.get(key, 0)
            self._failures[key] = n + 1
        self.add_after(key, min(self.base_delay * (2 ** n), self.max_delay))

    def forget(self, key: ReconcileKey) -> None:
        with self._lock:
            self._failures.pop(key, None)

    def _promote_due(self) -> None:
        now = time.monotonic()
        due = [k for t, k in self._delayed if t <= now]
        self._delayed = [(t, k) for t, k in self._delayed if t > now]
        for k in due:
            if k not in self._in_queue:
                self._pending.append(k)
                self._in_queue.add(k)

    def get(self, timeout: Optional[float] = None) -> Optional[ReconcileKey]:
        deadline = None if timeout is None else time.monotonic() + timeout
        with self._lock:
            while True:
                self._promote_due()
                if self._pending:
                    key = self._pending.pop(0)
                    self._in_queue.discard(key)
                    return key
                wait = None
                if self._delayed:
                    wait = max(0.0, min(t for t, _ in self._delayed) - time.monotonic())
                if deadline is not None:
                    rem = deadline - time.monotonic()
                    if rem <= 0:
                        return None
                    wait = rem if wait is None else min(wait, rem)
                if wait is None:
                    self._lock.wait()
                else:
                    self._lock.wait(wait)

    def empty(self) -> bool:
        with self._lock:
            self._promote_due()
            return not self._pending and not self._delayed


class Controller:
    """One controller: a primary-kind watch plus owned-kind watches mapped
    back to the owner, feeding a WorkQueue drained by ``reconcile_fn``."""

    def __init__(
        self,
        server,
        primary_gvk: str,
        reconcile_fn: Callable[[ReconcileKey], Optional[Result]],
        owned_gvks: Tuple[str, ...] = (),
        owner_label: Optional[str] = None,
    ):
        self.server = server
        self.primary_gvk = primary_gvk
        self.reconcile_fn = reconcile_fn
        self.owned_gvks = owned_gvks
        # owned objects map back to the owner by this label when the
        # ownerReference is absent (e.g. cluster-scoped watch fan-in)
        self.owner_label = owner_label
        self.queue = WorkQueue()
        self._watches = []
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()

    # -- event plumbing ----------------------------------------------------
    def _enqueue_from_event(self, ev: WatchEvent, primary: bool) -> None:
        md = ev.object.get("metadata", {})
        ns = md.get("namespace", "")
        if primary:
            self.queue.add((ns, md["name"]))
            return
        for ref in md.get("ownerReferences", []) or []:
            if ref.get("controller"):
                self.queue.add((ns, ref["name"]))
                return
        if self.owner_label:
            owner = (md.get("labels") or {}).get(self.owner_label)
            if owner:
                self.queue.add((ns, owner))

    def start_watches(self) -> None:
        if self._watches:  # idempotent: build() subscribes, start() reuses
            return
        w = self.server.watch(self.primary_gvk)
        self._watches.append((w, True))
        for g in self.owned_gvks:
            self._watches.append((self.server.watch(g), False))

    def pump_events(self, budget: Optional[float] = 0.0) -> int:
        """Drain available watch events into the queue (non-blocking when
        budget=0). Returns number of events consumed."""
        n = 0
        for w, primary in self._watches:
            while True:
                ev = w.next(timeout=budget)
                if ev is None:
                    break
                self._enqueue_from_event(ev, primary)
                n += 1
        return n

    # -- drive -------------------------------------------------------------
    def process_one(self, timeout: float = 0.0) -> bool:
        key = self.queue.get(timeout=timeout)
        if key is None:
            return False
        try:
            res = self.reconcile_fn(key)
        except Conflict:
            # stale read: immediate retry re-reads fresh state
            self.queue.add_rate_limited(key)
            return True
        except Exception:
            self.queue.add_rate_limited(key)
            return True
        self.queue.forget(key)
        if res and res.requeue_after is not None:
            self.queue.add_after(key, res.requeue_after)
        return True

    def run_until_idle(self, max_seconds: float = 5.0) -> None:
        """Test harness drive: pump events + process until both the event
        streams and the queue are quiet (or the deadline passes)."""
        deadline = time.monotonic() + max_seconds
        idle_rounds = 0
        while time.monotonic() < deadline:
            moved = self.pump_events(0.0) > 0
            moved |= self.process_one(timeout=0.01)
            if moved:
                idle_rounds = 0
                continue
            if self.queue.empty():
                idle_rounds += 1
                if idle_rounds >= 2:
                    return
            time.sleep(0.005)

    def start(self) -> None:
        """Background mode (real deployments): one pump thread + one worker."""
        self.start_watches()

        def pump():
            while not self._stop.is_set():
                if self.pump_events(budget=0.05) == 0:
                    time.sleep(0.01)

        def work():
            while not self._stop.is_set():
                self.process_one(timeout=0.1)

        for fn in (pump, work):
            t = threading.Thread(target=fn, daemon=True)
            t.start()
            self._threads.append(t)

    def stop(self) -> None:
        self._stop.set()
        for w, _ in self._watches:
            w.stop()
        for t in self._threads:
            t.join(timeout=1.0)


# -- apply helpers ----------------------------------------------------------

def _is_subset(desired, current) -> bool:
    """True when every field in desired equals the corresponding field in
    current (dicts recursively; lists compared whole). The semantic-equality
    guard: server-defaulted extra fields in current don't count as drift."""
    if isinstance(desired, dict) and isinstance(current, dict):
        return all(k in current and _is_subset(v, current[k]) for k, v in desired.items())
    return desired == current


def create_or_update(server, desired: Dict, owner: Optional[Dict] = None) -> Dict:
    """controller-runtime CreateOrUpdate: create if absent, else merge the
    desired fields over the live object and write only when something
    actually changes (semantic diff)."""
    desired = copy.deepcopy(desired)
    md = desired.setdefault("metadata", {})
    if owner is not None:
        refs = md.setdefault("ownerReferences", [])
        if not any(r.get("uid") == owner["metadata"]["uid"] for r in refs):
            refs.append(owner_reference(owner))
    ns, name = md.get("namespace", ""), md["name"]
    g = gvk_of(desired)
    current = server.try_get(g, ns, name)
    if current is None:
        return server.create(desired)
    if _is_subset(_manifest_fields(desired), current):
        return current
    merged = copy.deepcopy(current)
    _merge_manifest(merged, desired)
    merged["metadata"]["resourceVersion"] = current["metadata"]["resourceVersion"]
    return server.update(merged)


def _manifest_fields(obj: Dict) -> Dict:
    """The fields a manifest asserts (drop empty metadata maps)."""
    out = {k: v for k, v in obj.items() if k not in ("status",)}
    return out


def _merge_manifest(dst: Dict, src: Dict) -> None:
    """Overlay manifest fields: spec replaced wholesale (the controller owns
    it), metadata merged key-wise (labels/annotations/ownerReferences)."""
    for k, v in src.items():
        if k == "metadata":
            for mk, mv in v.items():
                if mk in ("labels", "annotations") and isinstance(
                    dst["metadata"].get(mk), dict
                ):
                    dst["metadata"][mk].update(mv)
                elif mk == "ownerReferences":
                    have = dst["metadata"].setdefault("ownerReferences", [])
                    for r in mv:
                        if not any(h.get("uid") == r.get("uid") for h in have):
                            have.append(r)
                elif mk not in ("resourceVersion", "uid", "generation",
                                "creationTimestamp"):
                    dst["metadata"][mk] = mv
        elif k != "status":
            dst[k] = copy.deepcopy(v)


def delete_if_exists(server, gvk: str, namespace: str, name: str) -> None:
    try:
        server.delete(gvk, namespace, name)
    except NotFound:
        pass


# -- status conditions -------------------------------------------------------

def set_condition(status: Dict, ctype: str, cstatus: str,
                  reason: str = "", message: str = "") -> bool:
    """Upsert a condition; returns True if it changed (knative-style
    condition set, reference inference_service_status.go)."""
    conds = status.setdefault("conditions", [])
    for c in conds:
        if c["type"] == ctype:
            if (c["status"], c.get("reason", ""), c.get("message", "")) == (
                cstatus, reason, message
            ):
                return False
            c.update(status=cstatus, reason=reason, message=message)
            return True
    conds.append(
        {"type": ctype, "status": cstatus, "reason": reason, "message": message}
    )
    return True


def get_condition(status: Dict, ctype: str) -> Optional[Dict]:
    for c in status.get("conditions", []) or []:
        if c["type"] == ctype:
            return c
    return None
