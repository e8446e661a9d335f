"""In-memory Kubernetes apiserver for the live control plane.

Round 1 shipped a dict store with resourceVersion bumping; this is the
envtest-grade upgrade (VERDICT round-1 item 1/2): the semantics the
reference's integration tier gets from real kube-apiserver binaries
(reference pkg/controller/suite_test.go:88-128) are modeled here —

- optimistic concurrency: update/update_status with a stale
  metadata.resourceVersion raises ConflictError (HTTP 409 analog);
- uid assignment on create and metadata.generation bumping on spec
  change (k8s bumps generation only for spec, not metadata/status);
- watch event streams (ADDED/MODIFIED/DELETED) fanned out to
  subscribers — what drives the manager's requeue-on-child-event loop
  (reference SetupWithManager Owns(×10),
  inferenceservice_controller.go:689-704);
- ownerReference garbage collection: deleting an owner cascade-deletes
  its dependents (matched by uid), emitting DELETED events;
- coordination Leases for leader election (reference cmd/main.go:84-86).

Thread-safe; no controllers run inside the store itself — LWS status is
set by the lws stub controller or a test helper, mirroring envtest where
no kubelet/LWS controller exists.
"""

from __future__ import annotations

import copy
import itertools
import queue
import threading
import uuid as uuid_mod
from typing import Any, Dict, List, Optional, Tuple

Key = Tuple[str, str, str]  # (kind, namespace, name)


class NotFoundError(KeyError):
    pass


class ConflictError(RuntimeError):
    """Stale resourceVersion on a write (kube-apiserver HTTP 409)."""


class AlreadyExistsError(ValueError):
    pass


class Watch:
    """One subscriber's event stream. Iterate or poll() with timeout."""

    def __init__(self, kinds: Optional[set] = None):
        self.kinds = kinds
        self._q: "queue.Queue[Tuple[str, Dict[str, Any]]]" = queue.Queue()
        self.closed = False

    def _emit(self, event_type: str, obj: Dict[str, Any]) -> None:
        if self.kinds is None or obj["kind"] in self.kinds:
            self._q.put((event_type, obj))

    def poll(self, timeout: Optional[float] = None):
        """Next (event_type, obj) or None on timeout/close."""
        try:
            ev = self._q.get(timeout=timeout)
        except queue.Empty:
            return None
        return None if ev is _CLOSE else ev

    def close(self) -> None:
        self.closed = True
        self._q.put(_CLOSE)


_CLOSE = ("__closed__", {})


class FakeClient:
    """In-memory apiserver (name kept for round-1 compatibility; the
    HTTP layer in httpapi.py serves this same store over REST)."""

    def __init__(self):
        self._objects: Dict[Key, Dict[str, Any]] = {}
        self._rv = itertools.count(1)
        self._lock = threading.RLock()
        self._watches: List[Watch] = []

    @staticmethod
    def _key(obj: Dict[str, Any]) -> Key:
        md = obj["metadata"]
        return (obj["kind"], md.get("namespace", "default"), md["name"])

    def _broadcast(self, event_type: str, obj: Dict[str, Any]) -> None:
        for w in self._watches:
            if not w.closed:
                w._emit(event_type, copy.deepcopy(obj))

    # ------------------------------------------------------------- verbs
    def create(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        with self._lock:
            key = self._key(obj)
            if key in self._objects:
                raise AlreadyExistsError(f"already exists: {key}")
            stored = copy.deepcopy(obj)
            md = stored["metadata"]
            md["resourceVersion"] = str(next(self._rv))
            md.setdefault("uid", str(uuid_mod.uuid4()))
            md.setdefault("generation", 1)
            self._objects[key] = stored
            self._broadcast("ADDED", stored)
            return copy.deepcopy(stored)

    def get(self, kind: str, name: str, namespace: str = "default") -> Dict[str, Any]:
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._objects:
                raise NotFoundError(key)
            return copy.deepcopy(self._objects[key])

    def try_get(self, kind: str, name: str, namespace: str = "default"):
        try:
            return self.get(kind, name, namespace)
        except NotFoundError:
            return None

    def update(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        """Spec/metadata update. If obj carries a resourceVersion it must
        match the stored one (optimistic concurrency); omitting it skips
        the check (k8s requires it; round-1 tests rely on the lenient
        form for direct store edits)."""
        with self._lock:
            key = self._key(obj)
            if key not in self._objects:
                raise NotFoundError(key)
            current = self._objects[key]
            sent_rv = obj["metadata"].get("resourceVersion")
            if sent_rv is not None and sent_rv != current["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"{key}: stale resourceVersion {sent_rv} "
                    f"(current {current['metadata']['resourceVersion']})"
                )
            stored = copy.deepcopy(obj)
            md = stored["metadata"]
            md.setdefault("uid", current["metadata"].get("uid"))
            # generation bumps only when spec changed (k8s semantics)
            old_gen = int(current["metadata"].get("generation", 1))
            if stored.get("spec") != current.get("spec"):
                md["generation"] = old_gen + 1
            else:
                md["generation"] = old_gen
            # preserve status unless explicitly set
            if "status" not in stored and "status" in current:
                stored["status"] = copy.deepcopy(current["status"])
            # no-op writes keep resourceVersion and emit no event (k8s
            # semantics — what lets the level-triggered watch loop
            # converge instead of reconciling its own identical writes)
            md["resourceVersion"] = current["metadata"]["resourceVersion"]
            if stored == current:
                return copy.deepcopy(current)
            md["resourceVersion"] = str(next(self._rv))
            self._objects[key] = stored
            self._broadcast("MODIFIED", stored)
            return copy.deepcopy(stored)

    def update_status(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        """Status subresource update: only .status is taken from obj;
        conflicts on stale resourceVersion like the main resource."""
        with self._lock:
            key = self._key(obj)
            if key not in self._objects:
                raise NotFoundError(key)
            current = self._objects[key]
            sent_rv = obj["metadata"].get("resourceVersion")
            if sent_rv is not None and sent_rv != current["metadata"]["resourceVersion"]:
                raise ConflictError(
                    f"{key}: stale resourceVersion {sent_rv} on status "
                    f"(current {current['metadata']['resourceVersion']})"
                )
            new_status = copy.deepcopy(obj.get("status", {}))
            if current.get("status") == new_status:
                return copy.deepcopy(current)  # no-op: no rv bump, no event
            current["status"] = new_status
            current["metadata"]["resourceVersion"] = str(next(self._rv))
            self._broadcast("MODIFIED", current)
            return copy.deepcopy(current)

    def delete(self, kind: str, name: str, namespace: str = "default") -> None:
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._objects:
                raise NotFoundError(key)
            obj = self._objects.pop(key)
            self._broadcast("DELETED", obj)
            self._gc_dependents(obj)

    def _gc_dependents(self, owner: Dict[str, Any]) -> None:
        """ownerReference cascade delete (kube GC, foreground-equivalent:
        synchronous here)."""
        uid = owner["metadata"].get("uid")
        if not uid:
            return
        dependents = [
            k for k, o in self._objects.items()
            if any(
                ref.get("uid") == uid
                for ref in o["metadata"].get("ownerReferences", [])
            )
        ]
        for k in dependents:
            if k in self._objects:  # may already be gone via recursion
                dep = self._objects.pop(k)
                self._broadcast("DELETED", dep)
                self._gc_dependents(dep)

    def list(
        self,
        kind: str,
        namespace: Optional[str] = "default",
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict[str, Any]]:
        with self._lock:
            out = []
            for (k, ns, _), obj in sorted(self._objects.items()):
                if k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                labels = obj["metadata"].get("labels", {})
                if label_selector and any(
                    labels.get(lk) != lv for lk, lv in label_selector.items()
                ):
                    continue
                out.append(copy.deepcopy(obj))
            return out

    # ------------------------------------------------------------- watch
    def watch(self, kinds: Optional[List[str]] = None,
              send_initial: bool = False) -> Watch:
        """Subscribe to events. With send_initial, current objects are
        replayed as ADDED first (k8s list+watch bootstrap)."""
        w = Watch(set(kinds) if kinds else None)
        with self._lock:
            if send_initial:
                for _, obj in sorted(self._objects.items()):
                    w._emit("ADDED", copy.deepcopy(obj))
            self._watches.append(w)
        return w

    def stop_watch(self, w: Watch) -> None:
        with self._lock:
            if w in self._watches:
                self._watches.remove(w)
        w.close()

    # ---------------------------------------------------- leader election
    def acquire_lease(self, name: str, identity: str, duration_s: float,
                      now: float, namespace: str = "default") -> bool:
        """coordination.k8s.io Lease acquire/renew. Returns True if this
        identity holds the lease after the call."""
        with self._lock:
            lease = self.try_get("Lease", name, namespace)
            if lease is None:
                self.create({
                    "kind": "Lease",
                    "metadata": {"name": name, "namespace": namespace},
                    "spec": {"holderIdentity": identity,
                             "renewTime": now,
                             "leaseDurationSeconds": duration_s},
                })
                return True
            spec = lease.get("spec", {})
            holder = spec.get("holderIdentity")
            expired = now - float(spec.get("renewTime", 0)) > float(
                spec.get("leaseDurationSeconds", duration_s)
            )
            if holder == identity or expired:
                lease["spec"] = {"holderIdentity": identity,
                                 "renewTime": now,
                                 "leaseDurationSeconds": duration_s}
                try:
                    self.update(lease)
                except ConflictError:
                    return False
                return True
            return False

    # ------------------------------------------------------ test helpers
    def set_lws_ready(self, name: str, namespace: str = "default",
                      ready_replicas: int = 1) -> None:
        """Status-faking helper: mark an LWS ready (no LWS controller runs
        here, mirroring envtest — SURVEY.md §4.2)."""
        obj = self.get("LeaderWorkerSet", name, namespace)
        obj["status"] = {"replicas": 1, "readyReplicas": ready_replicas}
        self.update_status(obj)


# Explicit alias for new code; FakeClient name retained for round-1 users.
InMemoryAPIServer = FakeClient
