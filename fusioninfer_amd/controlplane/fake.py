"""In-memory Kubernetes-like API server for control-plane tests.

Plays the role the reference's envtest (real kube-apiserver binaries)
plays in its integration tier: objects are stored with
resourceVersion bumping and label-selector list, but no controllers run —
LWS status stays empty unless a test fakes it (the status-faking helper
the survey calls out as a gap to close).
"""

from __future__ import annotations

import copy
import itertools
from typing import Any, Dict, List, Optional, Tuple

Key = Tuple[str, str, str]  # (kind, namespace, name)


class NotFoundError(KeyError):
    pass


class FakeClient:
    def __init__(self):
        self._objects: Dict[Key, Dict[str, Any]] = {}
        self._rv = itertools.count(1)

    @staticmethod
    def _key(obj: Dict[str, Any]) -> Key:
        md = obj["metadata"]
        return (obj["kind"], md.get("namespace", "default"), md["name"])

    # ------------------------------------------------------------- verbs
    def create(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        key = self._key(obj)
        if key in self._objects:
            raise ValueError(f"already exists: {key}")
        stored = copy.deepcopy(obj)
        stored["metadata"]["resourceVersion"] = str(next(self._rv))
        self._objects[key] = stored
        return copy.deepcopy(stored)

    def get(self, kind: str, name: str, namespace: str = "default") -> Dict[str, Any]:
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
        key = self._key(obj)
        if key not in self._objects:
            raise NotFoundError(key)
        stored = copy.deepcopy(obj)
        stored["metadata"]["resourceVersion"] = str(next(self._rv))
        # preserve status unless explicitly set (spec updates don't clear it)
        if "status" not in stored and "status" in self._objects[key]:
            stored["status"] = copy.deepcopy(self._objects[key]["status"])
        self._objects[key] = stored
        return copy.deepcopy(stored)

    def update_status(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        key = self._key(obj)
        if key not in self._objects:
            raise NotFoundError(key)
        self._objects[key]["status"] = copy.deepcopy(obj.get("status", {}))
        self._objects[key]["metadata"]["resourceVersion"] = str(next(self._rv))
        return copy.deepcopy(self._objects[key])

    def delete(self, kind: str, name: str, namespace: str = "default") -> None:
        key = (kind, namespace, name)
        if key not in self._objects:
            raise NotFoundError(key)
        del self._objects[key]

    def list(
        self,
        kind: str,
        namespace: Optional[str] = "default",
        label_selector: Optional[Dict[str, str]] = None,
    ) -> List[Dict[str, Any]]:
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

    # ------------------------------------------------------ test helpers
    def set_lws_ready(self, name: str, namespace: str = "default",
                      ready_replicas: int = 1) -> None:
        """Status-faking helper: mark an LWS ready (envtest runs no LWS
        controller — SURVEY.md §4.2 notes readiness aggregation was
        untested in the reference for exactly this reason)."""
        obj = self.get("LeaderWorkerSet", name, namespace)
        obj["status"] = {"replicas": 1, "readyReplicas": ready_replicas}
        self.update_status(obj)
