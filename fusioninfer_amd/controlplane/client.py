"""Typed client for external consumers (reference client-go parity).

The reference generates a typed Go clientset/informers/listers for the
InferenceService API (client-go/**, ~2.1k generated LoC; SURVEY.md §2.1
#15). The Python-native equivalent is this thin typed wrapper over any
store client (in-memory FakeClient or httpapi.HTTPClient): CRUD + watch
in terms of api.InferenceService objects instead of raw dicts, with the
typed informer analog (watch_typed) for controllers/tools built on top.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Tuple

from fusioninfer_amd.controlplane.api import InferenceService

KIND = "InferenceService"


class InferenceServiceClient:
    """Typed CRUD/watch over a duck-typed store client."""

    def __init__(self, client):
        self._c = client

    # ------------------------------------------------------------- CRUD
    def create(self, svc: InferenceService) -> InferenceService:
        return InferenceService.from_dict(self._c.create(svc.to_dict()))

    def get(self, name: str, namespace: str = "default") -> InferenceService:
        return InferenceService.from_dict(self._c.get(KIND, name, namespace))

    def try_get(self, name: str,
                namespace: str = "default") -> Optional[InferenceService]:
        obj = self._c.try_get(KIND, name, namespace)
        return InferenceService.from_dict(obj) if obj is not None else None

    def list(self, namespace: Optional[str] = "default",
             label_selector=None) -> List[InferenceService]:
        return [
            InferenceService.from_dict(o)
            for o in self._c.list(KIND, namespace, label_selector)
        ]

    def update(self, svc: InferenceService) -> InferenceService:
        """Optimistic-concurrency update: svc must carry the
        resourceVersion it was read at (from_dict preserves it)."""
        return InferenceService.from_dict(self._c.update(svc.to_dict()))

    def update_status(self, svc: InferenceService) -> InferenceService:
        return InferenceService.from_dict(
            self._c.update_status(svc.to_dict())
        )

    def delete(self, name: str, namespace: str = "default") -> None:
        self._c.delete(KIND, name, namespace)

    # ------------------------------------------------- informer analog
    def watch_typed(
        self, send_initial: bool = True
    ) -> Tuple[object, Callable[[Optional[float]],
               Optional[Tuple[str, InferenceService]]]]:
        """Returns (raw_watch, poll): poll(timeout) yields
        (event_type, InferenceService) or None. Stop with
        client.stop_watch(raw_watch)."""
        w = self._c.watch(kinds=[KIND], send_initial=send_initial)

        def poll(timeout: Optional[float] = None):
            ev = w.poll(timeout=timeout)
            if ev is None:
                return None
            etype, obj = ev
            return etype, InferenceService.from_dict(obj)

        return w, poll

    def stop_watch(self, w) -> None:
        self._c.stop_watch(w)
