"""Stub child controllers for live-loop testing.

The reference's envtest runs no LWS/Volcano controllers, so LWS status
stays empty and readiness aggregation is untested end-to-end (SURVEY.md
§4.2 calls this out as the gap to close). This stub IS that closure: a
watch-driven controller that marks every LeaderWorkerSet ready shortly
after creation — the control-plane analog of the CPU stub worker
container in BASELINE config #1 — so the manager's child-event requeue
path and Active-condition aggregation run live.
"""

from __future__ import annotations

import threading

from fusioninfer_amd.controlplane.fake import ConflictError, FakeClient


class StubLWSController:
    """Marks LeaderWorkerSets ready. With delay_s > 0 readiness arrives
    asynchronously (exercises Pending -> Deploying -> Running)."""

    def __init__(self, client: FakeClient, delay_s: float = 0.0,
                 ready: bool = True):
        self.client = client
        self.delay_s = delay_s
        self.ready = ready
        self._stop = threading.Event()
        self._thread = None

    def start(self) -> "StubLWSController":
        self._watch = self.client.watch(kinds=["LeaderWorkerSet"],
                                        send_initial=True)
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def _loop(self) -> None:
        while not self._stop.is_set():
            ev = self._watch.poll(timeout=0.2)
            if ev is None:
                continue
            event_type, obj = ev
            if event_type == "DELETED" or not self.ready:
                continue
            if self.delay_s:
                self._stop.wait(self.delay_s)
            md = obj["metadata"]
            replicas = int(obj.get("spec", {}).get("replicas", 1))
            st = {"replicas": replicas, "readyReplicas": replicas}
            if obj.get("status") == st:
                continue
            obj = dict(obj)
            obj["status"] = st
            try:
                self.client.update_status(obj)
            except (ConflictError, KeyError):
                pass  # deleted or raced; a later event retries

    def stop(self) -> None:
        self._stop.set()
        self.client.stop_watch(self._watch)
        if self._thread:
            self._thread.join(timeout=2.0)
