"""Controller manager: the live, watch-driven reconcile loop.

Round 1 had reconcile-as-a-function; this is the runtime around it,
mirroring controller-runtime's manager (reference cmd/main.go:68-222 and
SetupWithManager's For(InferenceService).Owns(×10 kinds),
pkg/controller/inferenceservice_controller.go:689-704):

- a watch on InferenceService enqueues the object itself;
- watches on every owned kind map child events back to the owning
  InferenceService via ownerReferences and requeue it (level-triggered:
  deleting a child re-creates it on the next pass, a child status change
  re-aggregates into .status.components);
- a deduplicating work queue + worker that runs the single-pass
  reconciler with retry-on-conflict (optimistic-concurrency loop around
  the ONE status update, reference :145-148);
- /healthz + /readyz probe endpoints and a Prometheus /metrics endpoint
  with controller-runtime metric names (controller_runtime_reconcile_total
  — what the reference's e2e tier asserts, test/e2e/e2e_test.go:176-261);
- optional leader election through coordination Leases
  (cmd/main.go:84-86): only the elected manager reconciles.
"""

from __future__ import annotations

import http.server
import threading
import time
import uuid
from typing import Dict, Optional, Set, Tuple

from fusioninfer_amd.controlplane.fake import ConflictError, FakeClient
from fusioninfer_amd.controlplane.reconciler import (
    InferenceServiceReconciler,
    ModelLoaderReconciler,
)

#: every kind the controller owns (reference Owns() list, :689-704)
OWNED_KINDS = [
    "PodGroup",
    "LeaderWorkerSet",
    "ServiceAccount",
    "Role",
    "RoleBinding",
    "ConfigMap",
    "Deployment",
    "Service",
    "InferencePool",
    "HTTPRoute",
]


#: controller-runtime's reconcile-duration bucket boundaries (seconds)
_TIME_BUCKETS = [0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0,
                 10.0]


class _Metrics:
    def __init__(self):
        self.lock = threading.Lock()
        self.reconcile_total: Dict[str, int] = {}
        self.requeues = 0
        self.time_hist = [0] * (len(_TIME_BUCKETS) + 1)
        self.time_sum = 0.0
        self.time_count = 0

    def inc(self, result: str) -> None:
        with self.lock:
            self.reconcile_total[result] = self.reconcile_total.get(result, 0) + 1

    def observe(self, seconds: float) -> None:
        with self.lock:
            self.time_sum += seconds
            self.time_count += 1
            for i, b in enumerate(_TIME_BUCKETS):
                if seconds <= b:
                    self.time_hist[i] += 1
                    return
            self.time_hist[-1] += 1

    def render(self, queue_depth: int) -> str:
        with self.lock:
            lines = [
                "# TYPE controller_runtime_reconcile_total counter",
            ]
            for result, n in sorted(self.reconcile_total.items()):
                lines.append(
                    "controller_runtime_reconcile_total{controller="
                    f'"inferenceservice",result="{result}"}} {n}'
                )
            lines.append("# TYPE workqueue_depth gauge")
            lines.append(
                'workqueue_depth{name="inferenceservice"} ' + str(queue_depth)
            )
            lines.append("# TYPE controller_runtime_reconcile_requeues counter")
            lines.append(
                "controller_runtime_reconcile_requeues{controller="
                f'"inferenceservice"}} {self.requeues}'
            )
            lines.append(
                "# TYPE controller_runtime_reconcile_time_seconds histogram"
            )
            cum = 0
            for b, c in zip(_TIME_BUCKETS, self.time_hist):
                cum += c
                lines.append(
                    "controller_runtime_reconcile_time_seconds_bucket"
                    f'{{controller="inferenceservice",le="{b}"}} {cum}'
                )
            cum += self.time_hist[-1]
            lines.append(
                "controller_runtime_reconcile_time_seconds_bucket"
                f'{{controller="inferenceservice",le="+Inf"}} {cum}'
            )
            lines.append(
                "controller_runtime_reconcile_time_seconds_sum"
                f'{{controller="inferenceservice"}} {self.time_sum}'
            )
            lines.append(
                "controller_runtime_reconcile_time_seconds_count"
                f'{{controller="inferenceservice"}} {self.time_count}'
            )
            return "\n".join(lines) + "\n"


class Manager:
    """Watch-driven controller manager over an apiserver client."""

    def __init__(
        self,
        client: FakeClient,
        probe_port: int = 0,
        metrics_port: int = 0,
        leader_elect: bool = False,
        lease_name: str = "fusioninfer-leader",
        identity: Optional[str] = None,
        lease_duration_s: float = 15.0,
        renew_period_s: float = 2.0,
        max_conflict_retries: int = 8,
    ):
        self.client = client
        self.reconciler = InferenceServiceReconciler(client)
        self.modelloader_reconciler = ModelLoaderReconciler(client)
        self.metrics = _Metrics()
        self.leader_elect = leader_elect
        self.lease_name = lease_name
        self.identity = identity or f"manager-{uuid.uuid4().hex[:8]}"
        self.lease_duration_s = lease_duration_s
        self.renew_period_s = renew_period_s
        self.max_conflict_retries = max_conflict_retries
        self.is_leader = not leader_elect
        self._probe_port = probe_port
        self._metrics_port = metrics_port
        self._queue: Set[Tuple[str, str, str]] = set()  # (kind, ns, name)
        self._cv = threading.Condition()
        self._stop = threading.Event()
        self._busy = 0
        self._threads = []
        self._servers = []
        self.probe_addr: Optional[Tuple[str, int]] = None
        self.metrics_addr: Optional[Tuple[str, int]] = None
        self._ready = threading.Event()

    # ------------------------------------------------------------ queue
    def enqueue(self, kind: str, namespace: str, name: str) -> None:
        with self._cv:
            self._queue.add((kind, namespace, name))
            self._cv.notify_all()

    def _map_event(self, obj: Dict) -> None:
        """Route an event to the work queue: the primary resource itself,
        or a child's controller ownerReference (reference Owns() mapping)."""
        kind = obj["kind"]
        ns = obj["metadata"].get("namespace", "default")
        if kind in ("InferenceService", "ModelLoader"):
            self.enqueue(kind, ns, obj["metadata"]["name"])
            return
        for ref in obj["metadata"].get("ownerReferences", []):
            if ref.get("kind") == "InferenceService" and ref.get("controller"):
                self.enqueue("InferenceService", ns, ref["name"])

    # ----------------------------------------------------------- workers
    def _watch_loop(self) -> None:
        w = self.client.watch(
            kinds=["InferenceService", "ModelLoader"] + OWNED_KINDS,
            send_initial=True,
        )
        self._watch = w
        self._ready.set()
        while not self._stop.is_set():
            ev = w.poll(timeout=0.2)
            if ev is None:
                continue
            _, obj = ev
            self._map_event(obj)
        self.client.stop_watch(w)

    def _work_loop(self) -> None:
        while not self._stop.is_set():
            with self._cv:
                while not self._queue and not self._stop.is_set():
                    self._cv.wait(timeout=0.2)
                if self._stop.is_set():
                    return
                key = self._queue.pop()
                self._busy += 1
            try:
                self._process(key)
            finally:
                with self._cv:
                    self._busy -= 1
                    self._cv.notify_all()

    def _process(self, key: Tuple[str, str, str]) -> None:
        kind, ns, name = key
        if not self.is_leader:
            return
        t0 = time.monotonic()
        try:
            self._process_inner(kind, ns, name)
        finally:
            self.metrics.observe(time.monotonic() - t0)

    def _process_inner(self, kind: str, ns: str, name: str) -> None:
        try:
            if kind == "ModelLoader":
                self.modelloader_reconciler.reconcile(name, ns)
                self.metrics.inc("success")
                return
            for attempt in range(self.max_conflict_retries):
                try:
                    self.reconciler.reconcile(name, ns)
                    self.metrics.inc("success")
                    return
                except ConflictError:
                    # optimistic-concurrency retry: re-read + re-reconcile
                    self.metrics.requeues += 1
                    time.sleep(0.01 * (attempt + 1))
            self.metrics.inc("error")
        except Exception:
            self.metrics.inc("error")
            # level-triggered: next event for this object retries

    # ------------------------------------------------------ leader loop
    def _leader_loop(self) -> None:
        while not self._stop.is_set():
            got = self.client.acquire_lease(
                self.lease_name,
                self.identity,
                self.lease_duration_s,
                now=time.monotonic(),
            )
            if got != self.is_leader:
                self.is_leader = got
                if got:
                    # became leader: full resync of primaries
                    for obj in self.client.list("InferenceService",
                                                namespace=None):
                        self._map_event(obj)
            self._stop.wait(self.renew_period_s)

    # ------------------------------------------------------- http servers
    def _serve(self, port: int, handler_cls) -> Tuple[str, int]:
        srv = http.server.ThreadingHTTPServer(("127.0.0.1", port), handler_cls)
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        self._servers.append(srv)
        self._threads.append(t)
        return srv.server_address

    def _start_http(self) -> None:
        mgr = self

        class ProbeHandler(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                if self.path in ("/healthz", "/readyz"):
                    ok = not mgr._stop.is_set()
                    self.send_response(200 if ok else 503)
                    self.end_headers()
                    self.wfile.write(b"ok" if ok else b"stopping")
                else:
                    self.send_response(404)
                    self.end_headers()

            def log_message(self, *a):  # quiet
                pass

        class MetricsHandler(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                if self.path == "/metrics":
                    body = mgr.metrics.render(len(mgr._queue)).encode()
                    self.send_response(200)
                    self.send_header("Content-Type", "text/plain")
                    self.end_headers()
                    self.wfile.write(body)
                else:
                    self.send_response(404)
                    self.end_headers()

            def log_message(self, *a):
                pass

        if self._probe_port is not None:
            self.probe_addr = self._serve(self._probe_port, ProbeHandler)
        if self._metrics_port is not None:
            self.metrics_addr = self._serve(self._metrics_port, MetricsHandler)

    # ------------------------------------------------------------- start
    def start(self) -> "Manager":
        self._start_http()
        t = threading.Thread(target=self._watch_loop, daemon=True)
        t.start()
        self._threads.append(t)
        self._ready.wait(timeout=5.0)
        w = threading.Thread(target=self._work_loop, daemon=True)
        w.start()
        self._threads.append(w)
        if self.leader_elect:
            le = threading.Thread(target=self._leader_loop, daemon=True)
            le.start()
            self._threads.append(le)
        return self

    def stop(self) -> None:
        self._stop.set()
        with self._cv:
            self._cv.notify_all()
        for srv in self._servers:
            srv.shutdown()
        for t in self._threads:
            t.join(timeout=2.0)

    def _pending(self) -> bool:
        w = getattr(self, "_watch", None)
        return bool(self._queue or self._busy
                    or (w is not None and not w._q.empty()))

    def wait_idle(self, timeout: float = 5.0) -> bool:
        """Block until watch events are drained, the queue is empty and no
        worker is busy — observed twice to close the watcher hand-off race."""
        deadline = time.monotonic() + timeout
        idle_streak = 0
        while time.monotonic() < deadline:
            if self._pending():
                idle_streak = 0
                time.sleep(0.01)
                continue
            idle_streak += 1
            if idle_streak >= 3:
                return True
            time.sleep(0.02)
        return not self._pending()
