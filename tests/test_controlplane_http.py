"""HTTP apiserver layer: the control plane across a real process boundary.

The store lives in one process and the manager in another, talking REST
+ watch streams — the offline analog of controller-runtime against
kube-apiserver (BASELINE config #1's 'controller against kind' plumbing).
Includes a true kill-and-resume: the manager process is SIGKILLed, state
survives in the store process, a fresh manager converges the backlog.
"""

import subprocess
import sys
import time

import pytest

from fusioninfer_amd.controlplane.fake import ConflictError, FakeClient
from fusioninfer_amd.controlplane.httpapi import HTTPClient, serve_store
from fusioninfer_amd.controlplane.manager import Manager
from fusioninfer_amd.controlplane.stubs import StubLWSController

from tests.test_controlplane import monolithic_svc
from tests.test_controlplane_live import _eventually


@pytest.fixture()
def http_store():
    store = FakeClient()
    srv = serve_store(store, port=0)
    host, port = srv.server_address
    client = HTTPClient(f"http://{host}:{port}")
    yield store, client
    srv.shutdown()


def test_http_verbs_roundtrip(http_store):
    store, client = http_store
    obj = client.create(
        {"kind": "ConfigMap", "metadata": {"name": "c", "labels": {"a": "1"}},
         "data": {"k": "v"}}
    )
    assert obj["metadata"]["resourceVersion"]
    assert client.get("ConfigMap", "c")["data"] == {"k": "v"}
    assert client.try_get("ConfigMap", "nope") is None

    # conflict on stale rv surfaces as ConflictError over HTTP 409
    fresh = client.get("ConfigMap", "c")
    stale = dict(fresh)
    fresh["data"] = {"k": "v2"}
    client.update(fresh)
    with pytest.raises(ConflictError):
        stale["data"] = {"k": "v3"}
        client.update(stale)

    # status subresource
    cur = client.get("ConfigMap", "c")
    cur["status"] = {"ok": True}
    client.update_status(cur)
    assert client.get("ConfigMap", "c")["status"] == {"ok": True}

    # list + label selector
    client.create({"kind": "ConfigMap",
                   "metadata": {"name": "d", "labels": {"a": "2"}}})
    assert len(client.list("ConfigMap")) == 2
    assert [o["metadata"]["name"]
            for o in client.list("ConfigMap", label_selector={"a": "1"})] == ["c"]

    # delete + ownerRef GC through HTTP
    owner = client.create({"kind": "InferenceService",
                           "metadata": {"name": "own"}})
    client.create({
        "kind": "Service",
        "metadata": {"name": "child", "ownerReferences": [{
            "kind": "InferenceService", "name": "own",
            "uid": owner["metadata"]["uid"], "controller": True,
        }]},
    })
    client.delete("InferenceService", "own")
    assert client.try_get("Service", "child") is None


def test_http_watch_stream(http_store):
    store, client = http_store
    w = client.watch(kinds=["ConfigMap"], send_initial=False)
    try:
        client.create({"kind": "ConfigMap", "metadata": {"name": "w1"}})
        ev = w.poll(timeout=3.0)
        assert ev is not None and ev[0] == "ADDED"
        assert ev[1]["metadata"]["name"] == "w1"
        client.delete("ConfigMap", "w1")
        ev = w.poll(timeout=3.0)
        assert ev[0] == "DELETED"
    finally:
        client.stop_watch(w)


def test_manager_over_http(http_store):
    """The full watch-driven manager running against the HTTP client."""
    store, client = http_store
    mgr = Manager(client).start()
    stub = StubLWSController(store).start()
    try:
        client.create(monolithic_svc("h1").to_dict())
        _eventually(lambda: client.try_get("LeaderWorkerSet", "h1-worker-0"),
                    timeout=8.0)
        svc = _eventually(
            lambda: (lambda o: o if any(
                c["type"] == "Active" and c["status"] == "True"
                for c in o.get("status", {}).get("conditions", [])
            ) else None)(client.get("InferenceService", "h1")),
            timeout=8.0,
        )
        assert svc["status"]["components"]["worker"]["phase"] == "Running"
    finally:
        stub.stop()
        mgr.stop()


def test_manager_subprocess_kill_and_resume(http_store, tmp_path):
    """Manager in a SEPARATE PROCESS against the store; SIGKILL it, state
    survives, a fresh manager process converges the unreconciled change."""
    store, client = http_store
    url = client.base

    def spawn():
        return subprocess.Popen(
            [sys.executable, "-m", "fusioninfer_amd.controlplane", "run",
             "--apiserver", url,
             "--health-probe-bind-address", ":0",
             "--metrics-bind-address", ":0"],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL,
        )

    stub = StubLWSController(store).start()
    proc = spawn()
    try:
        client.create(monolithic_svc("k1").to_dict())
        _eventually(lambda: client.try_get("LeaderWorkerSet", "k1-worker-0"),
                    timeout=15.0)
        proc.kill()  # exact PID of the process we spawned
        proc.wait(timeout=10)

        obj = client.get("InferenceService", "k1")
        obj["spec"]["roles"][0]["replicas"] = 2
        client.update(obj)
        time.sleep(0.3)
        assert len(client.list("LeaderWorkerSet")) == 1  # nobody home

        proc = spawn()
        _eventually(lambda: len(client.list("LeaderWorkerSet")) == 2,
                    timeout=15.0)
    finally:
        stub.stop()
        proc.kill()
        proc.wait(timeout=10)


def test_typed_client_crud_watch_and_conflict(http_store):
    """Typed clientset parity (reference client-go/**): CRUD + informer-
    style watch in InferenceService terms over the HTTP store, with
    optimistic concurrency through the dataclass resourceVersion."""
    import pytest as _pytest

    from fusioninfer_amd.controlplane.client import InferenceServiceClient
    from fusioninfer_amd.controlplane.fake import ConflictError as _Conflict

    store, raw = http_store
    c = InferenceServiceClient(raw)
    created = c.create(monolithic_svc("t1"))
    assert created.resource_version
    assert c.get("t1").name == "t1"
    assert [s.name for s in c.list()] == ["t1"]

    w, poll = c.watch_typed(send_initial=True)
    try:
        ev = poll(timeout=3.0)
        assert ev is not None and ev[0] == "ADDED" and ev[1].name == "t1"

        fresh = c.get("t1")
        stale = c.get("t1")
        fresh.roles[0].replicas = 2
        updated = c.update(fresh)
        assert updated.roles[0].replicas == 2
        with _pytest.raises(_Conflict):
            stale.roles[0].replicas = 5
            c.update(stale)
        ev = poll(timeout=3.0)
        assert ev is not None and ev[0] == "MODIFIED"
        assert ev[1].roles[0].replicas == 2

        c.delete("t1")
        assert c.try_get("t1") is None
        ev = poll(timeout=3.0)
        assert ev is not None and ev[0] == "DELETED"
    finally:
        c.stop_watch(w)
