"""Live control-plane loop tests (VERDICT round-1 item 1).

The reference's envtest scenarios
(pkg/controller/inferenceservice_controller_test.go:85-333 — create,
replica scale-up/down, image change propagates, metadata-only no-op with
stable resourceVersion) run here as LIVE-LOOP tests: a watch-driven
Manager reconciles against the in-memory apiserver while the test only
touches the apiserver — plus the parts envtest could not cover: child
status events driving readiness aggregation (via the stub LWS
controller), level-triggered child re-creation, conflict retry,
owner-reference GC, kill-and-resume, leader election, and the
probe/metrics endpoints the reference's e2e tier asserts
(test/e2e/e2e_test.go:176-261).
"""

import threading
import time
import urllib.request

import pytest

from fusioninfer_amd.controlplane import api
from fusioninfer_amd.controlplane.fake import (
    ConflictError,
    FakeClient,
)
from fusioninfer_amd.controlplane.manager import Manager
from fusioninfer_amd.controlplane.stubs import StubLWSController

from tests.test_controlplane import monolithic_svc, pod_template


@pytest.fixture()
def cluster():
    """store + manager (+ stub LWS controller) — stopped after the test."""
    client = FakeClient()
    mgr = Manager(client).start()
    stub = StubLWSController(client).start()
    yield client, mgr
    stub.stop()
    mgr.stop()


def _eventually(fn, timeout=5.0, interval=0.02):
    deadline = time.monotonic() + timeout
    last = None
    while time.monotonic() < deadline:
        last = fn()
        if last:
            return last
        time.sleep(interval)
    raise AssertionError(f"condition not met within {timeout}s: {last!r}")


def _get_cond(svc_obj, cond_type):
    for c in svc_obj.get("status", {}).get("conditions", []):
        if c["type"] == cond_type:
            return c
    return None


def test_live_create_reconciles_to_active(cluster):
    """envtest scenario 1 (create -> LWS exists) plus the aggregation
    envtest couldn't test: the stub LWS controller reports readiness and
    the CHILD EVENT requeues the service to Active."""
    client, mgr = cluster
    client.create(monolithic_svc("m1").to_dict())
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m1-worker-0"))
    svc = _eventually(
        lambda: (lambda o: o if _get_cond(o, "Active")
                 and _get_cond(o, "Active")["status"] == "True" else None)(
            client.get("InferenceService", "m1"))
    )
    comp = svc["status"]["components"]["worker"]
    assert comp["phase"] == "Running"
    assert comp["readyReplicas"] == 1
    assert _get_cond(svc, "Initialized")["status"] == "True"


def test_live_replica_scale_up_down(cluster):
    """envtest scenario 2: replicas 1->3 creates per-replica LWS; 3->1
    deletes the orphans."""
    client, mgr = cluster
    client.create(monolithic_svc("m2", replicas=1).to_dict())
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m2-worker-0"))

    obj = client.get("InferenceService", "m2")
    obj["spec"]["roles"][0]["replicas"] = 3
    client.update(obj)
    _eventually(lambda: len(client.list("LeaderWorkerSet")) == 3)

    obj = client.get("InferenceService", "m2")
    obj["spec"]["roles"][0]["replicas"] = 1
    client.update(obj)
    _eventually(lambda: len(client.list("LeaderWorkerSet")) == 1)
    assert client.try_get("LeaderWorkerSet", "m2-worker-0") is not None


def test_live_image_change_propagates(cluster):
    """envtest scenario 3: image change -> spec hash changes -> LWS
    updated with the new image."""
    client, mgr = cluster
    client.create(monolithic_svc("m3").to_dict())
    lws1 = _eventually(lambda: client.try_get("LeaderWorkerSet", "m3-worker-0"))

    obj = client.get("InferenceService", "m3")
    obj["spec"]["roles"][0]["template"] = pod_template(image="new:v2")
    client.update(obj)

    def updated():
        lws = client.try_get("LeaderWorkerSet", "m3-worker-0")
        if lws is None:
            return None
        img = lws["spec"]["leaderWorkerTemplate"]["leaderTemplate"]["spec"][
            "containers"][0]["image"]
        return lws if img == "new:v2" else None

    lws2 = _eventually(updated)
    assert lws2["metadata"]["resourceVersion"] != lws1["metadata"]["resourceVersion"]
    assert (lws2["metadata"]["labels"]["fusioninfer.io/spec-hash"]
            != lws1["metadata"]["labels"]["fusioninfer.io/spec-hash"])


def test_live_metadata_only_change_is_noop(cluster):
    """envtest scenario 4: metadata-only change -> NO LWS write (stable
    resourceVersion)."""
    client, mgr = cluster
    client.create(monolithic_svc("m4").to_dict())
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m4-worker-0"))
    mgr.wait_idle()
    rv_before = client.get("LeaderWorkerSet", "m4-worker-0")["metadata"][
        "resourceVersion"]

    obj = client.get("InferenceService", "m4")
    obj["metadata"].setdefault("labels", {})["team"] = "serving"
    client.update(obj)
    mgr.wait_idle()
    time.sleep(0.1)
    mgr.wait_idle()
    assert client.get("LeaderWorkerSet", "m4-worker-0")["metadata"][
        "resourceVersion"] == rv_before


def test_live_deleted_child_is_recreated(cluster):
    """Level-triggered ownership: deleting an owned LWS re-creates it from
    the child DELETED event alone (reference Owns() semantics)."""
    client, mgr = cluster
    client.create(monolithic_svc("m5").to_dict())
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m5-worker-0"))
    mgr.wait_idle()
    client.delete("LeaderWorkerSet", "m5-worker-0")
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m5-worker-0"))


def test_live_owner_gc_on_cr_delete(cluster):
    """Deleting the InferenceService garbage-collects every owned child."""
    client, mgr = cluster
    svc = pd_like_svc("m6")
    client.create(svc.to_dict())
    _eventually(lambda: client.try_get("Deployment", "m6-epp"))
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m6-prefiller-0"))
    mgr.wait_idle()
    client.delete("InferenceService", "m6")
    _eventually(lambda: not client.list("LeaderWorkerSet"))
    assert not client.list("Deployment")
    assert not client.list("PodGroup")
    assert not client.list("HTTPRoute")
    mgr.wait_idle()  # DELETED child events must not crash/recreate
    assert not client.list("LeaderWorkerSet")


def pd_like_svc(name):
    return api.InferenceService(
        name=name,
        uid=f"uid-{name}",
        roles=[
            api.Role(api.PREFILLER, 1, pod_template()),
            api.Role(api.DECODER, 1, pod_template()),
            api.Role(
                api.ROUTER, 1, routing_strategy=api.PD_DISAGGREGATION,
                httproute={"parentRefs": [{"name": "gw"}]},
            ),
        ],
    )


def test_live_concurrent_spec_updates_converge(cluster):
    """Optimistic-concurrency retry: spec updates hammered from another
    thread while the manager reconciles; the loop converges to the last
    spec with no lost children."""
    client, mgr = cluster
    client.create(monolithic_svc("m7").to_dict())

    def hammer():
        for i in range(10):
            for attempt in range(20):
                obj = client.get("InferenceService", "m7")
                obj["spec"]["roles"][0]["replicas"] = (i % 3) + 1
                try:
                    client.update(obj)
                    break
                except ConflictError:
                    time.sleep(0.002)
            time.sleep(0.005)

    t = threading.Thread(target=hammer)
    t.start()
    t.join()
    final = client.get("InferenceService", "m7")["spec"]["roles"][0]["replicas"]
    _eventually(lambda: len(client.list("LeaderWorkerSet")) == final)
    mgr.wait_idle()
    assert len(client.list("LeaderWorkerSet")) == final


def test_kill_and_resume():
    """Manager killed mid-flight; the store keeps state; a NEW manager
    resumes from list+watch bootstrap and converges the pending change."""
    client = FakeClient()
    mgr = Manager(client).start()
    stub = StubLWSController(client).start()
    try:
        client.create(monolithic_svc("m8").to_dict())
        _eventually(lambda: client.try_get("LeaderWorkerSet", "m8-worker-0"))
    finally:
        mgr.stop()
    # manager dead: scale up goes unreconciled
    obj = client.get("InferenceService", "m8")
    obj["spec"]["roles"][0]["replicas"] = 2
    client.update(obj)
    time.sleep(0.1)
    assert len(client.list("LeaderWorkerSet")) == 1

    mgr2 = Manager(client).start()
    try:
        _eventually(lambda: len(client.list("LeaderWorkerSet")) == 2)
    finally:
        mgr2.stop()
        stub.stop()


def test_leader_election_failover():
    """Two managers, one lease: only the leader reconciles; when it dies
    the follower takes over after lease expiry (reference
    cmd/main.go:84-86 leader-election semantics)."""
    client = FakeClient()
    m1 = Manager(client, leader_elect=True, identity="a",
                 lease_duration_s=0.4, renew_period_s=0.05).start()
    _eventually(lambda: m1.is_leader)
    m2 = Manager(client, leader_elect=True, identity="b",
                 lease_duration_s=0.4, renew_period_s=0.05).start()
    time.sleep(0.2)
    assert not m2.is_leader

    client.create(monolithic_svc("m9").to_dict())
    _eventually(lambda: client.try_get("LeaderWorkerSet", "m9-worker-0"))

    m1.stop()  # leader dies; lease expires; follower takes over
    _eventually(lambda: m2.is_leader, timeout=5.0)
    obj = client.get("InferenceService", "m9")
    obj["spec"]["roles"][0]["replicas"] = 2
    client.update(obj)
    try:
        _eventually(lambda: len(client.list("LeaderWorkerSet")) == 2)
    finally:
        m2.stop()


def test_probe_and_metrics_endpoints():
    """/healthz, /readyz and /metrics with controller-runtime metric names
    (what the reference's e2e asserts: controller_runtime_reconcile_total)."""
    client = FakeClient()
    mgr = Manager(client, probe_port=0, metrics_port=0).start()
    try:
        client.create(monolithic_svc("m10").to_dict())
        mgr.wait_idle()
        host, port = mgr.probe_addr
        for path in ("/healthz", "/readyz"):
            with urllib.request.urlopen(f"http://{host}:{port}{path}") as r:
                assert r.status == 200
        mh, mp = mgr.metrics_addr
        with urllib.request.urlopen(f"http://{mh}:{mp}/metrics") as r:
            body = r.read().decode()
        assert "controller_runtime_reconcile_total" in body
        assert 'result="success"' in body
        assert "workqueue_depth" in body
        assert "controller_runtime_reconcile_time_seconds_bucket" in body
        assert "controller_runtime_reconcile_time_seconds_count" in body
    finally:
        mgr.stop()


def test_store_conflict_semantics():
    """Stale resourceVersion on update/update_status raises ConflictError;
    no-op writes don't bump resourceVersion or emit watch events."""
    client = FakeClient()
    obj = client.create(
        {"kind": "ConfigMap", "metadata": {"name": "c"}, "data": {"a": "1"}}
    )
    w = client.watch(kinds=["ConfigMap"])
    stale = dict(obj)
    fresh = client.get("ConfigMap", "c")
    fresh["data"] = {"a": "2"}
    client.update(fresh)  # bumps rv
    stale["data"] = {"a": "3"}
    with pytest.raises(ConflictError):
        client.update(stale)
    cur = client.get("ConfigMap", "c")
    with pytest.raises(ConflictError):
        stale["status"] = {"x": 1}
        client.update_status(stale)
    # drain the one real event
    ev = w.poll(timeout=1.0)
    assert ev[0] == "MODIFIED" and ev[1]["data"] == {"a": "2"}
    # no-op write: same rv, no event
    rv = cur["metadata"]["resourceVersion"]
    client.update(dict(cur))
    assert client.get("ConfigMap", "c")["metadata"]["resourceVersion"] == rv
    assert w.poll(timeout=0.1) is None
    client.stop_watch(w)


def test_store_generation_bumps_on_spec_only():
    client = FakeClient()
    obj = client.create(monolithic_svc("g1").to_dict())
    assert obj["metadata"]["generation"] == 1
    obj["metadata"].setdefault("labels", {})["x"] = "y"
    obj = client.update(obj)
    assert obj["metadata"]["generation"] == 1  # metadata-only: no bump
    obj["spec"]["roles"][0]["replicas"] = 2
    obj = client.update(obj)
    assert obj["metadata"]["generation"] == 2
