"""Full-stack e2e: BASELINE config #1 without a cluster.

The reference's config #1 is "controller against kind, reconcile a
monolithic InferenceService with a CPU-only stub worker container".
Offline analog, one process short of kind: the live manager reconciles
the CR; the test plays kubelet by EXECUTING the container command the
reconciled LeaderWorkerSet actually rendered (the real engine server on
CPU, tiny model); readiness flows back through LWS status to the Active
condition; and a request is routed through the RECONCILED EPP config
via the Envoy ext-proc protocol to the running worker."""

import json
import subprocess
import sys
import urllib.request

import pytest

from fusioninfer_amd.controlplane import api
from fusioninfer_amd.controlplane.fake import FakeClient
from fusioninfer_amd.controlplane.manager import Manager
from fusioninfer_amd.epp.extproc import (
    DESTINATION_HEADER,
    ExtProcProcessor,
    build_request_body,
    build_request_headers,
    parse_processing_response,
)
from fusioninfer_amd.epp.picker import Endpoint, EndpointPicker

from tests.test_controlplane_live import _eventually, _get_cond


def _free_port():
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(240)
def test_config1_reconcile_exec_route_e2e():
    port = _free_port()
    svc = api.InferenceService(
        name="e2e1",
        roles=[
            api.Role(
                component_type=api.WORKER,
                replicas=1,
                template={"spec": {"containers": [{
                    "name": "engine",
                    "image": "fusioninfer-amd/engine:latest",
                    "command": ["python", "-m", "fusioninfer_amd.server"],
                    "args": ["--model", "tiny-qwen3",
                             "--host", "127.0.0.1",
                             "--port", str(port),
                             "--max-model-len", "256",
                             "--max-num-seqs", "8",
                             "--max-num-batched-tokens", "512"],
                    "resources": {"limits": {"amd.com/gpu": 1}},
                }]}},
            ),
            api.Role(component_type=api.ROUTER, replicas=1,
                     routing_strategy=api.PREFIX_CACHE),
        ],
    )

    client = FakeClient()
    mgr = Manager(client).start()
    proc = None
    try:
        client.create(svc.to_dict())
        lws = _eventually(
            lambda: client.try_get("LeaderWorkerSet", "e2e1-worker-0")
        )
        tmpl = lws["spec"]["leaderWorkerTemplate"]["leaderTemplate"]
        container = tmpl["spec"]["containers"][0]
        # kubelet stand-in: run the RENDERED command (image python -> ours)
        cmd = list(container["command"]) + list(container["args"])
        assert cmd[0] == "python"
        cmd[0] = sys.executable
        proc = subprocess.Popen(cmd, stdout=subprocess.DEVNULL,
                                stderr=subprocess.DEVNULL)

        def healthy():
            try:
                with urllib.request.urlopen(
                    f"http://127.0.0.1:{port}/health", timeout=2
                ) as r:
                    return r.status == 200 or None
            except Exception:
                return None

        _eventually(healthy, timeout=120.0, interval=0.5)

        # the data plane is up: report LWS readiness (kubelet/LWS
        # controller role) and watch the CR go Active via the manager
        client.set_lws_ready("e2e1-worker-0")
        svc_obj = _eventually(
            lambda: (lambda o: o if (_get_cond(o, "Active") or {}).get(
                "status") == "True" else None)(
                client.get("InferenceService", "e2e1"))
        )
        assert svc_obj["status"]["components"]["worker"]["phase"] == "Running"

        # route through the RECONCILED EPP config over ext-proc
        cm = client.get("ConfigMap", "e2e1-epp-config")
        picker = EndpointPicker(cm["data"]["config.yaml"])
        ep = Endpoint(f"127.0.0.1:{port}",
                      labels={"fusioninfer.io/component-type": "worker"})
        ext = ExtProcProcessor(picker, lambda: [ep])
        body = json.dumps({"prompt": "hello serving world",
                           "max_tokens": 6}).encode()
        resps = [
            parse_processing_response(r)
            for r in ext.process(iter([
                build_request_headers({":path": "/v1/completions"}),
                build_request_body(body),
            ]))
        ]
        dest = resps[1]["set_headers"][DESTINATION_HEADER]
        assert dest == f"127.0.0.1:{port}"

        # ...and the picked endpoint actually serves the request
        req = urllib.request.Request(
            f"http://{dest}/v1/completions", data=body,
            headers={"Content-Type": "application/json"},
        )
        with urllib.request.urlopen(req, timeout=60) as r:
            out = json.loads(r.read())
        assert r.status == 200
        assert len(out["choices"][0]["token_ids"]) == 6
    finally:
        if proc is not None:
            proc.kill()   # exact PID we spawned
            proc.wait(timeout=10)
        mgr.stop()
