"""Envoy ext-proc gRPC EPP: wire-level conversations drive the picker.

The reference's gateway path is Envoy -> ext-proc gRPC :9002 with gRPC
health on :9003 (pkg/router/epp.go:125-165). These tests run Envoy-format
ProcessingRequest/ProcessingResponse conversations (protobuf wire bytes)
against the processor, including a real grpcio server round-trip, the
prefix-cache pick, the PD prefill handshake, and the 503 immediate
response — VERDICT round-1 item 7's done-condition.
"""

import json

import pytest

from fusioninfer_amd.controlplane import api, router
from fusioninfer_amd.epp import Endpoint, EndpointPicker
from fusioninfer_amd.epp import pb
from fusioninfer_amd.epp.extproc import (
    DESTINATION_HEADER,
    ExtProcProcessor,
    build_request_body,
    build_request_headers,
    parse_processing_response,
    serve,
)
from tests.test_controlplane import monolithic_svc, pd_svc


def _cfg(strategy=None, svc=None):
    svc = svc or monolithic_svc()
    role = api.Role(api.ROUTER, routing_strategy=strategy)
    return router.generate_epp_config(svc, role)


def test_pb_roundtrip():
    for n in (0, 1, 127, 128, 300, 2**32, 2**63 - 1):
        data = pb.encode_varint(n)
        val, i = pb.decode_varint(data, 0)
        assert val == n and i == len(data)
    msg = pb.str_field(1, "key") + pb.varint_field(2, 99) + pb.len_field(3, b"\x00\x01")
    got = pb.fields(msg)
    assert got == [(1, 2, b"key"), (2, 0, 99), (3, 2, b"\x00\x01")]


def _conversation(proc, body: dict, headers=None):
    reqs = [
        build_request_headers(headers or {":path": "/v1/completions",
                                          ":method": "POST"}),
        build_request_body(json.dumps(body).encode()),
    ]
    return [parse_processing_response(r) for r in proc.process(iter(reqs))]


def test_extproc_routes_to_picked_endpoint():
    picker = EndpointPicker(_cfg(api.PREFIX_CACHE))
    eps = [Endpoint("10.0.0.1:8000"), Endpoint("10.0.0.2:8000")]
    proc = ExtProcProcessor(picker, lambda: eps)
    out = _conversation(proc, {"prompt": "hello world " * 30,
                               "max_tokens": 8})
    assert out[0]["phase"] == "request_headers"
    assert out[1]["phase"] == "request_body"
    dest = out[1]["set_headers"][DESTINATION_HEADER]
    assert dest in {e.address for e in eps}
    assert out[1]["clear_route_cache"]

    # prefix-cache affinity through the ext-proc surface: same prompt
    # sticks to the same endpoint
    out2 = _conversation(proc, {"prompt": "hello world " * 30 + "more"})
    assert out2[1]["set_headers"][DESTINATION_HEADER] == dest


def test_extproc_pd_prefill_handshake():
    svc = pd_svc()
    picker = EndpointPicker(
        router.generate_epp_config(svc, svc.router_roles()[0])
    )
    eps = [
        Endpoint("pre:8000", labels={"fusioninfer.io/component-type": "prefiller"}),
        Endpoint("dec:8000", labels={"fusioninfer.io/component-type": "decoder"}),
    ]
    calls = []

    def fake_prefill(address, token_ids):
        calls.append((address, len(token_ids)))
        return 42

    proc = ExtProcProcessor(picker, lambda: eps, prefill_call=fake_prefill)
    out = _conversation(proc, {"prompt": "abcdef" * 20})
    hdrs = out[1]["set_headers"]
    assert hdrs[DESTINATION_HEADER] == "dec:8000"
    assert hdrs["x-pd-tag"] == "42"
    assert calls and calls[0][0] == "pre:8000"


def test_extproc_no_endpoints_immediate_503():
    picker = EndpointPicker(_cfg(api.PREFIX_CACHE))
    proc = ExtProcProcessor(picker, lambda: [])
    out = _conversation(proc, {"prompt": "x"})
    assert out[1]["phase"] == "immediate"
    assert out[1]["immediate"]["code"] == 503


def test_extproc_bad_json_immediate_400():
    picker = EndpointPicker(_cfg(api.PREFIX_CACHE))
    proc = ExtProcProcessor(picker, lambda: [Endpoint("a:8000")])
    reqs = [build_request_body(b"{not json")]
    out = [parse_processing_response(r) for r in proc.process(iter(reqs))]
    assert out[0]["immediate"]["code"] == 400


def test_extproc_grpc_server_end_to_end():
    """Real grpcio round-trip on ephemeral ports: the Process stream and
    the :9003-style health service (reference probe contract)."""
    grpc = pytest.importorskip("grpc")
    picker = EndpointPicker(_cfg(api.PREFIX_CACHE))
    eps = [Endpoint("10.0.0.9:8000")]
    server, health_server, (host, port), (hh, hp) = serve(
        picker, lambda: eps, port=0, health_port=0
    )
    try:
        chan = grpc.insecure_channel(f"{host}:{port}")
        method = chan.stream_stream(
            "/envoy.service.ext_proc.v3.ExternalProcessor/Process",
            request_serializer=None, response_deserializer=None,
        )
        reqs = iter([
            build_request_headers({":path": "/v1/completions"}),
            build_request_body(json.dumps({"prompt": "hi there"}).encode()),
        ])
        resps = [parse_processing_response(r) for r in method(reqs)]
        assert resps[0]["phase"] == "request_headers"
        assert resps[1]["set_headers"][DESTINATION_HEADER] == "10.0.0.9:8000"
        chan.close()

        hchan = grpc.insecure_channel(f"{hh}:{hp}")
        check = hchan.unary_unary(
            "/grpc.health.v1.Health/Check",
            request_serializer=None, response_deserializer=None,
        )
        resp = check(b"")
        assert pb.fields(resp) == [(1, 0, 1)]  # status: SERVING
        hchan.close()
    finally:
        server.stop(0)
        health_server.stop(0)
