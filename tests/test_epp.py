"""First-party EPP picker tests against the control plane's rendered
EndpointPickerConfig YAMLs (one per routing strategy)."""

from fusioninfer_amd.controlplane import api, router
from fusioninfer_amd.epp import Endpoint, EndpointPicker
from fusioninfer_amd.epp.picker import PREFILL_HEADER
from tests.test_controlplane import monolithic_svc, pd_svc


def _cfg(strategy=None, svc=None):
    svc = svc or monolithic_svc()
    role = api.Role(api.ROUTER, routing_strategy=strategy)
    return router.generate_epp_config(svc, role)


def test_prefix_cache_affinity():
    picker = EndpointPicker(_cfg(api.PREFIX_CACHE))
    eps = [Endpoint("10.0.0.1:8000"), Endpoint("10.0.0.2:8000")]
    prompt = list(range(100))
    # first pick: no cache anywhere; records blocks on the chosen server
    r1 = picker.pick({"prompt_token_ids": prompt}, eps)
    assert r1.endpoint is not None
    # same prefix again: must stick to the same server
    r2 = picker.pick({"prompt_token_ids": prompt + [7, 8]}, eps)
    assert r2.endpoint.address == r1.endpoint.address
    # unrelated prompt with a busier chosen server: no affinity constraint
    r3 = picker.pick({"prompt_token_ids": [999] * 50}, eps)
    assert r3.endpoint is not None


def test_kv_utilization_prefers_free_server():
    picker = EndpointPicker(_cfg(api.KV_CACHE_UTILIZATION))
    eps = [
        Endpoint("a:8000", kv_cache_usage=0.9),
        Endpoint("b:8000", kv_cache_usage=0.1),
    ]
    r = picker.pick({"prompt_token_ids": [1, 2, 3]}, eps)
    assert r.endpoint.address == "b:8000"


def test_queue_scorer_prefers_short_queue():
    picker = EndpointPicker(_cfg(api.QUEUE_SIZE))
    eps = [
        Endpoint("a:8000", queue_depth=10),
        Endpoint("b:8000", queue_depth=0),
    ]
    r = picker.pick({"prompt_token_ids": [1]}, eps)
    assert r.endpoint.address == "b:8000"


def test_lora_affinity():
    picker = EndpointPicker(_cfg(api.LORA_AFFINITY))
    eps = [
        Endpoint("a:8000", active_loras=("sql-adapter",)),
        Endpoint("b:8000"),
    ]
    r = picker.pick({"prompt_token_ids": [1], "lora": "sql-adapter"}, eps)
    assert r.endpoint.address == "a:8000"


def test_pd_two_profile_routing():
    svc = pd_svc()
    picker = EndpointPicker(
        router.generate_epp_config(svc, svc.router_roles()[0])
    )
    assert picker.is_pd
    eps = [
        Endpoint("p:8000", labels={"fusioninfer.io/component-type": "prefiller"}),
        Endpoint("d:8000", labels={"fusioninfer.io/component-type": "decoder"}),
    ]
    r = picker.pick({"prompt_token_ids": list(range(30))}, eps)
    assert r.prefill_endpoint.address == "p:8000"
    assert r.endpoint.address == "d:8000"
    assert r.headers[PREFILL_HEADER] == "p:8000"


def test_pd_filters_respect_labels():
    svc = pd_svc()
    picker = EndpointPicker(
        router.generate_epp_config(svc, svc.router_roles()[0])
    )
    # only decoders available: prefill profile yields nothing
    eps = [Endpoint("d:8000", labels={"fusioninfer.io/component-type": "decoder"})]
    r = picker.pick({"prompt_token_ids": [1, 2]}, eps)
    assert r.prefill_endpoint is None
    assert r.endpoint.address == "d:8000"
    assert PREFILL_HEADER not in r.headers
