"""OpenAI server tests (CPU, tiny model) via httpx ASGI transport."""

import asyncio
import json

import httpx
import pytest

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.models.registry import get_model_config
from fusioninfer_amd.server.api_server import build_app, decode_tokens, encode_prompt
from fusioninfer_amd.server.serving import ServingEngine


@pytest.fixture(scope="module")
def serving():
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=128),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=512, max_model_len=256
        ),
    )
    s = ServingEngine(cfg, device="cpu")
    yield s
    s.shutdown()


@pytest.fixture(scope="module")
def app(serving):
    return build_app(serving, "tiny-qwen3")


def _client(app):
    return httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://test"
    )


def test_tokenizer_roundtrip():
    ids = encode_prompt("hello world", 1024)
    assert decode_tokens(ids) == "hello world"
    assert encode_prompt([1, 2, 3], 1024) == [1, 2, 3]


def test_health_and_models(app):
    async def run():
        async with _client(app) as c:
            r = await c.get("/health")
            assert r.status_code == 200 and r.json()["status"] == "ok"
            r = await c.get("/v1/models")
            assert r.json()["data"][0]["id"] == "tiny-qwen3"

    asyncio.run(run())


def test_completions_token_ids(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7, 8] * 6, "max_tokens": 4,
                      "temperature": 0, "ignore_eos": True},
            )
            body = r.json()
            assert len(body["choices"][0]["token_ids"]) == 4
            assert body["usage"]["prompt_tokens"] == 24
            assert body["usage"]["completion_tokens"] == 4

    asyncio.run(run())


def test_completions_streaming(app):
    async def run():
        async with _client(app) as c:
            toks = []
            async with c.stream(
                "POST",
                "/v1/completions",
                json={"prompt": "hi there", "max_tokens": 3, "stream": True,
                      "temperature": 0, "ignore_eos": True},
            ) as r:
                async for line in r.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        chunk = json.loads(line[6:])
                        toks.extend(chunk["choices"][0]["token_ids"])
            assert len(toks) == 3

    asyncio.run(run())


def test_chat_completions(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/chat/completions",
                json={
                    "messages": [{"role": "user", "content": "hello"}],
                    "max_tokens": 3,
                    "temperature": 0,
                    "ignore_eos": True,
                },
            )
            body = r.json()
            assert body["choices"][0]["message"]["role"] == "assistant"
            assert body["usage"]["completion_tokens"] == 3

    asyncio.run(run())


def test_metrics_vllm_names(app, serving):
    async def run():
        async with _client(app) as c:
            r = await c.get("/metrics")
            text = r.text
            # the metric names the EPP scorers scrape
            assert "vllm:gpu_cache_usage_perc" in text
            assert "vllm:num_requests_waiting" in text
            assert "vllm:num_requests_running" in text

    asyncio.run(run())


def test_cli_config_surface():
    from fusioninfer_amd.server.__main__ import build_engine_config, parse_args

    args = parse_args(
        [
            "--model", "tiny-qwen3",
            "--tensor-parallel-size", "2",
            "--max-model-len", "1024",
            "--kv-transfer-config",
            '{"kv_connector": "RcclConnector", "kv_role": "kv_producer"}',
        ]
    )
    cfg = build_engine_config(args)
    assert cfg.parallel.tensor_parallel_size == 2
    assert cfg.scheduler.max_model_len == 1024
    assert cfg.kv_transfer.kv_connector == "RcclConnector"
    assert cfg.kv_transfer.kv_role == "kv_producer"


def test_too_long_prompt_returns_400(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [1] * 10000, "max_tokens": 1},
            )
            assert r.status_code == 400
            assert "max_model_len" in r.json()["error"]["message"]

    asyncio.run(run())


def test_stop_strings(app):
    """OpenAI `stop`: generation halts at the stop string and the text is
    truncated before it (byte-fallback tokenizer makes matching exact:
    with temperature 0 on the tiny model we learn which byte repeats by
    generating once, then stop on it)."""

    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "abcabc", "max_tokens": 8,
                      "temperature": 0, "ignore_eos": True},
            )
            full = r.json()["choices"][0]["text"]
            assert len(full) > 1
            stop_ch = full[1]
            r2 = await c.post(
                "/v1/completions",
                json={"prompt": "abcabc", "max_tokens": 8, "stop": stop_ch,
                      "temperature": 0, "ignore_eos": True},
            )
            body = r2.json()
            text = body["choices"][0]["text"]
            assert stop_ch not in text
            assert text == full.split(stop_ch)[0]

    asyncio.run(run())


def test_n_choices(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7] * 8, "max_tokens": 3, "n": 3,
                      "temperature": 0.8, "seed": 42, "ignore_eos": True},
            )
            body = r.json()
            assert len(body["choices"]) == 3
            assert [ch["index"] for ch in body["choices"]] == [0, 1, 2]
            assert all(len(ch["token_ids"]) == 3 for ch in body["choices"])
            assert body["usage"]["completion_tokens"] == 9
            # distinct seeds -> at least two distinct outputs (overwhelmingly)
            outs = {tuple(ch["token_ids"]) for ch in body["choices"]}
            assert len(outs) >= 2

    asyncio.run(run())


def test_stop_strings_streaming(app):
    """Streaming requests honor `stop`: the stream ends at the match and
    the concatenated deltas exclude the stop string."""

    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "xyzxyz", "max_tokens": 8,
                      "temperature": 0, "ignore_eos": True},
            )
            full = r.json()["choices"][0]["text"]
            stop_ch = full[1]
            text = ""
            n_chunks = 0
            async with c.stream("POST", "/v1/completions", json={
                "prompt": "xyzxyz", "max_tokens": 8, "stream": True,
                "stop": stop_ch, "temperature": 0, "ignore_eos": True,
            }) as resp:
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        chunk = json.loads(line[6:])
                        text += chunk["choices"][0]["text"]
                        n_chunks += 1
            assert stop_ch not in text
            assert text == full.split(stop_ch)[0]
            assert n_chunks <= len(full)

    asyncio.run(run())


def test_guided_choice_and_json_object(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "pick one:", "max_tokens": 16,
                      "temperature": 0.0,
                      "guided_choice": ["yes", "no", "maybe"]},
            )
            assert r.status_code == 200
            assert r.json()["choices"][0]["text"] in {"yes", "no", "maybe"}

            r = await c.post(
                "/v1/completions",
                json={"prompt": "as json:", "max_tokens": 48,
                      "temperature": 0.0,
                      "guided_json": {
                          "type": "object",
                          "properties": {"ok": {"type": "boolean"}},
                      }},
            )
            assert r.status_code == 200
            obj = json.loads(r.json()["choices"][0]["text"])
            assert isinstance(obj.get("ok"), bool)

            # optional properties: required subset must be present, any
            # extra emitted key must be a declared optional in order
            r = await c.post(
                "/v1/completions",
                json={"prompt": "as json:", "max_tokens": 64,
                      "temperature": 0.0,
                      "guided_json": {
                          "type": "object",
                          "properties": {
                              "name": {"enum": ["a", "b"]},
                              "age": {"type": "integer",
                                      "minimum": 0, "maximum": 9},
                              "ok": {"type": "boolean"},
                          },
                          "required": ["name", "ok"],
                      }},
            )
            assert r.status_code == 200
            obj = json.loads(r.json()["choices"][0]["text"])
            assert obj["name"] in {"a", "b"} and isinstance(obj["ok"], bool)
            assert set(obj) <= {"name", "age", "ok"}

            # OpenAI response_format on chat: output must stay a viable
            # JSON prefix even if max_tokens truncates it
            r = await c.post(
                "/v1/chat/completions",
                json={"messages": [{"role": "user", "content": "json pls"}],
                      "max_tokens": 12, "temperature": 0.0,
                      "response_format": {"type": "json_object"}},
            )
            assert r.status_code == 200
            from fusioninfer_amd.guided import JsonGrammar

            text = r.json()["choices"][0]["message"]["content"]
            g = JsonGrammar(root_object=True)
            st = g.initial()
            for ch in text:
                st = g.step(st, ch)
                assert st is not None, text

    asyncio.run(run())


def test_guided_regex_and_bad_pattern(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "id:", "max_tokens": 20, "temperature": 0.0,
                      "guided_regex": "[a-f]{4}-[0-9]{2}"},
            )
            assert r.status_code == 200
            import re as _re

            assert _re.fullmatch(r"[a-f]{4}-[0-9]{2}",
                                 r.json()["choices"][0]["text"])

            r = await c.post(
                "/v1/completions",
                json={"prompt": "x", "max_tokens": 4,
                      "guided_regex": "(unclosed"},
            )
            assert r.status_code == 400

    asyncio.run(run())


def test_logit_bias_forces_token(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7], "max_tokens": 3,
                      "temperature": 0.0,
                      "logit_bias": {"42": 100.0}},
            )
            assert r.status_code == 200
            # +100 on a random-init model's logits dominates every step
            assert r.json()["choices"][0]["token_ids"] == [42, 42, 42]

    asyncio.run(run())


def test_chat_stream_stop_strings(app):
    async def run():
        async with _client(app) as c:
            # unconstrained streamed reference (same per-token decode as
            # the stop-matching path), then stop on a substring of it
            async with c.stream(
                "POST", "/v1/chat/completions",
                json={"messages": [{"role": "user", "content": "hi"}],
                      "max_tokens": 12, "temperature": 0.0, "stream": True},
            ) as resp:
                full = ""
                async for line in resp.aiter_lines():
                    if not line.startswith("data: ") or line == "data: [DONE]":
                        continue
                    chunk = json.loads(line[6:])
                    full += chunk["choices"][0]["delta"].get("content", "")
            stop = full[3:6]
            assert stop
            async with c.stream(
                "POST", "/v1/chat/completions",
                json={"messages": [{"role": "user", "content": "hi"}],
                      "max_tokens": 12, "temperature": 0.0, "stream": True,
                      "stop": [stop]},
            ) as resp:
                text = ""
                finished = False
                async for line in resp.aiter_lines():
                    if not line.startswith("data: ") or line == "data: [DONE]":
                        continue
                    chunk = json.loads(line[6:])
                    ch = chunk["choices"][0]
                    text += ch["delta"].get("content", "")
                    finished = finished or ch["finish_reason"] == "stop"
            assert finished
            assert stop not in text
            assert text == full[: full.find(stop)]

    asyncio.run(run())


def test_echo_and_stream_usage(app):
    async def run():
        async with _client(app) as c:
            # echo: response text starts with the prompt
            r = await c.post(
                "/v1/completions",
                json={"prompt": "hello", "max_tokens": 4,
                      "temperature": 0.0, "echo": True},
            )
            assert r.json()["choices"][0]["text"].startswith("hello")

            # stream_options.include_usage: final chunk carries usage
            async with c.stream(
                "POST", "/v1/completions",
                json={"prompt": "hi", "max_tokens": 5, "temperature": 0.0,
                      "stream": True,
                      "stream_options": {"include_usage": True}},
            ) as resp:
                chunks = []
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and line != "data: [DONE]":
                        chunks.append(json.loads(line[6:]))
            assert chunks[-1]["usage"]["completion_tokens"] == 5
            assert chunks[-1]["choices"] == []

    asyncio.run(run())


def test_truncate_prompt_and_include_stop(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [7] * 100, "max_tokens": 2,
                      "truncate_prompt_tokens": 10, "temperature": 0.0,
                      "ignore_eos": True},
            )
            assert r.json()["usage"]["prompt_tokens"] == 10

            # include_stop_str_in_output keeps the matched stop string
            r = await c.post(
                "/v1/completions",
                json={"prompt": "mnomno", "max_tokens": 8,
                      "temperature": 0, "ignore_eos": True},
            )
            full = r.json()["choices"][0]["text"]
            stop_ch = full[1]
            r2 = await c.post(
                "/v1/completions",
                json={"prompt": "mnomno", "max_tokens": 8, "stop": stop_ch,
                      "include_stop_str_in_output": True,
                      "temperature": 0, "ignore_eos": True},
            )
            text = r2.json()["choices"][0]["text"]
            assert text.endswith(stop_ch)
            assert text == full.split(stop_ch)[0] + stop_ch

    asyncio.run(run())


def test_prompt_logprobs_api(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7, 8] * 4, "max_tokens": 2,
                      "temperature": 0.0, "prompt_logprobs": 1,
                      "logprobs": 1, "ignore_eos": True},
            )
            body = r.json()
            ch = body["choices"][0]
            assert len(ch["prompt_logprobs"]) == 16
            assert ch["prompt_logprobs"][0] is None
            assert all(isinstance(e[0], float)
                       for e in ch["prompt_logprobs"][1:])
            assert len(ch["logprobs"]) == 2  # one entry per output token

    asyncio.run(run())


def test_dynamic_lora_endpoints(app, serving, tmp_path):
    from tests.test_lora import _write_peft_adapter

    _write_peft_adapter(tmp_path / "ad", serving.engine.cfg.model)

    async def run():
        async with _client(app) as c:
            r = await c.post("/v1/load_lora_adapter",
                             json={"lora_name": "dyn",
                                   "lora_path": str(tmp_path / "ad")})
            assert r.status_code == 200, r.text
            r = await c.get("/v1/models")
            assert any(m["id"] == "dyn" for m in r.json()["data"])
            # request routed to the adapter by model name
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7] * 4, "max_tokens": 2,
                      "model": "dyn", "temperature": 0.0,
                      "ignore_eos": True},
            )
            assert r.status_code == 200
            r = await c.post("/v1/unload_lora_adapter",
                             json={"lora_name": "dyn"})
            assert r.status_code == 200
            r = await c.post("/v1/unload_lora_adapter",
                             json={"lora_name": "dyn"})
            assert r.status_code == 404

    asyncio.run(run())


def test_bad_sampling_params_400(app):
    async def run():
        async with _client(app) as c:
            for bad in (
                {"max_tokens": 0},
                {"temperature": -1},
                {"top_p": 0.0},
                {"top_p": 1.5},
                {"top_k": -2},
            ):
                r = await c.post("/v1/completions",
                                 json={"prompt": "x", **bad})
                assert r.status_code == 400, bad

    asyncio.run(run())


def test_chat_n_choices(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/chat/completions",
                json={"messages": [{"role": "user", "content": "pick"}],
                      "max_tokens": 3, "n": 3, "temperature": 0.8,
                      "seed": 7, "ignore_eos": True},
            )
            body = r.json()
            assert [ch["index"] for ch in body["choices"]] == [0, 1, 2]
            assert body["usage"]["completion_tokens"] == 9

    asyncio.run(run())


def test_streaming_n_choices(app):
    async def run():
        async with _client(app) as c:
            per_choice = {}
            finish = {}
            async with c.stream(
                "POST", "/v1/completions",
                json={"prompt": [5, 6, 7] * 6, "max_tokens": 3, "n": 2,
                      "temperature": 0.8, "seed": 3, "stream": True,
                      "ignore_eos": True},
            ) as resp:
                async for line in resp.aiter_lines():
                    if not line.startswith("data: ") or line == "data: [DONE]":
                        continue
                    ch = json.loads(line[6:])["choices"][0]
                    per_choice.setdefault(ch["index"], []).extend(
                        ch["token_ids"]
                    )
                    if ch["finish_reason"] in ("stop", "length"):
                        finish[ch["index"]] = True
            assert set(per_choice) == {0, 1}
            assert all(len(v) == 3 for v in per_choice.values())
            assert finish == {0: True, 1: True}

    asyncio.run(run())


def test_chat_streaming_n_choices(app):
    async def run():
        async with _client(app) as c:
            text = {}
            roles = {}
            async with c.stream(
                "POST", "/v1/chat/completions",
                json={"messages": [{"role": "user", "content": "go"}],
                      "max_tokens": 3, "n": 2, "temperature": 0.8,
                      "seed": 5, "stream": True, "ignore_eos": True},
            ) as resp:
                async for line in resp.aiter_lines():
                    if not line.startswith("data: ") or line == "data: [DONE]":
                        continue
                    ch = json.loads(line[6:])["choices"][0]
                    d = ch["delta"]
                    if "role" in d:
                        roles[ch["index"]] = d["role"]
                    text[ch["index"]] = text.get(ch["index"], "") + \
                        d.get("content", "")
            assert set(roles) == {0, 1}
            assert set(text) == {0, 1}

    asyncio.run(run())


def test_stop_strings_streaming_keep(app):
    """include_stop_str_in_output applies to streams too: the
    concatenated deltas END WITH the stop string (same request param,
    same behavior as the non-stream path — ADVICE round-1)."""

    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "xyzxyz", "max_tokens": 8,
                      "temperature": 0, "ignore_eos": True},
            )
            full = r.json()["choices"][0]["text"]
            stop_ch = full[1]
            text = ""
            async with c.stream("POST", "/v1/completions", json={
                "prompt": "xyzxyz", "max_tokens": 8, "stream": True,
                "stop": stop_ch, "temperature": 0, "ignore_eos": True,
                "include_stop_str_in_output": True,
            }) as resp:
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        chunk = json.loads(line[6:])
                        text += chunk["choices"][0]["text"]
            assert text == full.split(stop_ch)[0] + stop_ch

            # n>1 interleaved stream honors it per choice
            texts = {0: "", 1: ""}
            async with c.stream("POST", "/v1/completions", json={
                "prompt": "xyzxyz", "max_tokens": 8, "stream": True, "n": 2,
                "stop": stop_ch, "temperature": 0, "ignore_eos": True,
                "include_stop_str_in_output": True,
            }) as resp:
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        chunk = json.loads(line[6:])
                        ch = chunk["choices"][0]
                        texts[ch["index"]] += ch["text"]
            for v in texts.values():
                assert v.endswith(stop_ch)
                assert stop_ch not in v[:-1]

    asyncio.run(run())


def test_streaming_usage_not_inflated(app):
    """Multi-choice stream usage counts the shared prompt ONCE
    (OpenAI/vLLM semantics — ADVICE round-1)."""

    async def run():
        async with _client(app) as c:
            usage = None
            n_toks = 0
            async with c.stream("POST", "/v1/completions", json={
                "prompt": "hello", "max_tokens": 4, "stream": True, "n": 2,
                "temperature": 0, "ignore_eos": True,
                "stream_options": {"include_usage": True},
            }) as resp:
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        chunk = json.loads(line[6:])
                        if chunk.get("usage"):
                            usage = chunk["usage"]
                        for ch in chunk.get("choices", []):
                            n_toks += len(ch.get("token_ids", []))
            assert usage is not None
            assert usage["prompt_tokens"] == 5  # len("hello") bytes, once
            assert usage["completion_tokens"] == n_toks == 8

    asyncio.run(run())


def test_finish_reason_length_vs_stop(app):
    """OpenAI semantics: max_tokens exhaustion -> "length"; stop-string
    or stop-token -> "stop" (both transports)."""

    async def run():
        async with _client(app) as c:
            r = await c.post("/v1/completions", json={
                "prompt": "abcabc", "max_tokens": 4, "temperature": 0,
                "ignore_eos": True,
            })
            assert r.json()["choices"][0]["finish_reason"] == "length"

            full = (await c.post("/v1/completions", json={
                "prompt": "abcabc", "max_tokens": 8, "temperature": 0,
                "ignore_eos": True,
            })).json()["choices"][0]["text"]
            stop_ch = full[1]
            r = await c.post("/v1/completions", json={
                "prompt": "abcabc", "max_tokens": 8, "temperature": 0,
                "ignore_eos": True, "stop": stop_ch,
            })
            assert r.json()["choices"][0]["finish_reason"] == "stop"

            # streaming: the final chunk carries "length"
            reasons = []
            async with c.stream("POST", "/v1/completions", json={
                "prompt": "xy", "max_tokens": 3, "stream": True,
                "temperature": 0, "ignore_eos": True,
            }) as resp:
                async for line in resp.aiter_lines():
                    if line.startswith("data: ") and "[DONE]" not in line:
                        fr = json.loads(line[6:])["choices"][0][
                            "finish_reason"]
                        if fr:
                            reasons.append(fr)
            assert reasons == ["length"]

    asyncio.run(run())


def test_guided_grammar(app):
    """vLLM guided_grammar (GBNF): output constrained to the grammar."""

    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "pick:", "max_tokens": 16,
                      "temperature": 0.0,
                      "guided_grammar":
                          'root ::= "left" | "right" | "straight"'},
            )
            assert r.status_code == 200
            assert r.json()["choices"][0]["text"] in {
                "left", "right", "straight"}

            r = await c.post(  # malformed grammar -> 400
                "/v1/completions",
                json={"prompt": "x", "max_tokens": 4,
                      "guided_grammar": 'root ::= root "x"'},
            )
            assert r.status_code == 400

    asyncio.run(run())


def test_embeddings_endpoint(app):
    """OpenAI /v1/embeddings: unit-norm pooled vectors, deterministic
    per input, distinct across inputs; token-id input and length
    validation."""
    import math as _math

    async def run():
        async with _client(app) as c:
            r = await c.post("/v1/embeddings",
                             json={"input": ["hello world", "other text"]})
            assert r.status_code == 200
            out = r.json()
            assert out["object"] == "list" and len(out["data"]) == 2
            v0 = out["data"][0]["embedding"]
            v1 = out["data"][1]["embedding"]
            assert abs(_math.fsum(x * x for x in v0) - 1.0) < 1e-3
            assert v0 != v1
            assert out["usage"]["prompt_tokens"] > 0

            r2 = await c.post("/v1/embeddings",
                              json={"input": "hello world"})
            assert r2.json()["data"][0]["embedding"] == v0  # deterministic

            r3 = await c.post("/v1/embeddings", json={"input": [5, 6, 7]})
            assert r3.status_code == 200
            r4 = await c.post("/v1/embeddings",
                              json={"input": list(range(500))})
            assert r4.status_code == 400

    asyncio.run(run())


def test_health_watchdog_detects_hung_step(app, serving):
    """A step stuck past step_timeout_s flips /health to 503 (k8s
    liveness -> pod restart is the recovery path)."""
    import time as _time

    async def run():
        async with _client(app) as c:
            assert (await c.get("/health")).status_code == 200
            old = serving.step_timeout_s
            serving.step_timeout_s = 0.01
            serving._step_started = _time.monotonic() - 1.0  # fake a hang
            try:
                r = await c.get("/health")
                assert r.status_code == 503
                assert "exceeded" in r.json()["error"]
            finally:
                # one-way latch: restore for other tests in this module
                serving.step_timeout_s = old
                serving.healthy = True
                serving.last_error = ""
                serving._step_started = None

    asyncio.run(run())


def test_api_key_auth(serving):
    """--api-key: bearer auth on API endpoints; /health and /metrics
    stay open for the kubelet/Prometheus/EPP scrapers."""

    async def run():
        app2 = build_app(serving, "tiny-qwen3", api_key="sk-test")
        async with _client(app2) as c:
            assert (await c.get("/health")).status_code == 200
            assert (await c.get("/metrics")).status_code == 200
            r = await c.post("/v1/completions",
                             json={"prompt": "x", "max_tokens": 1})
            assert r.status_code == 401
            r = await c.post(
                "/v1/completions",
                json={"prompt": "x", "max_tokens": 1, "temperature": 0,
                      "ignore_eos": True},
                headers={"Authorization": "Bearer sk-test"})
            assert r.status_code == 200
            r = await c.get("/v1/models",
                            headers={"Authorization": "Bearer wrong"})
            assert r.status_code == 401

    asyncio.run(run())


def test_moe_model_serves_http():
    """MoE model end-to-end through the HTTP surface (CPU bmm expert
    path): completions + embeddings against tiny-qwen3-moe."""

    async def run():
        cfg = EngineConfig(
            model=get_model_config("tiny-qwen3-moe"),
            cache=CacheConfig(num_gpu_blocks=128),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )
        s = ServingEngine(cfg, device="cpu")
        try:
            app2 = build_app(s, "tiny-qwen3-moe")
            async with _client(app2) as c:
                r = await c.post(
                    "/v1/completions",
                    json={"prompt": [7, 8, 9] * 8, "max_tokens": 4,
                          "temperature": 0, "ignore_eos": True},
                )
                assert r.status_code == 200
                assert len(r.json()["choices"][0]["token_ids"]) == 4
                r = await c.post("/v1/embeddings", json={"input": [3, 4, 5]})
                assert r.status_code == 200
        finally:
            s.shutdown()

    asyncio.run(run())


def test_out_of_vocab_logit_bias_is_safe(app):
    """logit_bias keys outside the vocab must not crash the engine loop
    (device-side index fault); they are dropped."""

    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [5, 6, 7] * 4, "max_tokens": 3,
                      "temperature": 0, "ignore_eos": True,
                      "logit_bias": {"999999999": 50, "7": 10}},
            )
            assert r.status_code == 200
            assert len(r.json()["choices"][0]["token_ids"]) == 3
            # engine stayed healthy
            assert (await c.get("/health")).status_code == 200

    asyncio.run(run())


def test_nan_sampling_params_rejected(app):
    """JSON NaN literals in sampling params must 400, not NaN the
    engine's multinomial."""

    async def run():
        async with _client(app) as c:
            for payload in (
                '{"prompt": "x", "max_tokens": 2, "temperature": NaN}',
                '{"prompt": "x", "max_tokens": 2, "temperature": 0.5, '
                '"presence_penalty": NaN}',
                '{"prompt": "x", "max_tokens": 2, "top_p": NaN}',
            ):
                r = await c.post(
                    "/v1/completions", content=payload,
                    headers={"Content-Type": "application/json"})
                assert r.status_code == 400, payload
            assert (await c.get("/health")).status_code == 200

    asyncio.run(run())


def test_excessive_n_rejected(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": "x", "max_tokens": 1, "n": 100000},
            )
            assert r.status_code == 400

    asyncio.run(run())


def test_malformed_prompt_is_400(app):
    async def run():
        async with _client(app) as c:
            r = await c.post(
                "/v1/completions",
                json={"prompt": [["nested"]], "max_tokens": 1},
            )
            assert r.status_code == 400
            r = await c.post(
                "/v1/completions",
                json={"prompt": {"not": "a prompt"}, "max_tokens": 1},
            )
            assert r.status_code == 400

    asyncio.run(run())


def test_malformed_chat_messages_400(app):
    async def run():
        async with _client(app) as c:
            for messages in ("not-a-list", [{"role": "user"}, 42]):
                r = await c.post(
                    "/v1/chat/completions",
                    json={"messages": messages, "max_tokens": 1},
                )
                assert r.status_code == 400, messages
            # lenient cases must not 500 either
            r = await c.post(
                "/v1/chat/completions",
                json={"messages": [{"content": 3.14}], "max_tokens": 1,
                      "temperature": 0, "ignore_eos": True},
            )
            assert r.status_code in (200, 400)

    asyncio.run(run())
