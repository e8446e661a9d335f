"""OpenAI-compatible HTTP server on :8000.

Capability parity with the engine surface the reference's InferencePool /
HTTPRoute target (`vllm serve` on port 8000; targetPort
pinned at reference pkg/router/inferencepool.go:31-32): /v1/completions,
/v1/chat/completions (streaming + non-streaming), /health, /v1/models,
and Prometheus /metrics with the vLLM-compatible gauge names the EPP
scorers scrape (vllm:gpu_cache_usage_perc, vllm:num_requests_waiting —
reference pkg/router/strategy.go:70-98 consumes these).

Tokenization: vendored tokenizer.json files load through the offline
`tokenizers` wheel (--tokenizer / checkpoint-dir autodetect); without one,
string prompts use a reversible byte-level fallback. OpenAI's
prompt-as-token-ids form is first-class either way and is what the
benchmark and EPP flows use.
"""

from __future__ import annotations

import asyncio
import json
import time
import uuid
from typing import Any, AsyncGenerator, Dict, List, Optional, Union

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse

from fusioninfer_amd.engine.sequence import SamplingParams
from fusioninfer_amd.server.serving import PDRejectedError, ServingEngine

# ----------------------------------------------------------- tokenization


def encode_prompt(prompt: Union[str, List[int]], vocab_size: int) -> List[int]:
    if isinstance(prompt, list):
        return [int(t) for t in prompt]
    # byte-level fallback: offset so ids stay in-vocab and reversible
    return [min(b + 3, vocab_size - 1) for b in prompt.encode("utf-8")]


def decode_tokens(token_ids: List[int]) -> str:
    try:
        return bytes(max(t - 3, 0) & 0xFF for t in token_ids).decode(
            "utf-8", errors="replace"
        )
    except Exception:
        return " ".join(str(t) for t in token_ids)


def _find_stop(text: str, stops: List[str]):
    """Earliest stop-string match (longest on index tie) or None."""
    best = None
    for s in stops:
        i = text.find(s)
        if i >= 0 and (best is None or i < best[0]
                       or (i == best[0] and len(s) > len(best[1]))):
            best = (i, s)
    return best


class StopStreamFilter:
    """Streaming stop-string matcher: holds back the longest possible
    stop prefix so a match that spans chunk boundaries is never partially
    emitted. With keep=True (include_stop_str_in_output) the matched stop
    string itself is emitted, same as the non-stream path. push() returns
    (text safe to emit now, matched?)."""

    def __init__(self, stops: List[str], keep: bool = False):
        self.stops = stops
        self.keep = keep
        self.hold = max((len(s) for s in stops), default=1) - 1
        self.text = ""
        self.sent = 0

    def push(self, piece: str):
        self.text += piece
        best = _find_stop(self.text, self.stops)
        if best is not None:
            end = best[0] + len(best[1]) if self.keep else best[0]
            emit = self.text[self.sent: end]
            self.sent = end
            return emit, True
        safe = max(len(self.text) - self.hold, self.sent)
        emit = self.text[self.sent: safe]
        self.sent = safe
        return emit, False

    def flush(self) -> str:
        """Emit the held-back tail (stream ended without a match)."""
        emit = self.text[self.sent:]
        self.sent = len(self.text)
        return emit


def build_app(serving: ServingEngine, model_name: str,
              tokenizer=None, api_key: Optional[str] = None) -> FastAPI:
    app = FastAPI(title="fusioninfer-amd")
    vocab = serving.engine.cfg.model.vocab_size
    if tokenizer is None:
        from fusioninfer_amd.tokenizer import ByteTokenizer

        tokenizer = ByteTokenizer(vocab)

    if api_key:
        # vLLM --api-key parity: bearer auth on everything except the
        # probes/metrics the platform (kubelet, Prometheus, EPP) scrapes
        @app.middleware("http")
        async def _auth(request: Request, call_next):
            path = request.url.path
            # exempt: platform scrapers (kubelet probes, Prometheus,
            # EPP metrics) and the cluster-internal PD handshake the
            # EPP issues without client credentials
            if (path not in ("/health", "/metrics")
                    and not path.startswith("/pd/")):
                if request.headers.get(
                        "Authorization") != f"Bearer {api_key}":
                    return JSONResponse(
                        {"error": {"message": "invalid API key",
                                   "type": "authentication_error"}}, 401)
            return await call_next(request)

    def _encode(prompt: Union[str, List[int]]) -> List[int]:
        try:
            if isinstance(prompt, list):
                return [int(t) for t in prompt]
            if not isinstance(prompt, str):
                raise TypeError(f"prompt must be a string or token-id "
                                f"list, got {type(prompt).__name__}")
            return tokenizer.encode(prompt)
        except (TypeError, ValueError, AttributeError) as e:
            raise ValueError(f"malformed prompt: {e}") from e

    @app.exception_handler(ValueError)
    async def _bad_request(request: Request, exc: ValueError):
        # malformed client input surfacing anywhere in a handler is a
        # 400, never a 500 stack trace
        return JSONResponse(
            {"error": {"message": str(exc),
                       "type": "invalid_request_error"}}, 400)

    @app.get("/health")
    async def health():
        # check_health also trips on a HUNG engine step (watchdog) so a
        # wedged collective fails the k8s liveness probe -> pod restart
        if not serving.check_health():
            return JSONResponse(
                {"status": "unhealthy", "error": serving.last_error}, 503
            )
        return {"status": "ok"}

    @app.get("/version")
    async def version():
        from fusioninfer_amd import __version__

        return {"version": __version__}

    @app.get("/v1/models")
    async def models():
        data = [
            {"id": model_name, "object": "model", "owned_by": "fusioninfer-amd"}
        ]
        for name in serving.engine.active_loras():
            data.append(
                {"id": name, "object": "model", "owned_by": "fusioninfer-amd",
                 "parent": model_name}
            )
        return {"object": "list", "data": data}

    @app.get("/metrics")
    async def metrics():
        m = serving.metrics()
        lines = []
        for name, mtype in [
            ("gpu_cache_usage_perc", "gauge"),
            ("num_requests_waiting", "gauge"),
            ("num_requests_running", "gauge"),
            ("generation_tokens_total", "counter"),
            ("prompt_tokens_total", "counter"),
            ("request_success_total", "counter"),
            ("time_to_first_token_seconds_sum", "counter"),
            ("time_to_first_token_seconds_count", "counter"),
            ("e2e_request_latency_seconds_sum", "counter"),
            ("e2e_request_latency_seconds_count", "counter"),
            ("engine_step_seconds_sum", "counter"),
            ("engine_step_seconds_count", "counter"),
            ("pipelined_decode_steps_total", "counter"),
            ("num_preemptions_total", "counter"),
            ("num_swap_outs_total", "counter"),
            ("spec_decode_num_draft_tokens_total", "counter"),
            ("spec_decode_num_accepted_tokens_total", "counter"),
            ("prefix_cache_queries_total", "counter"),
            ("prefix_cache_hits_total", "counter"),
        ]:
            lines.append(f"# TYPE vllm:{name} {mtype}")
            lines.append(
                f'vllm:{name}{{model_name="{model_name}"}} {m[name]}'
            )
        # vLLM-style histograms (cumulative buckets + +Inf)
        e = serving.engine
        for name, buckets, hist, total in [
            ("time_to_first_token_seconds", e.ttft_buckets, e.ttft_hist,
             e.ttft_sum),
            ("e2e_request_latency_seconds", e.e2e_buckets, e.e2e_hist,
             e.e2e_latency_sum),
            ("time_per_output_token_seconds", e.tpot_buckets, e.tpot_hist,
             0.0),
        ]:
            lines.append(f"# TYPE vllm:{name} histogram")
            cum = 0
            for b, c in zip(buckets, hist):
                cum += c
                lines.append(
                    f'vllm:{name}_bucket{{model_name="{model_name}",'
                    f'le="{b}"}} {cum}'
                )
            cum += hist[-1]
            lines.append(
                f'vllm:{name}_bucket{{model_name="{model_name}",'
                f'le="+Inf"}} {cum}'
            )
        return PlainTextResponse("\n".join(lines) + "\n")

    _vocab_cache: Dict[str, Any] = {}

    def _get_vocab():
        if "v" not in _vocab_cache:
            from fusioninfer_amd.guided import Vocabulary

            # byte-level decode keeps multi-byte UTF-8 representable in
            # grammar masks (decode_one collapses split chars to U+FFFD)
            decode = getattr(tokenizer, "decode_one_bytes",
                             tokenizer.decode_one)
            _vocab_cache["v"] = Vocabulary(vocab, decode)
        return _vocab_cache["v"]

    def _guided_from(body: Dict[str, Any]):
        """Structured outputs: OpenAI response_format plus the vLLM
        guided_json / guided_regex / guided_choice extensions. Raises
        ValueError for unsupported grammars (mapped to HTTP 400)."""
        from fusioninfer_amd.guided import build_guided

        if body.get("guided_regex"):
            return build_guided("regex", str(body["guided_regex"]),
                                _get_vocab())
        if body.get("guided_choice"):
            return build_guided("choice", list(body["guided_choice"]),
                                _get_vocab())
        if body.get("guided_json"):
            return build_guided("json_schema", body["guided_json"],
                                _get_vocab())
        if body.get("guided_grammar"):
            return build_guided("grammar", str(body["guided_grammar"]),
                                _get_vocab())
        rf = body.get("response_format")
        if isinstance(rf, dict):
            t = rf.get("type")
            if t == "json_object":
                return build_guided("json_object", None, _get_vocab())
            if t == "json_schema":
                schema = (rf.get("json_schema") or {}).get("schema")
                return build_guided("json_schema", schema, _get_vocab())
        return None

    def _sampling_from(body: Dict[str, Any]) -> SamplingParams:
        mt = int(body.get("max_tokens", 16))
        if mt < 1:
            raise ValueError("max_tokens must be >= 1")
        import math as _math

        temp = float(body.get("temperature", 1.0))
        # NaN compares False everywhere: an explicit finite check keeps
        # a JSON "NaN" literal from reaching torch.multinomial
        if not _math.isfinite(temp) or temp < 0:
            raise ValueError("temperature must be finite and >= 0")
        for pk in ("presence_penalty", "frequency_penalty",
                   "repetition_penalty"):
            v = body.get(pk)
            if v is not None and not _math.isfinite(float(v)):
                raise ValueError(f"{pk} must be finite")
        tp_ = float(body.get("top_p", 1.0))
        if not 0.0 < tp_ <= 1.0:
            raise ValueError("top_p must be in (0, 1]")
        if int(body.get("top_k", 0)) < 0:
            raise ValueError("top_k must be >= 0")
        if not 0.0 <= float(body.get("min_p", 0.0)) <= 1.0:
            raise ValueError("min_p must be in [0, 1]")
        stop_ids = list(body.get("stop_token_ids") or [])
        if tokenizer.eos_token_id is not None \
                and tokenizer.eos_token_id not in stop_ids:
            stop_ids.append(tokenizer.eos_token_id)
        return SamplingParams(
            max_tokens=int(body.get("max_tokens", 16)),
            min_tokens=int(body.get("min_tokens", 0)),
            temperature=float(body.get("temperature", 1.0)),
            top_p=float(body.get("top_p", 1.0)),
            top_k=int(body.get("top_k", 0)),
            min_p=float(body.get("min_p", 0.0)),
            repetition_penalty=float(body.get("repetition_penalty", 1.0)),
            presence_penalty=float(body.get("presence_penalty", 0.0)),
            frequency_penalty=float(body.get("frequency_penalty", 0.0)),
            seed=body.get("seed"),
            logprobs=body.get("logprobs"),
            prompt_logprobs=body.get("prompt_logprobs"),
            ignore_eos=bool(body.get("ignore_eos", False)),
            stop_token_ids=stop_ids,
            guided=_guided_from(body),
            logit_bias=body.get("logit_bias"),
        )

    def _parse_stops(body: Dict[str, Any]) -> List[str]:
        stops = body.get("stop")
        if stops is None:
            return []
        return [stops] if isinstance(stops, str) else list(stops)

    def _finish_reason(rid, by_stop_string):
        """OpenAI finish_reason for a finished stream: stop-string cuts
        are "stop"; otherwise the engine's reason ("length"/"stop")."""
        if by_stop_string:
            return "stop"
        fin = serving.peek_final(rid)
        return getattr(fin, "finish_reason", None) or "stop"

    async def _collect(q, stops=None, req_id=None, keep_stop=False):
        """Drain a request's token stream. With OpenAI `stop` strings the
        generated text is truncated BEFORE the first stop match (AFTER it
        with include_stop_str_in_output) and the engine request is
        aborted. Text accumulates through incremental detokenization so
        multi-byte characters split across BPE tokens decode correctly.
        Returns (token_ids, text, finish_reason)."""
        from fusioninfer_amd.tokenizer import IncrementalDetokenizer

        loop = asyncio.get_event_loop()
        toks: List[int] = []
        text = ""
        detok = IncrementalDetokenizer(tokenizer) if stops else None

        def engine_reason():
            fin = serving.peek_final(req_id)
            return getattr(fin, "finish_reason", None) or "stop"

        while True:
            tok, finished = await loop.run_in_executor(None, q.get)
            if tok is not None:
                toks.append(tok)
                if stops:
                    text += detok.push(tok)
                    hit = _find_stop(text, stops)
                    if hit is not None:
                        serving.abort(req_id)
                        end = hit[0] + len(hit[1]) if keep_stop else hit[0]
                        return toks, text[:end], "stop"
            if finished:
                if not stops:
                    return toks, tokenizer.decode(toks), engine_reason()
                text += detok.flush()
                hit = _find_stop(text, stops)
                if hit is not None:
                    end = hit[0] + len(hit[1]) if keep_stop else hit[0]
                    return toks, text[:end], "stop"
                return toks, text, engine_reason()

    async def _stream(q) -> AsyncGenerator:
        loop = asyncio.get_event_loop()
        while True:
            tok, finished = await loop.run_in_executor(None, q.get)
            yield tok, finished
            if finished:
                return

    @app.post("/v1/load_lora_adapter")
    async def load_lora(request: Request):
        body = await request.json()
        name = body.get("lora_name")
        path = body.get("lora_path")
        if not name or not path:
            return JSONResponse({"error": {
                "message": "lora_name and lora_path required",
                "type": "invalid_request_error"}}, 400)
        try:
            loop = asyncio.get_event_loop()
            await loop.run_in_executor(
                None, serving.engine.add_lora_from_path, name, path
            )
        except Exception as e:
            return JSONResponse({"error": {"message": repr(e),
                                           "type": "invalid_request_error"}},
                                400)
        return PlainTextResponse(f"Success: LoRA adapter '{name}' added")

    @app.post("/v1/unload_lora_adapter")
    async def unload_lora(request: Request):
        body = await request.json()
        name = body.get("lora_name")
        if not serving.engine.remove_lora(name or ""):
            return JSONResponse({"error": {
                "message": f"adapter {name!r} not found",
                "type": "invalid_request_error"}}, 404)
        return PlainTextResponse(f"Success: LoRA adapter '{name}' removed")

    @app.post("/tokenize")
    async def tokenize(request: Request):
        body = await request.json()
        ids = _encode(body.get("prompt", ""))
        return {
            "tokens": ids,
            "count": len(ids),
            "max_model_len": serving.engine.cfg.scheduler.max_model_len,
        }

    @app.post("/detokenize")
    async def detokenize(request: Request):
        body = await request.json()
        return {"prompt": tokenizer.decode(
            [int(t) for t in body.get("tokens", [])]
        )}

    @app.post("/v1/embeddings")
    async def embeddings(request: Request):
        """OpenAI embeddings: last-token pooled, L2-normalized hidden
        state of a decoder-LM prefill (vLLM embed-task semantics).
        input: str | [str] | [int] | [[int]]."""
        import asyncio

        body = await request.json()
        inp = body.get("input", "")
        if isinstance(inp, str):
            items = [inp]
        elif isinstance(inp, list) and inp and isinstance(inp[0], int):
            items = [inp]
        elif isinstance(inp, list):
            items = inp
        else:
            return JSONResponse(
                {"error": {"message": "invalid input",
                           "type": "invalid_request_error"}}, 400)
        max_len = serving.engine.cfg.scheduler.max_model_len
        pooling = body.get("pooling", "last")
        data = []
        n_prompt = 0
        for i, item in enumerate(items):
            ids = _encode(item) if isinstance(item, str) else \
                [int(t) for t in item]
            if not ids or len(ids) > max_len:
                return JSONResponse(
                    {"error": {"message": f"input {i}: length {len(ids)} "
                               f"not in [1, {max_len}]",
                               "type": "invalid_request_error"}}, 400)
            try:
                vec = await asyncio.to_thread(serving.embed, ids, pooling)
            except ValueError as e:
                return JSONResponse(
                    {"error": {"message": str(e),
                               "type": "invalid_request_error"}}, 400)
            n_prompt += len(ids)
            data.append({"object": "embedding", "index": i,
                         "embedding": vec})
        return {
            "object": "list",
            "data": data,
            "model": body.get("model") or model_name,
            "usage": {"prompt_tokens": n_prompt, "total_tokens": n_prompt},
        }

    @app.post("/pd/prefill")
    async def pd_prefill(request: Request):
        """PD producer: prefill the prompt, ship its KV to the decoder, and
        return the pd_tag the router forwards to the decode endpoint
        (x-pd-tag header) plus the prefiller-sampled first token."""
        if serving.kv_connector is None:
            return JSONResponse(
                {"error": {"message": "server is not a PD prefiller",
                           "type": "invalid_request_error"}}, 400)
        body = await request.json()
        prompt_ids = _encode(body.get("prompt", ""))
        loop = asyncio.get_event_loop()
        tag, first_token = await loop.run_in_executor(
            None, serving.prefill_via_pd, prompt_ids
        )
        return {"pd_tag": tag, "first_token": first_token,
                "prompt_tokens": len(prompt_ids)}

    @app.post("/v1/completions")
    async def completions(request: Request):
        body = await request.json()
        prompt = body.get("prompt", "")
        prompt_ids = _encode(prompt)
        tpt = body.get("truncate_prompt_tokens")
        if tpt:
            prompt_ids = prompt_ids[-int(tpt):]
        try:
            sampling = _sampling_from(body)
        except ValueError as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "invalid_request_error"}},
                                400)
        # OpenAI-style adapter selection: model == a registered LoRA name
        lora = body.get("model")
        if lora not in serving.engine.active_loras():
            lora = None
        stops = _parse_stops(body)
        keep_stop = bool(body.get("include_stop_str_in_output"))
        pd_tag = request.headers.get("x-pd-tag")
        try:
            if pd_tag is not None:
                # PD consumer: claim the KV the prefiller shipped under this
                # tag; the request joins the decode loop directly
                loop = asyncio.get_event_loop()
                req_id, q = await loop.run_in_executor(
                    None, serving.submit_imported, int(pd_tag), sampling
                )
            else:
                req_id, q = serving.submit(
                    prompt_ids, sampling, lora_name=lora,
                    priority=int(body.get("priority", 0)),
                )
        except ValueError as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "invalid_request_error"}}, 400)
        except PDRejectedError as e:
            # decoder out of KV blocks: tell the client/router to back off
            return JSONResponse({"error": {"message": str(e),
                                           "type": "rate_limit_exceeded"}},
                                429)
        except TimeoutError as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "server_error"}}, 504)
        created = int(time.time())
        cid = f"cmpl-{uuid.uuid4().hex[:16]}"

        echo = bool(body.get("echo"))
        include_usage = bool(
            (body.get("stream_options") or {}).get("include_usage")
        )

        n_req = max(int(body.get("n", 1)), 1)
        if n_req > 64:
            return JSONResponse({"error": {"message": "n must be <= 64",
                                           "type": "invalid_request_error"}},
                                400)
        if body.get("stream") and n_req > 1:
            # interleaved multi-choice stream: sibling requests advance in
            # lockstep with the engine, so round-robin draining is fair
            extra = []
            for i in range(1, n_req):
                s_i = _sampling_from(body)
                if s_i.seed is not None:
                    s_i.seed += i
                extra.append(serving.submit(
                    prompt_ids, s_i, lora_name=lora,
                    priority=int(body.get("priority", 0)),
                ))
            chans = [(0, req_id, q)] + [
                (i + 1, rid_i, q_i) for i, (rid_i, q_i) in enumerate(extra)
            ]

            async def sse_multi():
                from fusioninfer_amd.tokenizer import IncrementalDetokenizer

                loop = asyncio.get_event_loop()
                live = {idx: (rid, qq) for idx, rid, qq in chans}
                filts = {
                    idx: (StopStreamFilter(stops, keep_stop)
                          if stops else None)
                    for idx, _, _ in chans
                }
                detoks = {
                    idx: IncrementalDetokenizer(tokenizer)
                    for idx, _, _ in chans
                }
                n_out = 0
                try:
                    while live:
                        for idx in sorted(list(live)):
                            rid, qq = live[idx]
                            tok, finished = await loop.run_in_executor(
                                None, qq.get
                            )
                            delta = (
                                detoks[idx].push(tok)
                                if tok is not None else ""
                            )
                            if finished:
                                delta += detoks[idx].flush()
                            if tok is not None:
                                n_out += 1
                            f = filts[idx]
                            by_stop = False
                            if f is not None:
                                delta, hit = f.push(delta)
                                if hit:
                                    finished = True
                                    by_stop = True
                                    serving.abort(rid)
                                elif finished:
                                    delta += f.flush()
                            chunk = {
                                "id": cid,
                                "object": "text_completion",
                                "created": created,
                                "model": model_name,
                                "choices": [{
                                    "index": idx,
                                    "text": delta,
                                    "token_ids":
                                        [tok] if tok is not None else [],
                                    "finish_reason":
                                        _finish_reason(rid, by_stop)
                                        if finished else None,
                                }],
                            }
                            yield f"data: {json.dumps(chunk)}\n\n"
                            if finished:
                                del live[idx]
                    if include_usage:
                        yield "data: " + json.dumps({
                            "id": cid, "object": "text_completion",
                            "created": created, "model": model_name,
                            "choices": [],
                            "usage": {
                                "prompt_tokens": len(prompt_ids),
                                "completion_tokens": n_out,
                                "total_tokens":
                                    len(prompt_ids) + n_out,
                            },
                        }) + "\n\n"
                    yield "data: [DONE]\n\n"
                finally:
                    for idx in list(live):
                        serving.abort(live[idx][0])

            return StreamingResponse(sse_multi(),
                                     media_type="text/event-stream")

        if body.get("stream"):
            async def sse():
                from fusioninfer_amd.tokenizer import IncrementalDetokenizer

                done = False
                n_out = 0
                filt = StopStreamFilter(stops, keep_stop) if stops else None
                detok = IncrementalDetokenizer(tokenizer)
                try:
                    if echo:
                        first = {
                            "id": cid, "object": "text_completion",
                            "created": created, "model": model_name,
                            "choices": [{"index": 0,
                                         "text": tokenizer.decode(prompt_ids),
                                         "token_ids": [],
                                         "finish_reason": None}],
                        }
                        yield f"data: {json.dumps(first)}\n\n"
                    async for tok, finished in _stream(q):
                        if tok is not None:
                            n_out += 1
                        done = finished
                        delta = detok.push(tok) if tok is not None else ""
                        if finished:
                            delta += detok.flush()
                        by_stop = False
                        if filt is not None:
                            # hold back text that could extend into a stop
                            # string; cut the stream at the first match
                            delta, hit = filt.push(delta)
                            if hit:
                                finished = True
                                by_stop = True
                            elif finished:
                                delta += filt.flush()
                        chunk = {
                            "id": cid,
                            "object": "text_completion",
                            "created": created,
                            "model": model_name,
                            "choices": [
                                {
                                    "index": 0,
                                    "text": delta,
                                    "token_ids": [tok] if tok is not None else [],
                                    "finish_reason":
                                        _finish_reason(req_id, by_stop)
                                        if finished else None,
                                }
                            ],
                        }
                        yield f"data: {json.dumps(chunk)}\n\n"
                        if finished and not done:
                            done = True
                            serving.abort(req_id)
                            break
                    if include_usage:
                        usage = {
                            "id": cid, "object": "text_completion",
                            "created": created, "model": model_name,
                            "choices": [],
                            "usage": {
                                "prompt_tokens": len(prompt_ids),
                                "completion_tokens": n_out,
                                "total_tokens": len(prompt_ids) + n_out,
                            },
                        }
                        yield f"data: {json.dumps(usage)}\n\n"
                    yield "data: [DONE]\n\n"
                finally:
                    if not done:  # client disconnected mid-stream
                        serving.abort(req_id)

            return StreamingResponse(sse(), media_type="text/event-stream")

        # OpenAI `n`: extra choices run as sibling requests — prefix
        # caching dedups the shared-prompt KV, so the n-1 extra prefills
        # recompute at most one block each
        n = max(int(body.get("n", 1)), 1)
        if n > 64:
            return JSONResponse({"error": {"message": "n must be <= 64",
                                           "type": "invalid_request_error"}},
                                400)
        extra = []
        for i in range(1, n):
            s_i = _sampling_from(body)
            if s_i.seed is not None:
                s_i.seed += i  # distinct choices under a fixed seed
            extra.append(serving.submit(
                prompt_ids, s_i, lora_name=lora,
                priority=int(body.get("priority", 0)),
            ))
        choices = []
        try:
            toks, text, reason0 = await _collect(q, stops=stops,
                                                 req_id=req_id,
                                                 keep_stop=keep_stop)
            final = serving.take_final(req_id)
            choices.append((toks, text, reason0))
            for rid_i, q_i in extra:
                choices.append(await _collect(q_i, stops=stops,
                                              req_id=rid_i,
                                              keep_stop=keep_stop))
        except asyncio.CancelledError:
            # client disconnected: stop burning engine steps on it
            serving.abort(req_id)
            for rid_i, _ in extra:
                serving.abort(rid_i)
            raise
        if echo:
            prefix = tokenizer.decode(prompt_ids)
            choices = [(t, prefix + x, r) for t, x, r in choices]
        total_completion = sum(len(t) for t, _, _ in choices)
        return JSONResponse(
            {
                "id": cid,
                "object": "text_completion",
                "created": created,
                "model": model_name,
                "choices": [
                    {
                        "index": i,
                        "text": c_text,
                        "token_ids": c_toks,
                        "finish_reason": c_reason,
                        **({"logprobs": final.logprobs,
                            "prompt_logprobs": final.prompt_logprobs}
                           if i == 0 and final is not None
                           and (body.get("logprobs")
                                or body.get("prompt_logprobs"))
                           else {}),
                    }
                    for i, (c_toks, c_text, c_reason) in enumerate(choices)
                ],
                "usage": {
                    "prompt_tokens": len(prompt_ids),
                    "completion_tokens": total_completion,
                    "total_tokens": len(prompt_ids) + total_completion,
                },
            }
        )

    @app.post("/v1/chat/completions")
    async def chat_completions(request: Request):
        body = await request.json()
        messages = body.get("messages", [])
        try:
            # vendored chat template (tokenizer_config.json) when present;
            # flat role-prefixed transcript otherwise
            text = tokenizer.apply_chat_template(messages)
        except (ValueError, TypeError, KeyError, AttributeError) as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "invalid_request_error"}},
                                400)
        prompt_ids = _encode(text)
        tpt = body.get("truncate_prompt_tokens")
        if tpt:
            prompt_ids = prompt_ids[-int(tpt):]
        try:
            sampling = _sampling_from(body)
        except ValueError as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "invalid_request_error"}},
                                400)
        stops = _parse_stops(body)
        keep_stop = bool(body.get("include_stop_str_in_output"))
        lora = body.get("model")
        if lora not in serving.engine.active_loras():
            lora = None
        try:
            req_id, q = serving.submit(
                prompt_ids, sampling, lora_name=lora,
                priority=int(body.get("priority", 0)),
            )
        except ValueError as e:
            return JSONResponse({"error": {"message": str(e),
                                           "type": "invalid_request_error"}}, 400)
        created = int(time.time())
        cid = f"chatcmpl-{uuid.uuid4().hex[:16]}"

        n_req = max(int(body.get("n", 1)), 1)
        if n_req > 64:
            return JSONResponse({"error": {"message": "n must be <= 64",
                                           "type": "invalid_request_error"}},
                                400)
        if body.get("stream") and n_req > 1:
            extra = []
            for i in range(1, n_req):
                s_i = _sampling_from(body)
                if s_i.seed is not None:
                    s_i.seed += i
                extra.append(serving.submit(
                    prompt_ids, s_i, lora_name=lora,
                    priority=int(body.get("priority", 0)),
                ))
            chans = [(0, req_id, q)] + [
                (i + 1, rid_i, q_i) for i, (rid_i, q_i) in enumerate(extra)
            ]

            async def sse_multi():
                from fusioninfer_amd.tokenizer import IncrementalDetokenizer

                loop = asyncio.get_event_loop()
                live = {idx: (rid, qq) for idx, rid, qq in chans}
                firsts = {idx: True for idx, _, _ in chans}
                filts = {
                    idx: (StopStreamFilter(stops, keep_stop)
                          if stops else None)
                    for idx, _, _ in chans
                }
                detoks = {
                    idx: IncrementalDetokenizer(tokenizer)
                    for idx, _, _ in chans
                }
                try:
                    while live:
                        for idx in sorted(list(live)):
                            rid, qq = live[idx]
                            tok, finished = await loop.run_in_executor(
                                None, qq.get
                            )
                            delta: Dict[str, Any] = {}
                            if firsts[idx]:
                                delta["role"] = "assistant"
                                firsts[idx] = False
                            piece = (
                                detoks[idx].push(tok)
                                if tok is not None else ""
                            )
                            if finished:
                                piece += detoks[idx].flush()
                            f = filts[idx]
                            by_stop = False
                            if f is not None:
                                piece, hit = f.push(piece)
                                if hit:
                                    finished = True
                                    by_stop = True
                                    serving.abort(rid)
                                elif finished:
                                    piece += f.flush()
                            if piece:
                                delta["content"] = piece
                            chunk = {
                                "id": cid,
                                "object": "chat.completion.chunk",
                                "created": created,
                                "model": model_name,
                                "choices": [{
                                    "index": idx,
                                    "delta": delta,
                                    "finish_reason":
                                        _finish_reason(rid, by_stop)
                                        if finished else None,
                                }],
                            }
                            yield f"data: {json.dumps(chunk)}\n\n"
                            if finished:
                                del live[idx]
                    yield "data: [DONE]\n\n"
                finally:
                    for idx in list(live):
                        serving.abort(live[idx][0])

            return StreamingResponse(sse_multi(),
                                     media_type="text/event-stream")

        if body.get("stream"):
            async def sse():
                from fusioninfer_amd.tokenizer import IncrementalDetokenizer

                first = True
                done = False
                filt = StopStreamFilter(stops, keep_stop) if stops else None
                detok = IncrementalDetokenizer(tokenizer)
                try:
                    async for tok, finished in _stream(q):
                        done = finished
                        delta: Dict[str, Any] = {}
                        if first:
                            delta["role"] = "assistant"
                            first = False
                        piece = (
                            detok.push(tok)
                            if tok is not None else ""
                        )
                        if finished:
                            piece += detok.flush()
                        by_stop = False
                        if filt is not None:
                            # cut the stream at the first stop-string match
                            piece, hit = filt.push(piece)
                            if hit:
                                finished = True
                                by_stop = True
                            elif finished:
                                piece += filt.flush()
                        if piece:
                            delta["content"] = piece
                        chunk = {
                            "id": cid,
                            "object": "chat.completion.chunk",
                            "created": created,
                            "model": model_name,
                            "choices": [
                                {
                                    "index": 0,
                                    "delta": delta,
                                    "finish_reason": (
                                        _finish_reason(req_id, by_stop)
                                        if finished else None
                                    ),
                                }
                            ],
                        }
                        yield f"data: {json.dumps(chunk)}\n\n"
                        if finished and not done:
                            done = True
                            serving.abort(req_id)
                            break
                    yield "data: [DONE]\n\n"
                finally:
                    if not done:  # client disconnected mid-stream
                        serving.abort(req_id)

            return StreamingResponse(sse(), media_type="text/event-stream")

        # OpenAI `n` (non-stream): extra choices as sibling requests —
        # prefix caching dedups the shared-prompt KV
        n = max(int(body.get("n", 1)), 1)
        if n > 64:
            return JSONResponse({"error": {"message": "n must be <= 64",
                                           "type": "invalid_request_error"}},
                                400)
        extra = []
        for i in range(1, n):
            s_i = _sampling_from(body)
            if s_i.seed is not None:
                s_i.seed += i
            extra.append(serving.submit(
                prompt_ids, s_i, lora_name=lora,
                priority=int(body.get("priority", 0)),
            ))
        try:
            choices = [await _collect(q, stops=stops, req_id=req_id,
                                      keep_stop=keep_stop)]
            for rid_i, q_i in extra:
                choices.append(await _collect(q_i, stops=stops,
                                              req_id=rid_i,
                                              keep_stop=keep_stop))
        except asyncio.CancelledError:
            serving.abort(req_id)
            for rid_i, _ in extra:
                serving.abort(rid_i)
            raise
        total = sum(len(t) for t, _, _ in choices)
        return JSONResponse(
            {
                "id": cid,
                "object": "chat.completion",
                "created": created,
                "model": model_name,
                "choices": [
                    {
                        "index": i,
                        "message": {
                            "role": "assistant",
                            "content": c_text,
                        },
                        "finish_reason": c_reason,
                    }
                    for i, (_t, c_text, c_reason) in enumerate(choices)
                ],
                "usage": {
                    "prompt_tokens": len(prompt_ids),
                    "completion_tokens": total,
                    "total_tokens": len(prompt_ids) + total,
                },
            }
        )

    return app
