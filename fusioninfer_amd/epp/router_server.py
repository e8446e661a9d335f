"""First-party routing proxy: the request data path of SURVEY.md §3.3.

Plays the role of gateway + ext-proc EPP for environments without Envoy:
scores worker endpoints with the EndpointPickerConfig plugin graph
(epp/picker.py — the same YAML the control plane renders into the EPP
ConfigMap), scrapes each endpoint's vLLM-name /metrics for kv-cache
utilization and queue depth, and forwards OpenAI requests to the picked
endpoint (adding the PD prefill header when the config is PD).

Run: python -m fusioninfer_amd.epp --config epp.yaml \
         --endpoints 10.0.0.1:8000 10.0.0.2:8000 --port 8080
"""

from __future__ import annotations

import asyncio
import re
from typing import Dict, List, Optional

import httpx
from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from fusioninfer_amd.epp.picker import Endpoint, EndpointPicker

_METRIC_RE = re.compile(r"^vllm:(\w+)\{[^}]*\}\s+([0-9.eE+-]+)", re.M)


def parse_vllm_metrics(text: str) -> Dict[str, float]:
    return {m.group(1): float(m.group(2)) for m in _METRIC_RE.finditer(text)}


class MetricsScraper:
    """Keeps each endpoint's kv-util / queue-depth fresh."""

    def __init__(self, endpoints: List[Endpoint], interval_s: float = 1.0,
                 client: Optional[httpx.AsyncClient] = None):
        self.endpoints = endpoints
        self.interval_s = interval_s
        self.client = client or httpx.AsyncClient(timeout=2.0)
        self._task: Optional[asyncio.Task] = None

    async def scrape_once(self):
        for ep in self.endpoints:
            try:
                r = await self.client.get(f"http://{ep.address}/metrics")
                m = parse_vllm_metrics(r.text)
                ep.kv_cache_usage = m.get("gpu_cache_usage_perc", 0.0)
                ep.queue_depth = m.get("num_requests_waiting", 0.0)
            except Exception:
                pass  # endpoint temporarily unreachable; keep last values

    async def _loop(self):
        while True:
            await self.scrape_once()
            await asyncio.sleep(self.interval_s)

    def start(self):
        self._task = asyncio.get_event_loop().create_task(self._loop())

    def stop(self):
        if self._task:
            self._task.cancel()


def build_router_app(
    picker: EndpointPicker,
    endpoints: List[Endpoint],
    client: Optional[httpx.AsyncClient] = None,
    scrape: bool = True,
) -> FastAPI:
    app = FastAPI(title="fusioninfer-amd-router")
    http = client or httpx.AsyncClient(timeout=None)
    scraper = MetricsScraper(endpoints, client=http)

    @app.on_event("startup")
    async def _start():
        if scrape:
            scraper.start()

    @app.get("/health")
    async def health():
        return {"status": "ok", "endpoints": [e.address for e in endpoints]}

    async def _route(request: Request, path: str):
        body = await request.json()
        from fusioninfer_amd.server.api_server import encode_prompt

        prompt = body.get("prompt") or "".join(
            m.get("content", "") for m in body.get("messages", [])
        )
        token_ids = encode_prompt(prompt, 1 << 30)
        pick = picker.pick(
            {"prompt_token_ids": token_ids, "lora": body.get("model")},
            endpoints,
        )
        if pick.endpoint is None:
            return JSONResponse({"error": "no endpoint available"}, 503)
        headers = dict(pick.headers)
        if pick.prefill_endpoint is not None:
            # PD flow: prefill on the prefill-profile pick
            # (prefiller ships KV to the decoder via the connector), then
            # send the decode request claiming that KV batch by tag
            pr = await http.post(
                f"http://{pick.prefill_endpoint.address}/pd/prefill",
                json={"prompt": token_ids},
            )
            if pr.status_code != 200:
                return JSONResponse(
                    {"error": f"prefill failed: {pr.text}"}, 502
                )
            headers["x-pd-tag"] = str(pr.json()["pd_tag"])
        url = f"http://{pick.endpoint.address}{path}"
        if body.get("stream"):
            async def relay():
                async with http.stream(
                    "POST", url, json=body, headers=headers
                ) as r:
                    async for chunk in r.aiter_bytes():
                        yield chunk

            return StreamingResponse(relay(), media_type="text/event-stream")
        r = await http.post(url, json=body, headers=headers)
        return Response(
            content=r.content,
            status_code=r.status_code,
            media_type=r.headers.get("content-type"),
        )

    @app.post("/v1/completions")
    async def completions(request: Request):
        return await _route(request, "/v1/completions")

    @app.post("/v1/chat/completions")
    async def chat(request: Request):
        return await _route(request, "/v1/chat/completions")

    return app


def main(argv=None):
    import argparse

    import uvicorn

    p = argparse.ArgumentParser("fusioninfer-amd EPP router")
    p.add_argument("--config", required=True, help="EndpointPickerConfig YAML")
    p.add_argument("--endpoints", nargs="+", required=True,
                   help="worker endpoints host:port[,component-type]")
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8080)
    p.add_argument("--ext-proc", action="store_true",
                   help="serve the Envoy ext-proc gRPC protocol (the "
                        "reference's EPP wire contract) instead of the "
                        "HTTP reverse proxy")
    p.add_argument("--grpc-port", type=int, default=9002)
    p.add_argument("--health-port", type=int, default=9003)
    args = p.parse_args(argv)

    with open(args.config) as f:
        picker = EndpointPicker(f.read())
    endpoints = []
    for spec in args.endpoints:
        addr, _, ctype = spec.partition(",")
        labels = {"fusioninfer.io/component-type": ctype} if ctype else {}
        endpoints.append(Endpoint(addr, labels=labels))
    if args.ext_proc:
        import threading

        from fusioninfer_amd.epp import extproc

        server, health_server, addr, haddr = extproc.serve(
            picker, lambda: endpoints, port=args.grpc_port,
            health_port=args.health_port, host=args.host,
        )
        print(f"ext-proc gRPC on {addr}, health on {haddr}", flush=True)
        try:
            threading.Event().wait()
        finally:
            server.stop(0)
            health_server.stop(0)
        return
    app = build_router_app(picker, endpoints)
    uvicorn.run(app, host=args.host, port=args.port, log_level="warning")


if __name__ == "__main__":
    main()
