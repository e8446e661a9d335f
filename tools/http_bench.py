"""End-to-end HTTP serving check: start the OpenAI server (subprocess),
fire concurrent requests, report latency/throughput. Validates the full
server stack (FastAPI + serving thread + engine) on real hardware.

Usage: python tools/http_bench.py [--model Qwen3-8B] [--requests 64]
"""

import argparse
import asyncio
import json
import os
import signal
import subprocess
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import httpx


async def run_load(port, n_requests, prompt_len, max_tokens, concurrency):
    async with httpx.AsyncClient(timeout=300.0) as client:
        base = f"http://127.0.0.1:{port}"
        # wait for readiness
        for _ in range(600):
            try:
                r = await client.get(f"{base}/health")
                if r.status_code == 200:
                    break
            except Exception:
                pass
            await asyncio.sleep(1.0)
        else:
            raise RuntimeError("server never became healthy")

        sem = asyncio.Semaphore(concurrency)
        latencies = []

        async def one(i):
            async with sem:
                t0 = time.monotonic()
                r = await client.post(
                    f"{base}/v1/completions",
                    json={
                        "prompt": [(i * 7 + j) % 50000 for j in range(prompt_len)],
                        "max_tokens": max_tokens,
                        "temperature": 0,
                        "ignore_eos": True,
                    },
                )
                r.raise_for_status()
                body = r.json()
                assert body["usage"]["completion_tokens"] == max_tokens
                latencies.append(time.monotonic() - t0)

        t0 = time.monotonic()
        await asyncio.gather(*(one(i) for i in range(n_requests)))
        elapsed = time.monotonic() - t0
        m = await client.get(f"{base}/metrics")
        assert "vllm:gpu_cache_usage_perc" in m.text
        latencies.sort()
        return {
            "requests": n_requests,
            "elapsed_s": round(elapsed, 2),
            "req_per_s": round(n_requests / elapsed, 2),
            "p50_latency_s": round(latencies[len(latencies) // 2], 2),
            "p99_latency_s": round(latencies[int(len(latencies) * 0.99)], 2),
        }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="Qwen3-8B")
    p.add_argument("--port", type=int, default=8071)
    p.add_argument("--requests", type=int, default=64)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--max-tokens", type=int, default=32)
    p.add_argument("--concurrency", type=int, default=32)
    args = p.parse_args()

    server = subprocess.Popen(
        [
            sys.executable, "-m", "fusioninfer_amd.server",
            "--model", args.model, "--port", str(args.port),
            "--max-model-len", str(args.prompt_len + args.max_tokens + 64),
            "--enable-prefix-caching",
        ],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    )
    try:
        stats = asyncio.run(
            run_load(args.port, args.requests, args.prompt_len,
                     args.max_tokens, args.concurrency)
        )
        print(json.dumps(stats))
    finally:
        server.send_signal(signal.SIGTERM)  # exact PID, never pattern-kill
        try:
            server.wait(timeout=15)
        except subprocess.TimeoutExpired:
            server.kill()


if __name__ == "__main__":
    main()
