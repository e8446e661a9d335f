"""Serving wrapper: background engine-stepping thread + per-request streams.

The HTTP layer (api_server.py) submits token-id prompts; this class owns
the LLMEngine, steps it continuously on a dedicated thread, and fans out
per-request token deltas through thread-safe queues.
"""

from __future__ import annotations

import queue
import threading
from typing import Dict, List, Optional, Tuple

from fusioninfer_amd.config import EngineConfig
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams


class ServingEngine:
    def __init__(self, cfg: EngineConfig, device: Optional[str] = None):
        self.engine = LLMEngine(cfg, device=device)
        self._streams: Dict[str, "queue.Queue[Tuple[Optional[int], bool]]"] = {}
        self._lock = threading.Lock()
        self._work = threading.Event()
        self._stop = False
        self.healthy = True
        self.last_error: str = ""
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    # ------------------------------------------------------------- public
    def submit(
        self, prompt_token_ids: List[int], sampling: SamplingParams,
        lora_name=None,
    ) -> Tuple[str, "queue.Queue[Tuple[Optional[int], bool]]"]:
        q: "queue.Queue[Tuple[Optional[int], bool]]" = queue.Queue()
        with self._lock:
            req_id = self.engine.add_request(
                prompt_token_ids, sampling, lora_name=lora_name
            )
            self._streams[req_id] = q
        self._work.set()
        return req_id, q

    def metrics(self) -> Dict[str, float]:
        e = self.engine
        return {
            "gpu_cache_usage_perc": e.gpu_cache_usage(),
            "num_requests_waiting": float(e.num_waiting()),
            "num_requests_running": float(e.num_running()),
            "generation_tokens_total": float(e.num_generated_tokens),
            "prompt_tokens_total": float(e.num_prefilled_tokens),
            "request_success_total": float(e.num_finished),
            "time_to_first_token_seconds_sum": e.ttft_sum,
            "time_to_first_token_seconds_count": float(e.num_finished),
            "e2e_request_latency_seconds_sum": e.e2e_latency_sum,
            "e2e_request_latency_seconds_count": float(e.num_finished),
            "engine_step_seconds_sum": e.step_time_sum,
            "engine_step_seconds_count": float(e.num_steps),
        }

    def abort(self, request_id: str) -> bool:
        with self._lock:
            self._streams.pop(request_id, None)
            return self.engine.abort_request(request_id)

    def shutdown(self):
        self._stop = True
        self._work.set()
        self._thread.join(timeout=5)

    # -------------------------------------------------------------- loop
    def _loop(self):
        while not self._stop:
            with self._lock:
                has_work = self.engine.has_unfinished()
            if not has_work:
                self._work.clear()
                self._work.wait(timeout=0.25)
                continue
            try:
                with self._lock:
                    outputs = self.engine.step()
            except Exception as e:  # engine fault: fail requests, go unhealthy
                self.healthy = False
                self.last_error = repr(e)
                with self._lock:
                    for q in self._streams.values():
                        q.put((None, True))
                    self._streams.clear()
                return
            for out in outputs:
                q = self._streams.get(out.request_id)
                if q is None:
                    continue
                tok = out.output_token_ids[-1] if out.output_token_ids else None
                q.put((tok, out.finished))
                if out.finished:
                    with self._lock:
                        self._streams.pop(out.request_id, None)
