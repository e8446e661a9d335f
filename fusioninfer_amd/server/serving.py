"""Serving wrapper: background engine-stepping thread + per-request streams.

The HTTP layer (api_server.py) submits token-id prompts; this class owns
the LLMEngine, steps it continuously on a dedicated thread, and fans out
per-request token deltas through thread-safe queues.
"""

from __future__ import annotations

import itertools
import os
import queue
import threading
import time
from typing import Dict, List, Optional, Tuple

from fusioninfer_amd.config import EngineConfig
from fusioninfer_amd.distributed.kv_transfer import KV_CONSUMER, KV_PRODUCER
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams


class PDRejectedError(RuntimeError):
    """The decoder could not admit an imported KV batch within its wait
    budget (block exhaustion) — maps to HTTP 429."""


class ServingEngine:
    def __init__(
        self,
        cfg: EngineConfig,
        device: Optional[str] = None,
        kv_connector=None,
        import_block_wait_s: float = 30.0,
    ):
        self.engine = LLMEngine(cfg, device=device)
        self._streams: Dict[str, "queue.Queue[Tuple[Optional[int], bool]]"] = {}
        # finished RequestOutputs kept briefly so the HTTP layer can
        # attach logprobs / prompt_logprobs after draining the stream
        self._final: Dict[str, object] = {}
        self._lock = threading.Lock()
        self._work = threading.Event()
        self._stop = False
        self.healthy = True
        self.last_error: str = ""
        # engine-hang watchdog (failure detection is
        # Kubernetes-native — a wedged collective / kernel turns /health
        # 503 so the liveness probe restarts the pod). A step that has
        # been running longer than this is considered hung.
        self.step_timeout_s: float = float(
            os.environ.get("FI_STEP_TIMEOUT_S", "120")
        )
        self._step_started: Optional[float] = None
        # PD disaggregation: producer prefills+ships KV,
        # consumer receives KV on a background thread and admits on claim
        self.kv_connector = kv_connector
        self.engine.kv_connector = kv_connector
        self.import_block_wait_s = import_block_wait_s
        self._pd_tags = itertools.count(1)
        self._pending_imports: Dict[int, tuple] = {}
        self._rejected_imports: set = set()
        self._pd_cond = threading.Condition()
        # async export: finished prefills queue here; a dedicated sender
        # thread ships KV so the engine loop keeps prefilling the next
        # requests while transfers are in flight (SURVEY §7 stage-3
        # overlap; VERDICT round-1 item 8)
        self._pd_send_q: "queue.Queue" = queue.Queue()
        if kv_connector is not None and kv_connector.role == KV_PRODUCER:
            threading.Thread(target=self._pd_send_loop, daemon=True).start()
        if kv_connector is not None and kv_connector.role == KV_CONSUMER:
            threading.Thread(target=self._pd_recv_loop, daemon=True).start()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    # ------------------------------------------------------------- public
    def submit(
        self, prompt_token_ids: List[int], sampling: SamplingParams,
        lora_name=None, priority: int = 0,
    ) -> Tuple[str, "queue.Queue[Tuple[Optional[int], bool]]"]:
        q: "queue.Queue[Tuple[Optional[int], bool]]" = queue.Queue()
        with self._lock:
            req_id = self.engine.add_request(
                prompt_token_ids, sampling, lora_name=lora_name,
                priority=priority,
            )
            self._streams[req_id] = q
        self._work.set()
        return req_id, q

    # ---------------------------------------------------- PD disaggregation
    def prefill_via_pd(self, prompt_token_ids: List[int],
                       priority: int = 0) -> Tuple[int, int]:
        """Producer role: prefill a prompt (batched with other requests in
        the normal engine loop — concurrent PD prefills do NOT serialize),
        queue its KV for the sender thread, and return (pd_tag,
        first_token) IMMEDIATELY: the transfer overlaps both the next
        prefill and the router->decoder round trip. The tag travels in the
        KV header; out-of-order decode claims still match their own KV."""
        assert self.kv_connector is not None and \
            self.kv_connector.role == KV_PRODUCER, "not a PD prefiller"
        q: "queue.Queue[Tuple[Optional[int], bool]]" = queue.Queue()
        with self._lock:
            req_id = self.engine.add_export_request(prompt_token_ids,
                                                    priority=priority)
            self._streams[req_id] = q
        self._work.set()
        first_token = None
        finished = False
        while not finished:
            tok, finished = q.get(timeout=300.0)
            if tok is not None:
                first_token = tok
        tag = next(self._pd_tags)
        self._pd_send_q.put((req_id, tag, len(prompt_token_ids), first_token))
        return tag, first_token

    def _pd_send_loop(self):
        """Producer sender thread: ships held exports in tag order (one
        connector, one ordered p2p channel)."""
        while not self._stop:
            try:
                req_id, tag, prompt_len, first_token = self._pd_send_q.get(
                    timeout=0.25
                )
            except queue.Empty:
                continue
            try:
                with self._lock:
                    self.engine.pd_send_held(
                        self.kv_connector, req_id, prompt_len, first_token,
                        tag,
                    )
                    self.engine.release_held(req_id)
            except Exception as e:
                if self._stop:
                    return
                self.healthy = False
                self.last_error = repr(e)
                return

    def _pd_recv_loop(self):
        from fusioninfer_amd.engine.llm_engine import InsufficientBlocksError

        def alloc(n):
            """Backpressure: wait for decode completions to free blocks;
            past the budget return None — the connector drains the wire
            and the request is rejected (429)."""
            deadline = time.monotonic() + self.import_block_wait_s
            while not self._stop:
                with self._lock:
                    try:
                        ids = self.engine.allocate_import_blocks(n)
                        self.engine.pd_recv_broadcast(ids)
                        return ids
                    except InsufficientBlocksError:
                        pass
                if time.monotonic() >= deadline:
                    return None
                time.sleep(0.01)
            return None

        while not self._stop:
            try:
                kwargs = {}
                if hasattr(self.kv_connector, "channel"):  # in-memory: poll
                    kwargs["timeout"] = 0.25
                ids, prompt_len, first_token, tag = self.kv_connector.recv_kv(
                    self.engine.runner.kv_caches, alloc, **kwargs
                )
            except queue.Empty:
                continue
            except Exception as e:
                if self._stop:
                    return
                self.healthy = False
                self.last_error = repr(e)
                return
            if ids is None:  # rejected under block exhaustion
                with self._pd_cond:
                    self._rejected_imports.add(tag)
                    self._pd_cond.notify_all()
                continue
            with self._lock:
                holder = self.engine.take_import_holder()
            with self._pd_cond:
                self._pending_imports[tag] = (holder, prompt_len, first_token)
                self._pd_cond.notify_all()

    def submit_imported(
        self, pd_tag: int, sampling: SamplingParams, timeout: float = 60.0
    ) -> Tuple[str, "queue.Queue[Tuple[Optional[int], bool]]"]:
        """Consumer role: claim the KV batch tagged `pd_tag` (waiting for the
        connector to deliver it if needed) and admit the request straight
        into the decode loop."""
        with self._pd_cond:
            while pd_tag not in self._pending_imports:
                if pd_tag in self._rejected_imports:
                    self._rejected_imports.discard(pd_tag)
                    raise PDRejectedError(
                        f"KV for pd_tag={pd_tag} rejected: decoder out of "
                        f"KV blocks (retry later)"
                    )
                if not self._pd_cond.wait(timeout=timeout):
                    raise TimeoutError(f"KV for pd_tag={pd_tag} never arrived")
            holder, prompt_len, first_token = self._pending_imports.pop(pd_tag)
        q: "queue.Queue[Tuple[Optional[int], bool]]" = queue.Queue()
        with self._lock:
            req_id = self.engine.add_imported_request(
                prompt_len, first_token, sampling, holder=holder
            )
            if sampling.max_tokens <= 1:
                # first (prefiller-sampled) token already satisfies the
                # budget; finish without a decode step
                self.engine.abort_request(req_id)
                q.put((first_token, True))
                return req_id, q
            self._streams[req_id] = q
        q.put((first_token, False))
        self._work.set()
        return req_id, q

    def check_health(self) -> bool:
        """Liveness: False once the engine loop died on an exception OR
        the current step has been stuck past step_timeout_s (wedged
        collective, hung kernel). One-way: a hung engine cannot recover
        in-process — the pod restart is the recovery path."""
        started = self._step_started
        if (self.healthy and started is not None
                and time.monotonic() - started > self.step_timeout_s):
            self.healthy = False
            self.last_error = (
                f"engine step exceeded {self.step_timeout_s:.0f}s "
                "(hung collective or kernel)"
            )
        return self.healthy

    def embed(self, prompt_token_ids, pooling: str = "last"):
        """One pooled-embedding forward under the engine lock (runs
        between engine steps, like the PD sender)."""
        with self._lock:
            return self.engine.embed([prompt_token_ids], pooling=pooling)[0]

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
            "pipelined_decode_steps_total": float(e.num_async_steps),
            "num_preemptions_total": float(e.num_preemptions),
            "num_swap_outs_total": float(e.num_swap_outs),
            "spec_decode_num_draft_tokens_total": float(
                e.num_spec_draft_tokens
            ),
            "spec_decode_num_accepted_tokens_total": float(
                e.num_spec_accepted_tokens
            ),
            "prefix_cache_queries_total": float(
                e.block_manager.cache_query_tokens
            ),
            "prefix_cache_hits_total": float(
                e.block_manager.cache_hit_tokens
            ),
        }

    def take_final(self, request_id: str):
        """Pop the finished RequestOutput (None if aborted mid-flight)."""
        with self._lock:
            return self._final.pop(request_id, None)

    def peek_final(self, request_id: str):
        """Non-destructive read of the finished RequestOutput (finish
        reason lookups must not steal the logprobs consumer's pop; the
        _final dict stays bounded by the 4096-entry cap)."""
        with self._lock:
            return self._final.get(request_id)

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
                self._step_started = None
                self._work.clear()
                self._work.wait(timeout=0.25)
                continue
            try:
                self._step_started = time.monotonic()
                with self._lock:
                    outputs = self.engine.step()
                self._step_started = None
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
                # record the final BEFORE pushing the finished flag so a
                # consumer that wakes on it can take_final() immediately
                if out.finished:
                    with self._lock:
                        self._streams.pop(out.request_id, None)
                        self._final[out.request_id] = out
                        if len(self._final) > 4096:  # belt-and-braces cap
                            self._final.pop(next(iter(self._final)))
                # speculative steps emit up to k+1 tokens at once; the
                # finished flag rides on the last one
                toks = out.new_token_ids or [None]
                for i, tok in enumerate(toks):
                    q.put((tok, out.finished and i == len(toks) - 1))
