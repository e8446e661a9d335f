"""EndpointPickerConfig interpreter: plugin graph -> per-request pick."""

from __future__ import annotations

import dataclasses
from collections import OrderedDict
from typing import Any, Dict, List, Optional, Sequence

import yaml

PREFILL_HEADER = "x-prefiller-host-port"  # marks the decode-side request


@dataclasses.dataclass
class Endpoint:
    address: str                              # host:port
    labels: Dict[str, str] = dataclasses.field(default_factory=dict)
    # engine metrics (scraped from /metrics; names per SURVEY §2.3)
    kv_cache_usage: float = 0.0               # vllm:gpu_cache_usage_perc
    queue_depth: float = 0.0                  # vllm:num_requests_waiting
    active_loras: Sequence[str] = ()


@dataclasses.dataclass
class PickResult:
    endpoint: Optional[Endpoint]
    headers: Dict[str, str] = dataclasses.field(default_factory=dict)
    # PD: the prefill endpoint chosen by the prefill profile (if any)
    prefill_endpoint: Optional[Endpoint] = None


class _PrefixCacheLRU:
    """Per-endpoint LRU of prompt-block hashes (the upstream
    prefix-cache-scorer's model of each server's KV cache)."""

    def __init__(self, block_size: int, max_blocks_to_match: int,
                 capacity_per_server: int):
        self.block_size = max(int(block_size), 1)
        self.max_match = int(max_blocks_to_match)
        self.capacity = int(capacity_per_server)
        self._lru: Dict[str, OrderedDict] = {}

    def block_hashes(self, prompt: Sequence[int]) -> List[int]:
        hashes = []
        h = 0
        nblocks = min(len(prompt) // self.block_size, self.max_match)
        for b in range(nblocks):
            chunk = tuple(prompt[b * self.block_size : (b + 1) * self.block_size])
            h = hash((h, chunk))
            hashes.append(h)
        return hashes

    def score(self, endpoint: str, hashes: List[int]) -> float:
        """Fraction of the prompt's blocks this server likely has cached
        (longest matching prefix of blocks)."""
        lru = self._lru.get(endpoint)
        if not lru or not hashes:
            return 0.0
        matched = 0
        for h in hashes:
            if h in lru:
                matched += 1
            else:
                break
        return matched / len(hashes)

    def record(self, endpoint: str, hashes: List[int]) -> None:
        lru = self._lru.setdefault(endpoint, OrderedDict())
        for h in hashes:
            if h in lru:
                lru.move_to_end(h)
            else:
                lru[h] = None
            while len(lru) > self.capacity:
                lru.popitem(last=False)


class EndpointPicker:
    """Interprets the EndpointPickerConfig YAML (the exact config shape the
    control plane renders — controlplane/router.py)."""

    def __init__(self, config_yaml: str):
        cfg = yaml.safe_load(config_yaml)
        assert cfg.get("kind") == "EndpointPickerConfig", cfg
        self.plugins: Dict[str, Dict[str, Any]] = {}
        for p in cfg.get("plugins", []):
            name = p.get("name", p["type"])
            self.plugins[name] = p
        self.profiles: List[Dict[str, Any]] = cfg.get("schedulingProfiles", [])
        self.is_pd = any(
            p["type"] == "pd-profile-handler" for p in self.plugins.values()
        )
        pc = next(
            (
                p
                for p in self.plugins.values()
                if p["type"] == "prefix-cache-scorer"
            ),
            None,
        )
        params = (pc or {}).get("parameters", {})
        self.prefix_cache = _PrefixCacheLRU(
            params.get("blockSize", params.get("hashBlockSize", 5)),
            params.get("maxPrefixBlocksToMatch", 256),
            params.get("lruCapacityPerServer", 31250),
        )

    # ------------------------------------------------------------- scoring
    def _plugin_score(
        self, ptype: str, ep: Endpoint, hashes: List[int], request: Dict
    ) -> float:
        if ptype == "prefix-cache-scorer":
            return self.prefix_cache.score(ep.address, hashes)
        if ptype == "kv-cache-utilization-scorer":
            return 1.0 - min(max(ep.kv_cache_usage, 0.0), 1.0)
        if ptype == "queue-scorer":
            return 1.0 / (1.0 + max(ep.queue_depth, 0.0))
        if ptype == "lora-affinity-scorer":
            lora = request.get("model_adapter") or request.get("lora")
            return 1.0 if lora and lora in ep.active_loras else 0.0
        return 0.0

    def _filter(self, plugin: Dict, endpoints: List[Endpoint]) -> List[Endpoint]:
        if plugin["type"] == "by-label":
            label = plugin["parameters"]["label"]
            valid = set(plugin["parameters"]["validValues"])
            return [e for e in endpoints if e.labels.get(label) in valid]
        return endpoints

    def _run_profile(
        self,
        profile: Dict[str, Any],
        endpoints: List[Endpoint],
        hashes: List[int],
        request: Dict,
    ) -> Optional[Endpoint]:
        pool = list(endpoints)
        scored: Dict[str, float] = {e.address: 0.0 for e in pool}
        for pref in profile.get("plugins", []):
            plugin = self.plugins.get(pref["pluginRef"])
            if plugin is None:
                continue
            ptype = plugin["type"]
            if ptype == "by-label":
                pool = self._filter(plugin, pool)
            elif ptype.endswith("-scorer"):
                w = float(pref.get("weight", 1))
                for e in pool:
                    scored[e.address] = scored.get(e.address, 0.0) + w * (
                        self._plugin_score(ptype, e, hashes, request)
                    )
        if not pool:
            return None
        # max-score-picker: highest weighted score; ties -> least queue
        return max(pool, key=lambda e: (scored.get(e.address, 0.0),
                                        -e.queue_depth))

    # ---------------------------------------------------------------- pick
    def pick(self, request: Dict, endpoints: List[Endpoint]) -> PickResult:
        """request: {"prompt_token_ids": [...], ...}. Returns the endpoint
        to send to; for PD, `prefill_endpoint` + the prefill header are set
        (the pd-profile-handler's prefill-then-decode flow)."""
        prompt = request.get("prompt_token_ids", [])
        hashes = self.prefix_cache.block_hashes(prompt)

        if not self.is_pd:
            profile = self.profiles[0] if self.profiles else {"plugins": []}
            ep = self._run_profile(profile, endpoints, hashes, request)
            if ep is not None:
                self.prefix_cache.record(ep.address, hashes)
            return PickResult(endpoint=ep)

        by_name = {p["name"]: p for p in self.profiles}
        prefill_ep = self._run_profile(
            by_name.get("prefill", {}), endpoints, hashes, request
        )
        decode_ep = self._run_profile(
            by_name.get("decode", {}), endpoints, hashes, request
        )
        headers = {}
        if prefill_ep is not None:
            headers[PREFILL_HEADER] = prefill_ep.address
            self.prefix_cache.record(prefill_ep.address, hashes)
        if decode_ep is not None:
            self.prefix_cache.record(decode_ep.address, hashes)
        return PickResult(
            endpoint=decode_ep, headers=headers, prefill_endpoint=prefill_ep
        )
