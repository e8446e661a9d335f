"""PD-disaggregation KV connector: prefiller -> decoder KV handoff.

MI355X-native equivalent of the reference's PyNcclConnector /
NixlConnector flag surface (reference docs/.../core-design.md:88-111,
SURVEY.md §2.3): the prefiller (kv_producer) packs a request's KV blocks
from EVERY layer into ONE contiguous staging tensor (HIP gather kernel)
and ships it with a single RCCL send — xGMI is point-to-point (7 links x
~153 GB/s), so one large send per request hits link peak where many
per-layer sends would not. The decoder (kv_consumer)
receives and scatters into its own paged cache, then decodes.

Wire protocol per request (tags are implicit via ordering on the p2p pair;
the header carries an application-level pd_tag so the HTTP layer can match
a received KV batch to the decode request that claims it):
  1. header  int64[4]: [num_blocks, prompt_len, first_token, pd_tag]
  2. staging bf16 [layers, 2, num_blocks, Hk, bs, D]

`InMemoryKVConnector` is the same protocol over an in-process queue —
used by single-process tests and demos of the HTTP PD flow (the RCCL
transport itself is covered by the two-process gloo/RCCL tests).
"""

from __future__ import annotations

import queue
from typing import List, Tuple

import torch
import torch.distributed as dist

import fusioninfer_amd.ops as ops

KV_PRODUCER = "kv_producer"
KV_CONSUMER = "kv_consumer"


def pack_kv_blocks(
    kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
    block_ids: torch.Tensor,
) -> torch.Tensor:
    """One contiguous staging tensor across all layers:
    [layers, 2, num_blocks, Hk, bs, D]."""
    per_layer = [ops.gather_kv_blocks(kc, vc, block_ids) for kc, vc in kv_caches]
    return torch.stack(per_layer, dim=0).contiguous()


def unpack_kv_blocks(staging, kv_caches, block_ids: torch.Tensor) -> None:
    for layer, (kc, vc) in enumerate(kv_caches):
        ops.scatter_kv_blocks(staging[layer], kc, vc, block_ids)


class RcclKVConnector:
    """Point-to-point KV mover between a prefiller rank and a decoder rank."""

    def __init__(
        self,
        role: str,
        peer_rank: int,
        group=None,
        device: str = "cpu",
    ):
        assert role in (KV_PRODUCER, KV_CONSUMER), role
        self.role = role
        self.peer_rank = peer_rank
        self.group = group
        self.device = torch.device(device)

    # ------------------------------------------------------------- producer
    def send_kv(
        self,
        kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
        block_ids: List[int],
        prompt_len: int,
        first_token: int,
        tag: int = 0,
    ) -> None:
        assert self.role == KV_PRODUCER
        ids = torch.tensor(block_ids, dtype=torch.int32, device=self.device)
        header = torch.tensor(
            [len(block_ids), prompt_len, first_token, tag], dtype=torch.int64,
            device=self.device,
        )
        dist.send(header, self.peer_rank, group=self.group)
        dist.send(pack_kv_blocks(kv_caches, ids), self.peer_rank,
                  group=self.group)

    # ------------------------------------------------------------- consumer
    def recv_kv(
        self,
        kv_caches: List[Tuple[torch.Tensor, torch.Tensor]],
        allocate_blocks,  # callable(num_blocks) -> List[int]
    ) -> Tuple[List[int], int, int, int]:
        """Receives one request's KV; returns (block_ids, prompt_len,
        first_token, tag)."""
        assert self.role == KV_CONSUMER
        header = torch.zeros(4, dtype=torch.int64, device=self.device)
        dist.recv(header, self.peer_rank, group=self.group)
        num_blocks, prompt_len, first_token, tag = (
            int(header[0]), int(header[1]), int(header[2]), int(header[3])
        )
        block_ids = allocate_blocks(num_blocks)
        kc0, _ = kv_caches[0]
        layers = len(kv_caches)
        staging = torch.empty(
            (layers, 2, num_blocks, *kc0.shape[1:]),
            dtype=kc0.dtype,
            device=self.device,
        )
        dist.recv(staging, self.peer_rank, group=self.group)
        if block_ids is None:
            # backpressure overflow: the wire must stay consistent, so the
            # payload is drained into scratch and DROPPED (caller rejects
            # the request — HTTP 429)
            return None, prompt_len, first_token, tag
        assert len(block_ids) == num_blocks
        ids = torch.tensor(block_ids, dtype=torch.int32, device=self.device)
        unpack_kv_blocks(staging, kv_caches, ids)
        return block_ids, prompt_len, first_token, tag


class InMemoryKVConnector:
    """Same wire semantics over an in-process queue (single-process tests
    and demos of the HTTP PD flow). Construct a pair with
    make_inmemory_pair()."""

    def __init__(self, role: str, channel: "queue.Queue"):
        assert role in (KV_PRODUCER, KV_CONSUMER), role
        self.role = role
        self.channel = channel

    def send_kv(self, kv_caches, block_ids, prompt_len, first_token,
                tag: int = 0) -> None:
        assert self.role == KV_PRODUCER
        ids = torch.tensor(block_ids, dtype=torch.int32,
                           device=kv_caches[0][0].device)
        staging = pack_kv_blocks(kv_caches, ids).clone()
        self.channel.put((len(block_ids), prompt_len, first_token, tag, staging))

    def recv_kv(self, kv_caches, allocate_blocks, timeout=None):
        assert self.role == KV_CONSUMER
        num_blocks, prompt_len, first_token, tag, staging = self.channel.get(
            timeout=timeout
        )
        block_ids = allocate_blocks(num_blocks)
        if block_ids is None:  # backpressure overflow: drop (see Rccl form)
            return None, prompt_len, first_token, tag
        ids = torch.tensor(block_ids, dtype=torch.int32,
                           device=kv_caches[0][0].device)
        unpack_kv_blocks(staging.to(kv_caches[0][0].device), kv_caches, ids)
        return block_ids, prompt_len, first_token, tag


def make_inmemory_pair() -> Tuple[InMemoryKVConnector, InMemoryKVConnector]:
    ch: "queue.Queue" = queue.Queue()
    return (
        InMemoryKVConnector(KV_PRODUCER, ch),
        InMemoryKVConnector(KV_CONSUMER, ch),
    )
