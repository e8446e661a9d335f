"""Per-step batch metadata handed from the scheduler to the model forward.

Token layout in a step's flat batch: [prefill tokens ... | decode tokens].
"""

from __future__ import annotations

import dataclasses
from typing import Optional

import torch


@dataclasses.dataclass
class AttnMetadata:
    num_prefill_tokens: int
    num_decode_tokens: int
    positions: torch.Tensor            # [T] int32, all tokens
    slot_mapping: torch.Tensor         # [T] int32, cache slot per token
    # prefill segment (context attention over the paged cache: q rows are
    # only the NEW tokens; cached prefix tokens are skipped entirely)
    cu_seqlens: Optional[torch.Tensor] = None    # [P+1] int32, NEW tokens
    tile_seq: Optional[torch.Tensor] = None      # kernel tile table
    tile_row0: Optional[torch.Tensor] = None
    tile_rows: int = 0                           # rows per tile (0 = default)
    prefill_block_tables: Optional[torch.Tensor] = None  # [P, max_blocks]
    prefill_seq_lens_k: Optional[torch.Tensor] = None    # [P] total ctx
    # decode segment
    block_tables: Optional[torch.Tensor] = None  # [Dq, max_blocks] int32
    seq_lens: Optional[torch.Tensor] = None      # [Dq] int32 (ctx incl. current)
    # LoRA groups for this step (fusioninfer_amd.lora.LoRABatch) or None
    lora: object = None

    @property
    def num_tokens(self) -> int:
        return self.num_prefill_tokens + self.num_decode_tokens
