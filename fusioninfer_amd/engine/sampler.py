"""Token sampling: greedy / temperature / top-p over the logits rows."""

from __future__ import annotations

from typing import List

import torch

from fusioninfer_amd.engine.sequence import Sequence


class Sampler:
    def __init__(self, seed: int = 0, device: str = "cpu"):
        self.generator = None
        if device != "cpu":
            self.generator = torch.Generator(device=device)
            self.generator.manual_seed(seed)

    def sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> torch.Tensor:
        """logits: [S, V] fp32/bf16; returns [S] int64 token ids (on device)."""
        temps = torch.tensor(
            [s.sampling.temperature for s in seqs],
            dtype=torch.float32,
            device=logits.device,
        )
        if bool((temps == 0).all()):
            return logits.argmax(dim=-1)
        logits = logits.float() / temps.clamp(min=1e-5).unsqueeze(1)
        top_p = torch.tensor(
            [s.sampling.top_p for s in seqs], dtype=torch.float32,
            device=logits.device,
        )
        probs = torch.softmax(logits, dim=-1)
        if bool((top_p < 1.0).any()):
            sorted_probs, idx = probs.sort(dim=-1, descending=True)
            cum = sorted_probs.cumsum(dim=-1)
            mask = cum - sorted_probs > top_p.unsqueeze(1)
            sorted_probs[mask] = 0.0
            sorted_probs /= sorted_probs.sum(dim=-1, keepdim=True)
            choice = torch.multinomial(
                sorted_probs, 1, generator=self.generator
            ).squeeze(1)
            sampled = idx.gather(1, choice.unsqueeze(1)).squeeze(1)
        else:
            sampled = torch.multinomial(
                probs, 1, generator=self.generator
            ).squeeze(1)
        greedy = logits.argmax(dim=-1)
        return torch.where(temps == 0, greedy, sampled)
