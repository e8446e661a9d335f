"""Token sampling: penalties, temperature, top-k/top-p/min-p, per-request
seeds — the vLLM SamplingParams surface the reference's delegated engine
exposes."""

from __future__ import annotations

from typing import List

import torch

from fusioninfer_amd.engine.sequence import Sequence


class Sampler:
    def __init__(self, seed: int = 0, device: str = "cpu"):
        self.device = device
        self.base_seed = seed
        self.generator = None
        if device != "cpu":
            self.generator = torch.Generator(device=device)
            self.generator.manual_seed(seed)

    def _apply_penalties(self, logits: torch.Tensor, seqs: List[Sequence]):
        """repetition / presence / frequency penalties on already-emitted
        (prompt + output) tokens, OpenAI/vLLM semantics."""
        for i, s in enumerate(seqs):
            sp = s.sampling
            if (
                sp.repetition_penalty == 1.0
                and sp.presence_penalty == 0.0
                and sp.frequency_penalty == 0.0
            ):
                continue
            token_ids = torch.tensor(
                s.all_token_ids, dtype=torch.long, device=logits.device
            )
            uniq, counts = token_ids.unique(return_counts=True)
            row = logits[i]
            if sp.repetition_penalty != 1.0:
                vals = row[uniq]
                row[uniq] = torch.where(
                    vals > 0,
                    vals / sp.repetition_penalty,
                    vals * sp.repetition_penalty,
                )
            if sp.presence_penalty != 0.0:
                row[uniq] -= sp.presence_penalty
            if sp.frequency_penalty != 0.0:
                row[uniq] -= sp.frequency_penalty * counts.to(row.dtype)
        return logits

    def sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> torch.Tensor:
        """logits: [S, V] fp32; returns [S] int64 token ids."""
        logits = self._apply_penalties(logits, seqs)
        greedy = logits.argmax(dim=-1)
        # host-side check: avoids a device sync on the (common) all-greedy
        # decode step
        if all(s.sampling.temperature == 0 for s in seqs):
            return greedy
        temps = torch.tensor(
            [s.sampling.temperature for s in seqs],
            dtype=torch.float32,
            device=logits.device,
        )
        scaled = logits / temps.clamp(min=1e-5).unsqueeze(1)

        # top-k then top-p filtering, batched over the rows that need it
        # (the per-row Python loop cost milliseconds/step at batch 256)
        V = logits.shape[-1]
        k_rows = [
            i for i, s in enumerate(seqs)
            if s.sampling.temperature != 0 and 0 < s.sampling.top_k < V
        ]
        if k_rows:
            ks = torch.tensor(
                [seqs[i].sampling.top_k for i in k_rows],
                device=logits.device,
            )
            sub = scaled[k_rows]
            top_vals = torch.topk(sub, int(ks.max()), dim=-1).values
            kth = top_vals.gather(1, (ks - 1).unsqueeze(1))
            scaled[k_rows] = sub.masked_fill(sub < kth, float("-inf"))
        p_rows = [
            i for i, s in enumerate(seqs)
            if s.sampling.temperature != 0 and s.sampling.top_p < 1.0
        ]
        if p_rows:
            ps = torch.tensor(
                [seqs[i].sampling.top_p for i in p_rows],
                device=logits.device,
            ).unsqueeze(1)
            sub = scaled[p_rows]
            sorted_logits, idx = sub.sort(dim=-1, descending=True)
            probs = torch.softmax(sorted_logits, dim=-1)
            cut = probs.cumsum(dim=-1) - probs > ps
            sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
            scaled[p_rows] = torch.empty_like(sub).scatter_(
                1, idx, sorted_logits
            )

        # min-p (vLLM surface): drop tokens whose probability is below
        # min_p * max-probability of the row
        m_rows = [
            i for i, s in enumerate(seqs)
            if s.sampling.temperature != 0 and s.sampling.min_p > 0.0
        ]
        if m_rows:
            mps = torch.tensor(
                [seqs[i].sampling.min_p for i in m_rows],
                device=logits.device,
            ).unsqueeze(1)
            sub = scaled[m_rows]
            pr = torch.softmax(sub, dim=-1)
            cut = pr < mps * pr.max(dim=-1, keepdim=True).values
            scaled[m_rows] = sub.masked_fill(cut, float("-inf"))

        probs = torch.softmax(scaled, dim=-1)
        sampled = torch.empty_like(greedy)
        # group rows by generator: per-request seeds get their own draw
        default_rows = []
        for i, s in enumerate(seqs):
            sp = s.sampling
            if sp.temperature == 0:
                sampled[i] = greedy[i]
            elif sp.seed is not None:
                g = torch.Generator(device=logits.device)
                g.manual_seed(sp.seed + len(s.output_token_ids))
                sampled[i] = torch.multinomial(probs[i], 1, generator=g)[0]
            else:
                default_rows.append(i)
        if default_rows:
            rows = torch.tensor(default_rows, device=logits.device)
            draw = torch.multinomial(
                probs[rows], 1, generator=self.generator
            ).squeeze(1)
            sampled[rows] = draw
        return sampled
