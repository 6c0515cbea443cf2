"""Sampler: logits -> token ids.

Pipeline follows the reference sampler (vllm/v1/sample/sampler.py:72):
logit bias / allowed-tokens mask -> penalties -> temperature -> min-p ->
top-k/top-p -> sample (greedy fast path) -> logprobs. Vectorized torch;
runs on GPU or CPU. Hot paths (top-k/p) get HIP kernels later.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

import torch

from vllm_amd.sampling_params import SamplingParams


@dataclass
class SamplingMetadata:
    """Per-batch tensors for the requests being sampled this step."""

    temperature: torch.Tensor  # [B] float
    top_p: torch.Tensor  # [B] float
    top_k: torch.Tensor  # [B] int
    min_p: torch.Tensor  # [B] float
    repetition_penalty: torch.Tensor  # [B]
    presence_penalty: torch.Tensor  # [B]
    frequency_penalty: torch.Tensor  # [B]
    all_greedy: bool
    all_random: bool
    no_penalties: bool
    no_top_k: bool
    no_top_p: bool
    no_min_p: bool
    max_num_logprobs: int  # 0 -> no logprobs requested
    num_logprobs: list[int] = field(default_factory=list)  # per request
    generators: dict[int, torch.Generator] = field(default_factory=dict)
    # Token-count matrices for penalties (built lazily, [B, vocab]).
    prompt_token_ids: Optional[list[list[int]]] = None
    output_token_ids: Optional[list[list[int]]] = None
    logit_bias: Optional[list[Optional[dict[int, float]]]] = None
    allowed_token_ids: Optional[list[Optional[list[int]]]] = None
    # Per-row tokens banned THIS step (bad_words sequence completion).
    bad_token_ids: Optional[list[Optional[list[int]]]] = None
    # Per-row custom processors fn(output_ids, logits_row) -> logits_row.
    logits_processors: Optional[list] = None
    min_tokens_mask: Optional[list[Optional[tuple[int, set[int]]]]] = None
    # Structured output: per-row allowed-token sets for THIS step
    # (attached by the runner; None rows are unconstrained).
    grammar_masks: Optional[list[Optional[set]]] = None

    @classmethod
    def build(
        cls,
        params: list[SamplingParams],
        prompt_token_ids: list[list[int]],
        output_token_ids: list[list[int]],
        device: torch.device,
        seeds_offset: list[int],
    ) -> "SamplingMetadata":
        B = len(params)
        temp = torch.tensor([p.temperature for p in params],
                            dtype=torch.float32, device=device)
        top_p = torch.tensor([p.top_p for p in params], dtype=torch.float32,
                             device=device)
        top_k = torch.tensor([p.top_k for p in params], dtype=torch.int64,
                             device=device)
        min_p = torch.tensor([p.min_p for p in params], dtype=torch.float32,
                             device=device)
        rep = torch.tensor([p.repetition_penalty for p in params],
                           dtype=torch.float32, device=device)
        pres = torch.tensor([p.presence_penalty for p in params],
                            dtype=torch.float32, device=device)
        freq = torch.tensor([p.frequency_penalty for p in params],
                            dtype=torch.float32, device=device)
        generators: dict[int, torch.Generator] = {}
        for i, p in enumerate(params):
            if p.temperature > 0 and p.seed is not None:
                g = torch.Generator(device=device)
                # Seed advanced by tokens generated so far -> reproducible
                # across preemption/resume.
                g.manual_seed(p.seed + seeds_offset[i])
                generators[i] = g
        no_pen = all(
            p.repetition_penalty == 1.0
            and p.presence_penalty == 0.0
            and p.frequency_penalty == 0.0
            for p in params
        )
        logit_bias = [p.logit_bias for p in params]
        allowed = [p.allowed_token_ids for p in params]
        min_tok = []
        for p, out in zip(params, output_token_ids):
            if p.min_tokens and len(out) < p.min_tokens:
                min_tok.append((p.min_tokens, p.all_stop_token_ids))
            else:
                min_tok.append(None)
        bad_ids: list = []
        for p, prompt, out in zip(params, prompt_token_ids,
                                  output_token_ids):
            seqs = p._bad_words_token_ids
            if not seqs:
                bad_ids.append(None)
                continue
            tail = (list(prompt) + list(out))
            banned = []
            for seq in seqs:
                if len(seq) == 1 or tail[-(len(seq) - 1):] == seq[:-1]:
                    banned.append(seq[-1])
            bad_ids.append(banned or None)
        num_logprobs = [p.logprobs or 0 for p in params]
        return cls(
            temperature=temp,
            top_p=top_p,
            top_k=top_k,
            min_p=min_p,
            repetition_penalty=rep,
            presence_penalty=pres,
            frequency_penalty=freq,
            all_greedy=all(p.temperature == 0.0 for p in params),
            all_random=all(p.temperature > 0.0 for p in params),
            no_penalties=no_pen,
            no_top_k=all(p.top_k == 0 for p in params),
            no_top_p=all(p.top_p >= 1.0 for p in params),
            no_min_p=all(p.min_p == 0.0 for p in params),
            max_num_logprobs=max(num_logprobs) if num_logprobs else 0,
            num_logprobs=num_logprobs,
            generators=generators,
            prompt_token_ids=prompt_token_ids,
            output_token_ids=output_token_ids,
            logit_bias=(None if all(b is None for b in logit_bias)
                        else logit_bias),
            allowed_token_ids=(None if all(a is None for a in allowed)
                               else allowed),
            min_tokens_mask=(None if all(m is None for m in min_tok)
                             else min_tok),
            bad_token_ids=(None if all(b is None for b in bad_ids)
                           else bad_ids),
            logits_processors=(
                None if all(not p.logits_processors for p in params)
                else [p.logits_processors for p in params]),
        )


@dataclass
class SamplerOutput:
    sampled_token_ids: torch.Tensor  # [B] int64
    # logprobs[i] = list over sampled positions of {token_id: logprob}
    logprobs: Optional[list[Optional[dict[int, float]]]] = None


class Sampler(torch.nn.Module):

    def forward(
        self, logits: torch.Tensor, meta: SamplingMetadata
    ) -> SamplerOutput:
        logits = logits.float()
        B, V = logits.shape

        if meta.logit_bias is not None:
            for i, bias in enumerate(meta.logit_bias):
                if bias:
                    ids = torch.tensor(list(bias.keys()), device=logits.device)
                    vals = torch.tensor(list(bias.values()),
                                        dtype=torch.float32,
                                        device=logits.device)
                    logits[i].index_add_(0, ids, vals)
        if meta.logits_processors is not None:
            for i, procs in enumerate(meta.logits_processors):
                if procs:
                    row = logits[i]
                    for fn in procs:
                        row = fn(meta.output_token_ids[i], row)
                    logits[i] = row
        if meta.bad_token_ids is not None:
            for i, banned in enumerate(meta.bad_token_ids):
                if banned:
                    logits[i, torch.tensor(banned,
                                           device=logits.device)] = \
                        float("-inf")
        if meta.allowed_token_ids is not None:
            for i, allowed in enumerate(meta.allowed_token_ids):
                if allowed is not None:
                    mask = torch.ones(V, dtype=torch.bool,
                                      device=logits.device)
                    mask[torch.tensor(allowed, device=logits.device)] = False
                    logits[i].masked_fill_(mask, float("-inf"))
        if meta.grammar_masks is not None:
            for i, allowed in enumerate(meta.grammar_masks):
                if allowed is not None:
                    mask = torch.ones(V, dtype=torch.bool,
                                      device=logits.device)
                    if allowed:
                        mask[torch.tensor(sorted(allowed),
                                          device=logits.device)] = False
                    logits[i].masked_fill_(mask, float("-inf"))
        if meta.min_tokens_mask is not None:
            for i, mt in enumerate(meta.min_tokens_mask):
                if mt is not None:
                    _, stop_ids = mt
                    for tid in stop_ids:
                        logits[i, tid] = float("-inf")

        if not meta.no_penalties:
            logits = self._apply_penalties(logits, meta)

        if meta.all_greedy:
            sampled = logits.argmax(dim=-1)
        else:
            sampled = self._sample(logits, meta)

        logprobs_out = None
        if meta.max_num_logprobs > 0:
            lp = torch.log_softmax(logits, dim=-1)
            # Clamp to vocab: a larger request means "full vocab", and
            # an unclamped topk would crash the step.
            k = min(meta.max_num_logprobs, logits.shape[-1])
            topv, topi = lp.topk(k, dim=-1)
            sampled_lp = lp.gather(-1, sampled.unsqueeze(-1))
            logprobs_out = []
            for i in range(B):
                n = min(meta.num_logprobs[i], k)
                if n == 0:
                    logprobs_out.append(None)
                    continue
                d = {int(topi[i, j]): float(topv[i, j]) for j in range(n)}
                d[int(sampled[i])] = float(sampled_lp[i, 0])
                logprobs_out.append(d)
        return SamplerOutput(sampled_token_ids=sampled, logprobs=logprobs_out)

    def _apply_penalties(
        self, logits: torch.Tensor, meta: SamplingMetadata
    ) -> torch.Tensor:
        B, V = logits.shape
        device = logits.device
        out_counts = torch.zeros(B, V, dtype=torch.float32, device=device)
        prompt_mask = torch.zeros(B, V, dtype=torch.bool, device=device)
        for i in range(B):
            out = meta.output_token_ids[i]
            if out:
                ids = torch.tensor(out, device=device)
                out_counts[i].index_add_(
                    0, ids, torch.ones(len(out), device=device)
                )
            pt = meta.prompt_token_ids[i]
            if pt:
                prompt_mask[i, torch.tensor(pt, device=device)] = True
        out_mask = out_counts > 0
        seen_mask = prompt_mask | out_mask
        # Repetition penalty (divides positive, multiplies negative logits).
        rp = meta.repetition_penalty.unsqueeze(1)
        penalized = torch.where(
            logits > 0, logits / rp, logits * rp
        )
        logits = torch.where(seen_mask, penalized, logits)
        # Frequency / presence penalties (output tokens only).
        logits -= meta.frequency_penalty.unsqueeze(1) * out_counts
        logits -= meta.presence_penalty.unsqueeze(1) * out_mask.float()
        return logits

    def _sample(
        self, logits: torch.Tensor, meta: SamplingMetadata
    ) -> torch.Tensor:
        B, V = logits.shape
        temp = meta.temperature.clamp(min=1e-5).unsqueeze(1)
        scaled = logits / temp

        if not meta.no_top_k:
            k = meta.top_k.clamp(min=0)
            for i in range(B):
                ki = int(k[i])
                if 0 < ki < V:
                    kth = scaled[i].topk(ki).values[-1]
                    scaled[i][scaled[i] < kth] = float("-inf")
        if not meta.no_top_p:
            sorted_logits, sorted_idx = scaled.sort(dim=-1, descending=True)
            probs = sorted_logits.softmax(dim=-1)
            cum = probs.cumsum(dim=-1)
            # Keep tokens until cumulative prob exceeds top_p (always >=1).
            keep = cum - probs < meta.top_p.unsqueeze(1)
            sorted_logits[~keep] = float("-inf")
            scaled = torch.full_like(scaled, float("-inf")).scatter(
                -1, sorted_idx, sorted_logits
            )
        if not meta.no_min_p:
            probs = scaled.softmax(dim=-1)
            maxp = probs.max(dim=-1, keepdim=True).values
            scaled[probs < meta.min_p.unsqueeze(1) * maxp] = float("-inf")

        probs = scaled.softmax(dim=-1)
        sampled = torch.empty(B, dtype=torch.int64, device=logits.device)
        # Requests with shared default generator sampled batched; seeded
        # ones individually.
        unseeded = [i for i in range(B) if i not in meta.generators]
        greedy_rows = (meta.temperature == 0.0).nonzero().flatten().tolist()
        greedy_set = set(greedy_rows)
        if unseeded:
            idx = torch.tensor(unseeded, device=logits.device)
            sampled[idx] = torch.multinomial(
                probs[idx].clamp(min=0), 1
            ).squeeze(-1)
        for i, g in meta.generators.items():
            sampled[i] = torch.multinomial(probs[i].clamp(min=0), 1,
                                           generator=g).squeeze(-1)
        for i in greedy_set:
            sampled[i] = logits[i].argmax()
        return sampled
