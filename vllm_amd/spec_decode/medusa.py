"""Medusa-style multi-head draft proposer.

Role of the reference's vllm/model_executor/models/medusa.py +
v1/spec_decode medusa wiring: K extra heads hang off the target model's
last hidden state; head i predicts the token at offset i+2 (the normal
lm_head samples offset +1). Proposals are verified by the existing
greedy in-place verification with KV rollback (model_runner), so a bad
head only costs acceptance rate, never correctness.

Unlike EAGLE there is no autoregressive draft loop and no extra KV —
one matmul chain per head on [rows, hidden], which on MI355X is noise
next to the decode step itself.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class MedusaHeads(nn.Module):
    """K residual-MLP heads + per-head vocab projections.

    Head i: x = h + SiLU(W_i h + b_i); logits_i = V_i x  (the classic
    Medusa-1 ResBlock head shape)."""

    def __init__(self, hidden_size: int, vocab_size: int, k: int,
                 dtype: torch.dtype):
        super().__init__()
        self.k = k
        self.blocks = nn.ModuleList([
            nn.Linear(hidden_size, hidden_size, bias=True, dtype=dtype)
            for _ in range(k)
        ])
        self.lm_heads = nn.ModuleList([
            nn.Linear(hidden_size, vocab_size, bias=False, dtype=dtype)
            for _ in range(k)
        ])
        for p in self.parameters():
            p.requires_grad_(False)

    @torch.inference_mode()
    def propose(self, hidden: torch.Tensor) -> torch.Tensor:
        """hidden: [rows, H] at each row's last accepted position ->
        draft token ids [rows, k] (greedy argmax per head)."""
        toks = []
        for blk, lm in zip(self.blocks, self.lm_heads):
            x = hidden + F.silu(blk(hidden))
            toks.append(lm(x).argmax(dim=-1))
        return torch.stack(toks, dim=1)

    def init_dummy(self, seed: int) -> None:
        g = torch.Generator().manual_seed(seed ^ 0x6D647361)  # 'mdsa'
        for p in self.parameters():
            with torch.no_grad():
                cpu = torch.empty(p.shape, dtype=torch.float32).normal_(
                    0.0, 0.02, generator=g)
                p.copy_(cpu.to(p.dtype))

    def load_safetensors(self, path: str, dtype: torch.dtype) -> None:
        """Load a medusa-head checkpoint (HF medusa layout:
        `{i}.{j}.linear.weight/bias` res-blocks + `{i}.lm_head.weight`,
        one res-block deep)."""
        from vllm_amd.models.weight_loader import _iter_safetensors

        for name, w in _iter_safetensors(path):
            parts = name.split(".")
            i = int(parts[0])
            if i >= self.k:
                continue
            if "lm_head" in name:
                self.lm_heads[i].weight.data.copy_(w.to(dtype))
            elif name.endswith("linear.weight") or parts[-1] == "weight":
                self.blocks[i].weight.data.copy_(w.to(dtype))
            elif name.endswith("bias"):
                self.blocks[i].bias.data.copy_(w.to(dtype))


class MedusaProposer:
    """Scheduler-side marker: drafts come from the runner's heads (in
    ModelRunnerOutput.draft_token_ids), not from a CPU propose()."""

    model_based = True
