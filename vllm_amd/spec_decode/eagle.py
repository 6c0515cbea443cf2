"""EAGLE-1 draft proposer (role of the reference's
vllm/v1/spec_decode/eagle.py + vllm/model_executor/models/eagle.py).

One llama-style decoder layer autoregresses over FEATURES: the input at
sequence slot p is concat(embed(token_{p+1}), f_p) -> fc: 2H -> H ->
decoder layer -> g_p, and lm_head(g_p) predicts token_{p+2}. During
observation f_p is the TARGET model's hidden state (captured by the
model runner each step); during the k-step draft loop the draft feeds
its own g back as the feature. Verification is the existing greedy
in-place check with KV rollback (model_runner), so a bad draft costs
acceptance rate, never correctness.

The draft layer keeps its own per-request dense KV (one layer, a few
hundred KB per request at 4k context) — on MI355X a paged pool for a
single-layer draft is pointless bookkeeping next to 288 GB of HBM3E.
Speculative slots appended during the draft loop are truncated before
returning, and `observe` truncates on position regressions, so target
KV rollback and preemption re-runs stay consistent automatically.
"""

from __future__ import annotations

import math
from typing import Callable

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.layers.rotary import _compute_inv_freq


def _rms(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-6):
    dt = x.dtype
    x = x.float()
    x = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + eps)
    return (x * w.float()).to(dt)


class _ReqKV:
    __slots__ = ("k", "v", "len")

    def __init__(self, cap: int, n_kv: int, hd: int, dtype, device):
        self.k = torch.zeros(cap, n_kv, hd, dtype=dtype, device=device)
        self.v = torch.zeros(cap, n_kv, hd, dtype=dtype, device=device)
        self.len = 0


class EagleDraft(nn.Module):
    """Single llama decoder layer with a 2H->H input fc (EAGLE-1
    architecture: no input layernorm on the first/only layer)."""

    def __init__(self, hidden_size: int, num_heads: int,
                 num_kv_heads: int, intermediate_size: int,
                 dtype: torch.dtype, rope_theta: float = 10000.0):
        super().__init__()
        H = hidden_size
        self.h = H
        self.nh = num_heads
        self.nkv = num_kv_heads
        self.hd = H // num_heads
        self.fc = nn.Linear(2 * H, H, bias=True, dtype=dtype)
        self.q_proj = nn.Linear(H, H, bias=False, dtype=dtype)
        self.k_proj = nn.Linear(H, self.nkv * self.hd, bias=False,
                                dtype=dtype)
        self.v_proj = nn.Linear(H, self.nkv * self.hd, bias=False,
                                dtype=dtype)
        self.o_proj = nn.Linear(H, H, bias=False, dtype=dtype)
        self.gate_proj = nn.Linear(H, intermediate_size, bias=False,
                                   dtype=dtype)
        self.up_proj = nn.Linear(H, intermediate_size, bias=False,
                                 dtype=dtype)
        self.down_proj = nn.Linear(intermediate_size, H, bias=False,
                                   dtype=dtype)
        self.post_norm_w = nn.Parameter(torch.ones(H, dtype=dtype))
        inv = _compute_inv_freq(self.hd, rope_theta)
        self.register_buffer("inv_freq", inv, persistent=False)
        for p in self.parameters():
            p.requires_grad_(False)

    def _rope(self, x: torch.Tensor, pos: torch.Tensor) -> torch.Tensor:
        # x: [T, n, hd]; NeoX halves rotation, fp32 math.
        half = self.hd // 2
        freqs = pos.float().unsqueeze(1) * self.inv_freq.unsqueeze(0)
        cos = freqs.cos().unsqueeze(1)  # [T, 1, half]
        sin = freqs.sin().unsqueeze(1)
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * cos - x2 * sin,
                          x1 * sin + x2 * cos], dim=-1).to(x.dtype)

    @torch.inference_mode()
    def forward_chunk(self, inp: torch.Tensor, pos0: int,
                      kv: _ReqKV) -> torch.Tensor:
        """inp: [T, 2H] concat(embed(next_tok), feature) at slots
        pos0..pos0+T-1; appends to kv and returns g [T, H]."""
        T = inp.shape[0]
        pos = torch.arange(pos0, pos0 + T, device=inp.device)
        x = self.fc(inp)
        q = self._rope(self.q_proj(x).view(T, self.nh, self.hd), pos)
        k = self._rope(self.k_proj(x).view(T, self.nkv, self.hd), pos)
        v = self.v_proj(x).view(T, self.nkv, self.hd)
        kv.k[pos0:pos0 + T] = k
        kv.v[pos0:pos0 + T] = v
        kv.len = pos0 + T
        L = kv.len
        rep = self.nh // self.nkv
        kk = kv.k[:L].repeat_interleave(rep, dim=1)  # [L, nh, hd]
        vv = kv.v[:L].repeat_interleave(rep, dim=1)
        # [nh, T, L] causal scores (query slot pos0+i attends <= itself).
        scores = torch.einsum("tnd,lnd->ntl", q.float(), kk.float())
        scores *= 1.0 / math.sqrt(self.hd)
        qpos = pos.unsqueeze(1)  # [T, 1]
        kpos = torch.arange(L, device=inp.device).unsqueeze(0)
        scores.masked_fill_((kpos > qpos).unsqueeze(0), float("-inf"))
        attn = torch.softmax(scores, dim=-1)
        out = torch.einsum("ntl,lnd->tnd", attn, vv.float())
        x = x + self.o_proj(out.reshape(T, self.h).to(x.dtype))
        y = _rms(x, self.post_norm_w)
        x = x + self.down_proj(F.silu(self.gate_proj(y)) * self.up_proj(y))
        return x

    def init_dummy(self, seed: int) -> None:
        g = torch.Generator().manual_seed(seed ^ 0x65676C31)  # 'egl1'
        for p in self.parameters():
            with torch.no_grad():
                cpu = torch.empty(p.shape, dtype=torch.float32).normal_(
                    0.0, 0.02, generator=g)
                p.copy_(cpu.to(p.dtype))

    def load_safetensors(self, path: str, dtype: torch.dtype) -> None:
        """HF EAGLE-llama layout: fc.{weight,bias},
        layers.0.self_attn.{q,k,v,o}_proj.weight,
        layers.0.mlp.{gate,up,down}_proj.weight,
        layers.0.post_attention_layernorm.weight."""
        from vllm_amd.models.weight_loader import _iter_safetensors

        dest = {
            "fc.weight": self.fc.weight, "fc.bias": self.fc.bias,
            "layers.0.self_attn.q_proj.weight": self.q_proj.weight,
            "layers.0.self_attn.k_proj.weight": self.k_proj.weight,
            "layers.0.self_attn.v_proj.weight": self.v_proj.weight,
            "layers.0.self_attn.o_proj.weight": self.o_proj.weight,
            "layers.0.mlp.gate_proj.weight": self.gate_proj.weight,
            "layers.0.mlp.up_proj.weight": self.up_proj.weight,
            "layers.0.mlp.down_proj.weight": self.down_proj.weight,
            "layers.0.post_attention_layernorm.weight": self.post_norm_w,
        }
        for name, w in _iter_safetensors(path):
            name = name.removeprefix("model.")
            if name in dest:
                dest[name].data.copy_(w.to(dtype))


class EagleRunnerSide:
    """Runner-side EAGLE state: per-request draft KV + the observe /
    propose protocol driven by model_runner after each step."""

    def __init__(self, draft: EagleDraft,
                 embed: Callable[[torch.Tensor], torch.Tensor],
                 compute_logits: Callable[[torch.Tensor], torch.Tensor],
                 k: int, max_len: int, dtype, device):
        self.draft = draft
        self.embed = embed
        self.compute_logits = compute_logits
        self.k = k
        self.max_len = max_len
        self.dtype = dtype
        self.device = device
        self._kv: dict[str, _ReqKV] = {}
        self._last_g: dict[str, torch.Tensor] = {}

    def free(self, rid: str) -> None:
        self._kv.pop(rid, None)
        self._last_g.pop(rid, None)

    def _state(self, rid: str) -> _ReqKV:
        st = self._kv.get(rid)
        if st is None:
            st = _ReqKV(self.max_len + self.k + 1,
                        self.draft.nkv, self.draft.hd,
                        self.dtype, self.device)
            self._kv[rid] = st
        return st

    @torch.inference_mode()
    def observe(self, rid: str, start: int, feats: torch.Tensor,
                next_tokens: list[int]) -> None:
        """feats: target hiddens at positions start..start+T-1;
        next_tokens[j] = token at position start+j+1. Draft slot p gets
        input concat(embed(token_{p+1}), f_p)."""
        st = self._state(rid)
        if start > st.len:
            # Draft never saw some positions (shouldn't happen: every
            # computed position is observed) — drop stale state.
            st.len = 0
            start = 0
            return
        toks = torch.tensor(next_tokens, device=self.device)
        inp = torch.cat(
            [self.embed(toks).to(self.dtype),
             feats.to(self.dtype)], dim=-1)
        g = self.draft.forward_chunk(inp, start, st)
        self._last_g[rid] = g[-1]

    @torch.inference_mode()
    def propose(self, rid: str, last_token: int) -> list[int]:
        """k greedy draft tokens continuing after `last_token` (the
        newest accepted token). Speculative KV slots are truncated."""
        st = self._kv.get(rid)
        g = self._last_g.get(rid)
        if st is None or g is None or st.len == 0:
            return []
        base_len = st.len
        drafts: list[int] = []
        tok = last_token
        for _ in range(self.k):
            e = self.embed(
                torch.tensor([tok], device=self.device)).to(self.dtype)
            inp = torch.cat([e, g.unsqueeze(0)], dim=-1)
            g = self.draft.forward_chunk(inp, st.len, st)[-1]
            tok = int(self.compute_logits(g.unsqueeze(0)).argmax())
            drafts.append(tok)
        st.len = base_len  # roll back speculative slots
        return drafts


class EagleProposer:
    """Scheduler-side marker (like MedusaProposer): drafts come from the
    runner (ModelRunnerOutput.draft_token_ids)."""

    model_based = True
