"""Pure Mamba (selective-state-space) causal LM.

Role of the reference's vllm/model_executor/models/mamba.py
(MambaForCausalLM) + layers/mamba/ mixer; state management is the
constant-size analogue of paged KV (reference MambaSpec,
kv_cache_interface.py:710 / MambaManager, single_type_kv_cache_manager
.py:1253): the runner owns per-layer state tensors indexed by each
request's persistent batch row —
  conv state [rows, d_inner, d_conv-1]  (causal-conv lookback window)
  ssm  state [rows, d_inner, d_state]   (recurrent hidden state, fp32)
Rows are zeroed when a request is (re)admitted; prefix caching is
disabled for SSM models (engine/arg_utils.py) because state is not
content-addressable, so every (re)start scans from position 0 and
chunked prefill carries state across chunks naturally.

Compute is torch-native: the within-chunk scan is sequential over time
(vectorized over channels/state), decode is the O(1) recurrence
vectorized over the whole decode batch. A fused CDNA4 scan kernel is a
tracked follow-up; the op is elementwise-bound, so eager torch on ROCm
is already HBM-bound for decode.

Tensor parallelism: mixer weights are replicated and every rank computes
the full mixer (the residual stream is replicated between blocks in this
engine, so this is correct with zero collectives; sharding d_inner needs
the reference's split-B/C treatment and is not worth it before a fused
kernel exists). Pipeline parallelism slices layers as in llama.py.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import RMSNorm
from vllm_amd.parallel.state import (
    is_first_pp_rank,
    is_last_pp_rank,
    pp_layer_range,
)
from vllm_amd.worker.forward_context import get_forward_context


class MambaMixer(nn.Module):
    """Mamba-1 selective SSM mixer (in_proj -> causal conv -> selective
    scan -> gated out_proj)."""

    def __init__(self, spec: ModelSpec, cache_idx: int,
                 dtype: torch.dtype):
        super().__init__()
        hidden = spec.hidden_size
        self.d_inner = spec.mamba_expand * hidden
        self.d_state = spec.mamba_d_state
        self.d_conv = spec.mamba_d_conv
        self.dt_rank = spec.mamba_dt_rank or -(-hidden // 16)
        self.cache_idx = cache_idx
        self.in_proj = nn.Linear(hidden, 2 * self.d_inner, bias=False,
                                 dtype=dtype)
        # Depthwise causal conv stored as [d_inner, d_conv] (HF conv1d
        # weight is [d_inner, 1, d_conv]).
        self.conv_weight = nn.Parameter(
            torch.zeros(self.d_inner, self.d_conv, dtype=dtype))
        self.conv_bias = nn.Parameter(
            torch.zeros(self.d_inner, dtype=dtype))
        self.x_proj = nn.Linear(self.d_inner,
                                self.dt_rank + 2 * self.d_state,
                                bias=False, dtype=dtype)
        self.dt_proj = nn.Linear(self.dt_rank, self.d_inner, bias=True,
                                 dtype=dtype)
        # A kept as log (A = -exp(A_log) < 0 keeps the recurrence
        # contractive); state math runs in fp32 regardless of model dtype.
        self.A_log = nn.Parameter(
            torch.zeros(self.d_inner, self.d_state, dtype=torch.float32))
        self.D = nn.Parameter(torch.ones(self.d_inner,
                                         dtype=torch.float32))
        self.out_proj = nn.Linear(self.d_inner, hidden, bias=False,
                                  dtype=dtype)

    def _conv_decode(self, x: torch.Tensor, rows: torch.Tensor,
                     conv_cache: torch.Tensor) -> torch.Tensor:
        """One-token depthwise conv for the decode batch; advances the
        per-row lookback window in place."""
        window = torch.cat([conv_cache[rows], x.unsqueeze(-1)], dim=-1)
        conv_cache[rows] = window[..., 1:]
        out = (window * self.conv_weight).sum(-1) + self.conv_bias
        return F.silu(out)

    def _ssm(self, xc: torch.Tensor, h: torch.Tensor,
             step: bool) -> tuple[torch.Tensor, torch.Tensor]:
        """Selective scan. xc: [L, d_inner] (or [B, d_inner] when
        step=True, one token per batch row); h: matching fp32 state
        [*, d_inner, d_state]. Returns (y, h_out)."""
        dbc = self.x_proj(xc)
        dt = F.softplus(self.dt_proj(dbc[:, :self.dt_rank]).float())
        b = dbc[:, self.dt_rank:self.dt_rank + self.d_state].float()
        c = dbc[:, -self.d_state:].float()
        a = -torch.exp(self.A_log)  # [d_inner, d_state]
        xf = xc.float()
        da = torch.exp(dt.unsqueeze(-1) * a)        # [*, d_inner, ds]
        dbx = dt.unsqueeze(-1) * b.unsqueeze(1) * xf.unsqueeze(-1)
        if step:
            h = h * da + dbx
            y = (h * c.unsqueeze(1)).sum(-1) + self.D * xf
            return y, h
        ys = []
        for t in range(xc.shape[0]):
            h = h * da[t] + dbx[t]
            ys.append((h * c[t].unsqueeze(0)).sum(-1))
        y = torch.stack(ys) + self.D * xf
        return y, h

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        ctx = get_forward_context()
        meta = ctx.attn_metadata
        states = ctx.mamba_states
        if states is not None:
            conv_cache = states[0][self.cache_idx]
            ssm_cache = states[1][self.cache_idx]
            rows = meta.state_rows
        else:
            # Memory-profiling dummy forward: transient zero state.
            conv_cache = x.new_zeros(meta.num_reqs, self.d_inner,
                                     self.d_conv - 1)
            ssm_cache = torch.zeros(meta.num_reqs, self.d_inner,
                                    self.d_state, dtype=torch.float32,
                                    device=x.device)
            rows = torch.arange(meta.num_reqs, device=x.device)

        xs, z = self.in_proj(x).chunk(2, dim=-1)
        y = torch.empty_like(xs, dtype=torch.float32)
        nd = meta.num_decodes
        if nd:
            r = rows[:nd]
            xc = self._conv_decode(xs[:nd], r, conv_cache)
            yd, h = self._ssm(xc, ssm_cache[r], step=True)
            ssm_cache[r] = h
            y[:nd] = yd
        if nd < meta.num_reqs:
            qsl = meta.query_start_loc.tolist()
            for i in range(nd, meta.num_reqs):
                s, e = qsl[i], qsl[i + 1]
                if s == e:
                    continue
                r = int(rows[i])
                seq_t = xs[s:e].t()  # [d_inner, L]
                ext = torch.cat([conv_cache[r], seq_t], dim=-1)
                conv_cache[r] = ext[:, ext.shape[-1] - (self.d_conv - 1):]
                win = ext.unfold(-1, self.d_conv, 1)  # [d_inner, L, dc]
                xc = F.silu(
                    (win * self.conv_weight.unsqueeze(1)).sum(-1)
                    + self.conv_bias.unsqueeze(-1)).t().contiguous()
                yp, h = self._ssm(xc, ssm_cache[r], step=False)
                ssm_cache[r] = h
                y[s:e] = yp
        return self.out_proj((y * F.silu(z.float())).to(x.dtype))


class MambaDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int,
                 dtype: torch.dtype, cache_idx: int):
        super().__init__()
        self.norm = RMSNorm(spec.hidden_size, spec.rms_norm_eps,
                            dtype=dtype)
        self.mixer = MambaMixer(spec, cache_idx, dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        return hidden + self.mixer(self.norm(hidden))


class MambaModel(nn.Module):
    """Same PP layout rules as llama.py LlamaModel: layer slice per
    stage, embedding first, final norm last; nn.Identity placeholders
    keep global layer indices in parameter names."""

    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.lo, self.hi = pp_layer_range(spec.num_layers)
        self.embed_tokens = (
            VocabParallelEmbedding(spec.vocab_size, spec.hidden_size,
                                   dtype=dtype)
            if is_first_pp_rank() else None
        )
        self.layers = nn.ModuleList([
            MambaDecoderLayer(spec, i, dtype, cache_idx=i - self.lo)
            if self.lo <= i < self.hi else nn.Identity()
            for i in range(spec.num_layers)
        ])
        self.norm_f = (RMSNorm(spec.hidden_size, spec.rms_norm_eps,
                               dtype=dtype)
                       if is_last_pp_rank() else None)

    def forward(self, input_ids, positions, hidden_in=None):
        if self.embed_tokens is not None:
            hidden = self.embed_tokens(input_ids)
        else:
            hidden = hidden_in
        for layer in self.layers[self.lo:self.hi]:
            hidden = layer(hidden)
        if self.norm_f is not None:
            hidden = self.norm_f(hidden)
        return hidden


class MambaForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = MambaModel(config)
        self.lm_head = (
            ParallelLMHead(spec.vocab_size, spec.hidden_size,
                           dtype=config.torch_dtype)
            if is_last_pp_rank() else None
        )
        if spec.tie_word_embeddings and self.lm_head is not None:
            if self.model.embed_tokens is None:
                raise ValueError(
                    "tie_word_embeddings requires the embedding and the "
                    "lm_head on the same PP stage (pp=1)")
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, positions, hidden_in=None):
        return self.model(input_ids, positions, hidden_in)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)

    def load_hf_weights(self, tensors, config: ModelConfig) -> None:
        """HF MambaForCausalLM checkpoint layout: backbone.embeddings,
        backbone.layers.N.{norm,mixer.*}, backbone.norm_f; lm_head tied."""
        spec = config.spec
        for name, w in tensors:
            if name == "backbone.embeddings.weight":
                self.model.embed_tokens.load_weight(w)
                continue
            if name == "backbone.norm_f.weight":
                self.model.norm_f.weight.data.copy_(w)
                continue
            if name == "lm_head.weight":
                if not spec.tie_word_embeddings:
                    self.lm_head.load_weight(w)
                continue
            if not name.startswith("backbone.layers."):
                continue
            parts = name.split(".")
            layer = self.model.layers[int(parts[2])]
            rest = ".".join(parts[3:])
            mixer = layer.mixer
            if rest == "norm.weight":
                layer.norm.weight.data.copy_(w)
            elif rest == "mixer.conv1d.weight":
                mixer.conv_weight.data.copy_(w.squeeze(1))
            elif rest == "mixer.conv1d.bias":
                mixer.conv_bias.data.copy_(w)
            elif rest == "mixer.A_log":
                mixer.A_log.data.copy_(w.float())
            elif rest == "mixer.D":
                mixer.D.data.copy_(w.float())
            elif rest.startswith("mixer."):
                mod = mixer
                path = rest.split(".")[1:]
                for p in path[:-1]:
                    mod = getattr(mod, p)
                getattr(mod, path[-1]).data.copy_(w)
