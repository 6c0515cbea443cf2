"""Jamba-style hybrid attention+SSM causal LM.

Role of the reference's vllm/model_executor/models/jamba.py: most layers
are Mamba mixers (constant-size recurrent state), every
``attn_layer_period``-th layer (at ``attn_layer_offset``) is standard
GQA attention with paged KV and NO positional encoding (Jamba is NoPE —
the SSM layers carry position). This is the hybrid-KV-coordinator
story's third group kind: paged full-attention KV and SSM state rows
coexist in one model (reference HybridKVCacheCoordinator with
FullAttentionManager + MambaManager, kv_cache_coordinator.py:521).

Each layer is pre-norm mixer/attention + pre-norm gated MLP (the
reference Jamba alternates dense/MoE MLPs; this implementation is
all-dense — MoE Jamba variants plug into layers/fused_moe.py the same
way mixtral.py does, tracked follow-up). Checkpoint loading: dummy init
(partition-invariant) is the primary path, as for the other families.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.attention import Attention
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import RMSNorm
from vllm_amd.layers.linear import QKVParallelLinear, RowParallelLinear
from vllm_amd.models.llama import LlamaMLP
from vllm_amd.models.mamba import MambaMixer
from vllm_amd.parallel.state import (
    is_first_pp_rank,
    is_last_pp_rank,
    pp_layer_range,
)


class JambaAttention(nn.Module):
    """GQA attention without rotary embeddings (NoPE)."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype,
                 kv_cache_idx: int):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.head_dim, spec.num_heads,
            spec.num_kv_heads, bias=False, dtype=dtype)
        self.num_heads = self.qkv_proj.num_heads
        self.num_kv_heads = self.qkv_proj.num_kv_heads
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=False, dtype=dtype)
        self.attn = Attention(
            self.num_heads, spec.head_dim, scale=spec.head_dim**-0.5,
            num_kv_heads=self.num_kv_heads, layer_idx=kv_cache_idx)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        t = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        out = self.attn(q.view(t, self.num_heads, self.head_dim),
                        k.view(t, self.num_kv_heads, self.head_dim), v)
        return self.o_proj(out)


class JambaDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int,
                 dtype: torch.dtype, kv_cache_idx: int,
                 mamba_cache_idx: int):
        super().__init__()
        self.is_attn = spec.is_attn_layer(layer_idx)
        self.input_layernorm = RMSNorm(spec.hidden_size,
                                       spec.rms_norm_eps, dtype=dtype)
        if self.is_attn:
            self.self_attn = JambaAttention(spec, dtype, kv_cache_idx)
        else:
            self.mamba = MambaMixer(spec, mamba_cache_idx, dtype)
        self.pre_ff_layernorm = RMSNorm(spec.hidden_size,
                                        spec.rms_norm_eps, dtype=dtype)
        self.feed_forward = LlamaMLP(spec, dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        inner = (self.self_attn if self.is_attn else self.mamba)
        hidden = hidden + inner(self.input_layernorm(hidden))
        return hidden + self.feed_forward(self.pre_ff_layernorm(hidden))


class JambaModel(nn.Module):
    """PP layout as in llama.py; kv/mamba cache indices are each layer's
    ordinal among its KIND within this stage's local slice (the runner
    allocates exactly that many paged-KV layers / state layers)."""

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
        layers = []
        n_kv = n_mamba = 0
        for i in range(spec.num_layers):
            if not (self.lo <= i < self.hi):
                layers.append(nn.Identity())
                continue
            layers.append(JambaDecoderLayer(
                spec, i, dtype, kv_cache_idx=n_kv,
                mamba_cache_idx=n_mamba))
            if spec.is_attn_layer(i):
                n_kv += 1
            else:
                n_mamba += 1
        self.layers = nn.ModuleList(layers)
        self.final_layernorm = (
            RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype=dtype)
            if is_last_pp_rank() else None
        )

    def forward(self, input_ids, positions, hidden_in=None):
        hidden = (self.embed_tokens(input_ids)
                  if self.embed_tokens is not None else hidden_in)
        for layer in self.layers[self.lo:self.hi]:
            hidden = layer(hidden)
        if self.final_layernorm is not None:
            hidden = self.final_layernorm(hidden)
        return hidden


class JambaForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = JambaModel(config)
        self.lm_head = (
            ParallelLMHead(spec.vocab_size, spec.hidden_size,
                           dtype=config.torch_dtype)
            if is_last_pp_rank() else None
        )
        if spec.tie_word_embeddings and self.lm_head is not None:
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, positions, hidden_in=None):
        return self.model(input_ids, positions, hidden_in)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
