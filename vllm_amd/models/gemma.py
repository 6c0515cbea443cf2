"""Gemma3-family decoder.

Role of the reference's gemma3.py (vllm/model_executor/models/gemma3.py):
sandwich norms (post-attn / post-MLP norms before the residual add),
GeGLU MLP, sqrt(hidden)-scaled embeddings, (1+w) RMSNorm gains
(folded into the stored weight at load time — see weight_loader),
per-head q/k RMSNorm, and the 5-local:1-global sliding-window layer
pattern with a separate rope theta for local layers.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.activation import GeluAndMul
from vllm_amd.layers.attention import Attention
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import RMSNorm
from vllm_amd.layers.linear import (
    MergedColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)
from vllm_amd.layers.rotary import RotaryEmbedding


class GemmaMLP(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        self.gate_up_proj = MergedColumnParallelLinear(
            spec.hidden_size,
            [spec.intermediate_size, spec.intermediate_size],
            bias=False,
            dtype=dtype,
        )
        self.down_proj = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, bias=False, dtype=dtype
        )
        self.act_fn = GeluAndMul()

    def forward(self, x):
        return self.down_proj(self.act_fn(self.gate_up_proj(x)))


class GemmaAttention(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.head_dim, spec.num_heads,
            spec.num_kv_heads, bias=False, dtype=dtype,
        )
        self.num_heads = self.qkv_proj.num_heads
        self.num_kv_heads = self.qkv_proj.num_kv_heads
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=False, dtype=dtype,
        )
        every = spec.global_attn_every_n_layers
        self.is_global = (not spec.sliding_window) or (
            every and (layer_idx + 1) % every == 0)
        theta = spec.rope_theta if self.is_global else (
            spec.rope_local_theta or spec.rope_theta)
        self.rotary_emb = RotaryEmbedding(
            spec.head_dim, spec.head_dim, max_position, theta=theta,
            rope_scaling=spec.rope_scaling if self.is_global else None,
        )
        self.q_norm = RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype=dtype)
        self.k_norm = RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype=dtype)
        scalar = spec.query_pre_attn_scalar or spec.head_dim
        self.attn = Attention(
            self.num_heads, spec.head_dim, scale=scalar**-0.5,
            num_kv_heads=self.num_kv_heads, layer_idx=layer_idx,
            sliding_window=0 if self.is_global else spec.sliding_window,
            kv_group="full" if self.is_global else "window",
        )

    def forward(self, positions, hidden):
        T = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        # contiguous(): reshape of a strided head slice can stay a view
        # (num_kv_heads=1) and the HIP rms_norm requires contiguous rows.
        q = self.q_norm(
            q.reshape(-1, self.head_dim).contiguous()).view(q.shape)
        k = self.k_norm(
            k.reshape(-1, self.head_dim).contiguous()).view(k.shape)
        self.rotary_emb(positions, q, k)
        return self.o_proj(self.attn(q, k, v))


class GemmaDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position):
        super().__init__()
        self.self_attn = GemmaAttention(spec, layer_idx, dtype, max_position)
        self.mlp = GemmaMLP(spec, dtype)
        eps = spec.rms_norm_eps
        self.input_layernorm = RMSNorm(spec.hidden_size, eps, dtype=dtype)
        self.post_attention_layernorm = RMSNorm(spec.hidden_size, eps,
                                                dtype=dtype)
        self.pre_feedforward_layernorm = RMSNorm(spec.hidden_size, eps,
                                                 dtype=dtype)
        self.post_feedforward_layernorm = RMSNorm(spec.hidden_size, eps,
                                                  dtype=dtype)

    def forward(self, positions, hidden):
        res = hidden
        h = self.input_layernorm(hidden)
        h = self.self_attn(positions, h)
        h = self.post_attention_layernorm(h)
        hidden = res + h
        res = hidden
        h = self.pre_feedforward_layernorm(hidden)
        h = self.mlp(h)
        h = self.post_feedforward_layernorm(h)
        return res + h


class GemmaModel(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.spec = spec
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype
        )
        self.layers = nn.ModuleList(
            [GemmaDecoderLayer(spec, i, dtype, config.max_model_len)
             for i in range(spec.num_layers)]
        )
        self.norm = RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype=dtype)
        # sqrt(hidden) embedding normalizer, stored in the model dtype the
        # way HF does (affects numerics in bf16).
        self.register_buffer(
            "embed_scale",
            torch.tensor(spec.hidden_size**0.5, dtype=dtype),
            persistent=False,
        )

    def forward(self, input_ids, positions):
        hidden = self.embed_tokens(input_ids)
        if self.spec.scale_embeddings:
            hidden = hidden * self.embed_scale
        for layer in self.layers:
            hidden = layer(positions, hidden)
        return self.norm(hidden)


class GemmaForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = GemmaModel(config)
        self.lm_head = ParallelLMHead(
            spec.vocab_size, spec.hidden_size, dtype=config.torch_dtype
        )
        if spec.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, positions):
        return self.model(input_ids, positions)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
