"""Mixtral (sparse MoE decoder; BASELINE config 4: Mixtral 8x7B).

Structure mirrors the reference's mixtral.py (vllm/model_executor/models/
mixtral.py) — Llama-style GQA attention + per-layer top-2-of-8 FusedMoE
block replacing the dense MLP; two TP all-reduces per layer.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.attention import Attention
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.fused_moe import FusedMoE
from vllm_amd.layers.layernorm import RMSNorm
from vllm_amd.layers.linear import QKVParallelLinear, RowParallelLinear
from vllm_amd.layers.rotary import RotaryEmbedding


class MixtralAttention(nn.Module):
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
            spec.num_heads * spec.head_dim, spec.hidden_size, bias=False,
            dtype=dtype,
        )
        self.rotary_emb = RotaryEmbedding(
            spec.head_dim, spec.head_dim, max_position, theta=spec.rope_theta,
            rope_scaling=spec.rope_scaling,
        )
        # Qwen3-MoE: per-head RMSNorm on q/k pre-RoPE (same knob as the
        # dense Qwen3 trunk).
        self.q_norm = (RMSNorm(spec.head_dim, spec.rms_norm_eps,
                               dtype=dtype) if spec.qk_norm else None)
        self.k_norm = (RMSNorm(spec.head_dim, spec.rms_norm_eps,
                               dtype=dtype) if spec.qk_norm else None)
        self.attn = Attention(
            self.num_heads, spec.head_dim, scale=spec.head_dim**-0.5,
            num_kv_heads=self.num_kv_heads, layer_idx=layer_idx,
        )

    def forward(self, positions, hidden):
        T = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        if self.q_norm is not None:
            q = self.q_norm(
                q.reshape(-1, self.head_dim).contiguous()).view(q.shape)
            k = self.k_norm(
                k.reshape(-1, self.head_dim).contiguous()).view(k.shape)
        self.rotary_emb(positions, q, k)
        out = self.attn(q, k, v)
        return self.o_proj(out)


class MixtralDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position,
                 ep: bool = False, eplb_window: int = 0):
        super().__init__()
        self.self_attn = MixtralAttention(spec, layer_idx, dtype,
                                          max_position)
        self.block_sparse_moe = FusedMoE(
            num_experts=spec.num_experts,
            top_k=spec.num_experts_per_tok,
            hidden_size=spec.hidden_size,
            intermediate_size=spec.moe_intermediate_size,
            renormalize=spec.norm_topk_prob,
            dtype=dtype,
            enable_expert_parallel=ep,
            eplb_window=eplb_window,
        )
        self.input_layernorm = RMSNorm(spec.hidden_size, spec.rms_norm_eps,
                                       dtype=dtype)
        self.post_attention_layernorm = RMSNorm(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype
        )

    def forward(self, positions, hidden, residual):
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        hidden = self.self_attn(positions, hidden)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        hidden = self.block_sparse_moe(hidden)
        return hidden, residual


class MixtralModel(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype
        )
        self.layers = nn.ModuleList([
            MixtralDecoderLayer(spec, i, dtype, config.max_model_len,
                                ep=config.enable_expert_parallel,
                                eplb_window=config.eplb_window)
            for i in range(spec.num_layers)
        ])
        self.norm = RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype=dtype)

    def forward(self, input_ids, positions):
        hidden = self.embed_tokens(input_ids)
        residual = None
        for layer in self.layers:
            hidden, residual = layer(positions, hidden, residual)
        hidden, _ = self.norm(hidden, residual)
        return hidden


class MixtralForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = MixtralModel(config)
        self.lm_head = ParallelLMHead(
            spec.vocab_size, spec.hidden_size, dtype=config.torch_dtype
        )
        if spec.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, positions):
        return self.model(input_ids, positions)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
