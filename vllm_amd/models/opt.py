"""OPT decoder (OPT-125m CPU plumbing config; reference models/opt.py).

Learned positional embeddings (offset +2), pre-LayerNorm, ReLU MLP with
biases, no RoPE, tied LM head.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.config import ModelConfig
from vllm_amd.layers.attention import Attention
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import LayerNorm
from vllm_amd.layers.linear import (
    ColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)


class OPTAttention(nn.Module):
    def __init__(self, spec, layer_idx: int, dtype):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.head_dim, spec.num_heads,
            spec.num_kv_heads, bias=True, dtype=dtype,
        )
        self.out_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size, bias=True,
            dtype=dtype,
        )
        self.attn = Attention(
            self.qkv_proj.num_heads,
            spec.head_dim,
            scale=spec.head_dim**-0.5,
            num_kv_heads=self.qkv_proj.num_kv_heads,
            layer_idx=layer_idx,
        )

    def forward(self, hidden):
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        out = self.attn(q, k, v)
        return self.out_proj(out)


class OPTDecoderLayer(nn.Module):
    def __init__(self, spec, layer_idx, dtype):
        super().__init__()
        self.self_attn = OPTAttention(spec, layer_idx, dtype)
        self.self_attn_layer_norm = LayerNorm(spec.hidden_size, dtype=dtype)
        self.fc1 = ColumnParallelLinear(
            spec.hidden_size, spec.intermediate_size, bias=True, dtype=dtype
        )
        self.fc2 = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, bias=True, dtype=dtype
        )
        self.final_layer_norm = LayerNorm(spec.hidden_size, dtype=dtype)

    def forward(self, hidden):
        residual = hidden
        hidden = self.self_attn_layer_norm(hidden)
        hidden = self.self_attn(hidden)
        hidden = residual + hidden
        residual = hidden
        hidden = self.final_layer_norm(hidden)
        hidden = F.relu(self.fc1(hidden))
        hidden = self.fc2(hidden)
        return residual + hidden


class OPTModel(nn.Module):
    POS_OFFSET = 2

    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype
        )
        self.embed_positions = nn.Embedding(
            spec.max_position_embeddings + self.POS_OFFSET,
            spec.hidden_size,
        )
        self.embed_positions.weight.requires_grad = False
        self.embed_positions.to(dtype)
        self.layers = nn.ModuleList(
            [
                OPTDecoderLayer(spec, i, dtype)
                for i in range(spec.num_layers)
            ]
        )
        self.final_layer_norm = LayerNorm(spec.hidden_size, dtype=dtype)

    def forward(self, input_ids, positions):
        hidden = self.embed_tokens(input_ids)
        hidden = hidden + self.embed_positions(positions + self.POS_OFFSET)
        for layer in self.layers:
            hidden = layer(hidden)
        return self.final_layer_norm(hidden)


class OPTForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = OPTModel(config)
        self.lm_head = ParallelLMHead(
            spec.vocab_size, spec.hidden_size, dtype=config.torch_dtype
        )
        if spec.tie_word_embeddings:
            self.lm_head.weight = self.model.embed_tokens.weight

    def forward(self, input_ids, positions):
        return self.model(input_ids, positions)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
