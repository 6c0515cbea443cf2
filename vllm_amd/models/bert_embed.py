"""BERT-style bidirectional encoder for embedding serving.

Role of the reference's vllm/model_executor/models/bert.py
(BertEmbeddingModel behind /v1/embeddings): a pooling-only model — no
decode, no KV cache at all. Each request's prompt is encoded in ONE
forward pass (bidirectional attention cannot span prefill chunks, so
chunked prefill and prefix caching are force-disabled in
engine/arg_utils.py and over-budget prompts are rejected up front), and
the engine's existing pooling path (last/mean) reads the hidden states.

Attention is plain torch sdpa per request segment (no causal mask, no
paged cache); projections are TP-sharded through the same
QKV/RowParallel layers as the decoder families. Post-norm residuals and
GELU FFN per the original BERT encoder block.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.embedding import VocabParallelEmbedding
from vllm_amd.layers.layernorm import LayerNorm
from vllm_amd.layers.linear import (
    ColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)
from vllm_amd.worker.forward_context import get_forward_context


class BertSelfAttention(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.head_dim, spec.num_heads,
            spec.num_kv_heads, bias=True, dtype=dtype)
        self.num_heads = self.qkv_proj.num_heads
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=True, dtype=dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        meta = get_forward_context().attn_metadata
        t = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        q = q.view(t, self.num_heads, self.head_dim)
        k = k.view(t, self.num_heads, self.head_dim)
        v = v.view(t, self.num_heads, self.head_dim)
        out = torch.empty_like(q)
        qsl = meta.query_start_loc.tolist()
        for i in range(meta.num_reqs):
            s, e = int(qsl[i]), int(qsl[i + 1])
            if s == e:
                continue
            # [heads, L, hd]; full bidirectional within the segment.
            out[s:e] = F.scaled_dot_product_attention(
                q[s:e].transpose(0, 1), k[s:e].transpose(0, 1),
                v[s:e].transpose(0, 1)).transpose(0, 1)
        return self.o_proj(out.reshape(t, -1))


class BertEncoderLayer(nn.Module):
    """Post-norm encoder block: LN(x + attn(x)), LN(x + ffn(x))."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        self.attention = BertSelfAttention(spec, dtype)
        self.attn_norm = LayerNorm(spec.hidden_size, spec.rms_norm_eps,
                                   dtype=dtype)
        self.up_proj = ColumnParallelLinear(
            spec.hidden_size, spec.intermediate_size, bias=True,
            dtype=dtype)
        self.down_proj = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, bias=True,
            dtype=dtype)
        self.ffn_norm = LayerNorm(spec.hidden_size, spec.rms_norm_eps,
                                  dtype=dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        hidden = self.attn_norm(hidden + self.attention(hidden))
        ff = self.down_proj(F.gelu(self.up_proj(hidden)))
        return self.ffn_norm(hidden + ff)


class BertEmbeddingModel(nn.Module):
    """Pooling-only: forward returns hidden states; there is no lm_head
    and compute_logits must never be called (the engine rejects
    generation requests for pooling-only architectures)."""

    pooling_only = True

    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.embed_positions = nn.Embedding(
            spec.max_position_embeddings, spec.hidden_size)
        self.embed_positions.weight.requires_grad = False
        self.embed_positions.to(dtype)
        self.embed_norm = LayerNorm(spec.hidden_size, spec.rms_norm_eps,
                                    dtype=dtype)
        self.layers = nn.ModuleList([
            BertEncoderLayer(spec, dtype) for _ in range(spec.num_layers)
        ])

    def forward(self, input_ids, positions, hidden_in=None):
        hidden = self.embed_tokens(input_ids)
        hidden = hidden + self.embed_positions(positions)
        hidden = self.embed_norm(hidden)
        for layer in self.layers:
            hidden = layer(hidden)
        return hidden

    def compute_logits(self, hidden):
        raise RuntimeError(
            "bert embedding models are pooling-only; generation "
            "requests are rejected at admission")
