"""Whisper-style speech-to-text model (decoder half).

Role of the reference's vllm/model_executor/models/whisper.py: an
encoder-decoder where the audio encoder (vllm_amd/audio.py, run once
per request by the model runner) produces cross-attention states and
this decoder generates text over them — causal self-attention uses the
normal paged KV cache; cross-attention reads the request's cached
encoder states from ForwardContext.cross_feats (the CrossAttention
cache-group role, reference single_type_kv_cache_manager.py:1747,
realized as per-request constant encoder states rather than paged
blocks — they never grow during decode).

Decoder block (pre-norm): self-attn -> cross-attn -> GELU MLP.
Learned decoder positions (OPT-style, no offset). TP shards the
projections; the encoder itself is replicated (runs once per request).
PP is not supported for encoder-decoder this round (the encoder states
would need to ship between stages) — single-stage only, enforced at
construction.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.attention import Attention
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import LayerNorm
from vllm_amd.layers.linear import (
    ColumnParallelLinear,
    MergedColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)
from vllm_amd.parallel.state import get_pp_world_size
from vllm_amd.worker.forward_context import get_forward_context


class WhisperSelfAttention(nn.Module):
    """Causal GQA self-attention over paged KV (no rope — Whisper uses
    learned absolute positions added at the embedding)."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype,
                 layer_idx: int):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size, spec.head_dim, spec.num_heads,
            spec.num_kv_heads, bias=True, dtype=dtype)
        self.num_heads = self.qkv_proj.num_heads
        self.num_kv_heads = self.qkv_proj.num_kv_heads
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=True, dtype=dtype)
        self.attn = Attention(
            self.num_heads, spec.head_dim, scale=spec.head_dim**-0.5,
            num_kv_heads=self.num_kv_heads, layer_idx=layer_idx)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        t = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        out = self.attn(q.view(t, self.num_heads, self.head_dim),
                        k.view(t, self.num_kv_heads, self.head_dim), v)
        return self.o_proj(out)


class WhisperCrossAttention(nn.Module):
    """Full (non-causal) attention from decoder tokens to the request's
    cached audio-encoder states."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        self.q_proj = ColumnParallelLinear(
            spec.hidden_size, spec.num_heads * spec.head_dim, bias=True,
            dtype=dtype)
        # Merged column-parallel so each rank's shard holds ITS OWN
        # K and V slices (a single 2N column split would give rank 0
        # only K columns and rank 1 only V columns under tp=2).
        n = spec.num_heads * spec.head_dim
        self.kv_proj = MergedColumnParallelLinear(
            spec.hidden_size, [n, n], bias=False, dtype=dtype)
        from vllm_amd.parallel.state import get_tp_world_size

        self.num_heads = spec.num_heads // max(1, get_tp_world_size())
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=True, dtype=dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        ctx = get_forward_context()
        meta = ctx.attn_metadata
        feats = ctx.cross_feats
        t = hidden.shape[0]
        q = self.q_proj(hidden).view(t, self.num_heads, self.head_dim)
        out = torch.zeros_like(q)
        if feats is not None:
            qsl = meta.query_start_loc.tolist()
            for i in range(meta.num_reqs):
                f = feats[i] if i < len(feats) else None
                if f is None:
                    continue  # no audio (or padded row): contributes 0
                s, e = int(qsl[i]), int(qsl[i + 1])
                if s == e:
                    continue
                kv = self.kv_proj(f.to(hidden.dtype))
                k, v = kv.chunk(2, dim=-1)
                k = k.view(-1, self.num_heads, self.head_dim)
                v = v.view(-1, self.num_heads, self.head_dim)
                out[s:e] = F.scaled_dot_product_attention(
                    q[s:e].transpose(0, 1), k.transpose(0, 1),
                    v.transpose(0, 1)).transpose(0, 1)
        return self.o_proj(out.reshape(t, -1))


class WhisperDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int,
                 dtype: torch.dtype):
        super().__init__()
        self.self_attn_layer_norm = LayerNorm(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype)
        self.self_attn = WhisperSelfAttention(spec, dtype, layer_idx)
        self.encoder_attn_layer_norm = LayerNorm(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype)
        self.encoder_attn = WhisperCrossAttention(spec, dtype)
        self.final_layer_norm = LayerNorm(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype)
        self.fc1 = ColumnParallelLinear(
            spec.hidden_size, spec.intermediate_size, bias=True,
            dtype=dtype)
        self.fc2 = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, bias=True,
            dtype=dtype)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        hidden = hidden + self.self_attn(
            self.self_attn_layer_norm(hidden))
        hidden = hidden + self.encoder_attn(
            self.encoder_attn_layer_norm(hidden))
        ff = self.fc2(F.gelu(self.fc1(self.final_layer_norm(hidden))))
        return hidden + ff


class WhisperForConditionalGeneration(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.config = config
        if get_pp_world_size() > 1:
            raise ValueError(
                "pipeline parallelism is not supported for "
                "encoder-decoder models")
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.embed_positions = nn.Embedding(
            spec.max_position_embeddings, spec.hidden_size)
        self.embed_positions.weight.requires_grad = False
        self.embed_positions.to(dtype)
        self.layers = nn.ModuleList([
            WhisperDecoderLayer(spec, i, dtype)
            for i in range(spec.num_layers)
        ])
        self.norm = LayerNorm(spec.hidden_size, spec.rms_norm_eps,
                              dtype=dtype)
        self.lm_head = ParallelLMHead(spec.vocab_size, spec.hidden_size,
                                      dtype=dtype)
        if spec.tie_word_embeddings:
            self.lm_head.weight = self.embed_tokens.weight

    def forward(self, input_ids, positions, hidden_in=None):
        hidden = self.embed_tokens(input_ids)
        hidden = hidden + self.embed_positions(positions)
        for layer in self.layers:
            hidden = layer(hidden)
        return self.norm(hidden)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
