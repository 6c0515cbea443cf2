"""LLaMA-family decoder (Llama-3 8B/70B targets).

Structure mirrors the reference's llama.py (vllm/model_executor/models/
llama.py:116,222,311) — fused QKV and gate_up GEMMs, RMSNorm with fused
residual add, RoPE, GQA attention over the paged cache, two TP
all-reduces per layer (o_proj + down_proj).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.activation import SiluAndMul
from vllm_amd.layers.attention import Attention
from vllm_amd.worker.forward_context import get_forward_context
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.layernorm import LayerNorm, RMSNorm
from vllm_amd.layers.linear import (
    MergedColumnParallelLinear,
    QKVParallelLinear,
    RowParallelLinear,
)
from vllm_amd.layers.rotary import RotaryEmbedding
from vllm_amd.parallel.state import (
    is_first_pp_rank,
    tensor_model_parallel_all_reduce,
    is_last_pp_rank,
    pp_layer_range,
)


class LlamaMLP(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype):
        super().__init__()
        self.gated = spec.gated_mlp
        if self.gated:
            self.gate_up_proj = MergedColumnParallelLinear(
                spec.hidden_size,
                [spec.intermediate_size, spec.intermediate_size],
                bias=spec.use_bias,
                dtype=dtype,
            )
            self.act_fn = SiluAndMul()
        else:
            # NeoX/Falcon plain MLP: up -> act -> down.
            from vllm_amd.layers.linear import ColumnParallelLinear
            self.up_proj = ColumnParallelLinear(
                spec.hidden_size, spec.intermediate_size,
                bias=spec.use_bias, dtype=dtype)
            self.plain_act = spec.activation
        self.down_proj = RowParallelLinear(
            spec.intermediate_size, spec.hidden_size, bias=spec.use_bias,
            dtype=dtype,
        )

    def forward(self, x, reduce_results: bool = True):
        if self.gated:
            x = self.gate_up_proj(x)
            x = self.act_fn(x)
        else:
            x = self.up_proj(x)
            if self.plain_act == "gelu":
                x = torch.nn.functional.gelu(x, approximate="tanh")
            else:
                x = torch.nn.functional.relu(x)
        return self.down_proj(x, reduce_results=reduce_results)


class LlamaAttention(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype: torch.dtype,
                 max_position: int, cache_idx: int = -1):
        super().__init__()
        self.qkv_proj = QKVParallelLinear(
            spec.hidden_size,
            spec.head_dim,
            spec.num_heads,
            spec.num_kv_heads,
            bias=spec.use_bias or spec.qkv_bias,
            dtype=dtype,
        )
        self.num_heads = self.qkv_proj.num_heads
        self.num_kv_heads = self.qkv_proj.num_kv_heads
        self.head_dim = spec.head_dim
        self.o_proj = RowParallelLinear(
            spec.num_heads * spec.head_dim, spec.hidden_size,
            bias=spec.use_bias, dtype=dtype,
        )
        rotary_dim = int(spec.head_dim * spec.partial_rotary_factor)
        rotary_dim -= rotary_dim % 2
        self.rotary_emb = RotaryEmbedding(
            spec.head_dim,
            rotary_dim,
            max_position,
            theta=spec.rope_theta,
            rope_scaling=spec.rope_scaling,
        )
        # Qwen3-style per-head q/k RMSNorm (reference qwen3.py:60-61).
        self.q_norm = (RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype=dtype)
                       if spec.qk_norm else None)
        self.k_norm = (RMSNorm(spec.head_dim, spec.rms_norm_eps, dtype=dtype)
                       if spec.qk_norm else None)
        # Per-layer window: all layers windowed (Mistral) unless the spec
        # marks every Nth layer global (Gemma3 pattern).
        window = spec.sliding_window
        if window and spec.global_attn_every_n_layers:
            if (layer_idx + 1) % spec.global_attn_every_n_layers == 0:
                window = 0
        self.attn = Attention(
            self.num_heads,
            spec.head_dim,
            scale=spec.head_dim**-0.5,
            num_kv_heads=self.num_kv_heads,
            layer_idx=cache_idx if cache_idx >= 0 else layer_idx,
            sliding_window=window,
            kv_group=("window" if window and spec.is_mixed_attn
                      else "full"),
        )

    def forward(self, positions, hidden, reduce_results: bool = True):
        T = hidden.shape[0]
        qkv = self.qkv_proj(hidden)
        q, k, v = self.qkv_proj.split_qkv(qkv)
        q = q.view(T, self.num_heads, self.head_dim)
        k = k.view(T, self.num_kv_heads, self.head_dim)
        if self.q_norm is not None:
            # reshape() of a fused-QKV head slice may stay a strided VIEW
            # (e.g. num_kv_heads=1); the HIP rms_norm wants contiguous.
            q = self.q_norm(
                q.reshape(-1, self.head_dim).contiguous()).view(q.shape)
            k = self.k_norm(
                k.reshape(-1, self.head_dim).contiguous()).view(k.shape)
        self.rotary_emb(positions, q, k)
        out = self.attn(q, k, v)
        return self.o_proj(out, reduce_results=reduce_results)


class LlamaDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position,
                 cache_idx: int = -1):
        super().__init__()
        self.self_attn = LlamaAttention(spec, layer_idx, dtype, max_position,
                                        cache_idx)
        self.mlp = LlamaMLP(spec, dtype)
        norm_cls = LayerNorm if spec.use_layernorm else RMSNorm
        self.parallel_residual = spec.parallel_residual
        self.input_layernorm = norm_cls(spec.hidden_size, spec.rms_norm_eps,
                                        dtype=dtype)
        self.post_attention_layernorm = norm_cls(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype
        )

    def forward(self, positions, hidden, residual):
        if self.parallel_residual:
            # GPT-NeoX / Falcon form: x + attn(ln1(x)) + mlp(ln2(x)).
            # Both branches produce row-parallel PARTIALS summed before
            # ONE all-reduce — half the per-layer comm of sequential
            # blocks under TP.
            if residual is not None:
                hidden = hidden + residual
            attn_p = self.self_attn(positions,
                                    self.input_layernorm(hidden),
                                    reduce_results=False)
            mlp_p = self.mlp(self.post_attention_layernorm(hidden),
                             reduce_results=False)
            out = tensor_model_parallel_all_reduce(attn_p + mlp_p)
            return out, hidden
        sp = get_forward_context().sp_size
        if sp > 1:
            return self._forward_sp(positions, hidden, residual)
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        hidden = self.self_attn(positions, hidden)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        hidden = self.mlp(hidden)
        return hidden, residual

    def _forward_sp(self, positions, hidden, residual):
        """Sequence-parallel block (role of the reference's SP compile
        pass, parallel_state.py:164-250): the residual stream is sharded
        across TP ranks between blocks — norms, residual adds and
        activations run on T/tp rows; the per-layer all-reduces become
        all-gather (before the sharded-weight matmuls) + reduce-scatter
        (after the row-parallel ones). Same bytes on the wire as TP, but
        the elementwise work is 1/tp per rank and the RS output lands
        pre-sharded for the next block's norm."""
        from vllm_amd.parallel.state import (sp_all_gather_rows,
                                             sp_reduce_scatter_rows)
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        full = sp_all_gather_rows(hidden)
        part = self.self_attn(positions, full, reduce_results=False)
        hidden = sp_reduce_scatter_rows(part)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        full = sp_all_gather_rows(hidden)
        part = self.mlp(full, reduce_results=False)
        hidden = sp_reduce_scatter_rows(part)
        return hidden, residual


class LlamaModel(nn.Module):
    """With pipeline parallelism, this stage materializes only its layer
    slice [lo, hi) — other slots hold nn.Identity so parameter NAMES keep
    their global layer indices (checkpoint loading and the name-seeded
    dummy init stay partition-invariant). The embedding lives on the
    first stage, final norm on the last; stage boundaries carry one
    [T, hidden] activation (hidden + residual pre-combined)."""

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
        self.layers = nn.ModuleList(
            [
                LlamaDecoderLayer(spec, i, dtype, config.max_model_len,
                                  cache_idx=i - self.lo)
                if self.lo <= i < self.hi else nn.Identity()
                for i in range(spec.num_layers)
            ]
        )
        norm_cls = LayerNorm if spec.use_layernorm else RMSNorm
        self.norm = (norm_cls(spec.hidden_size, spec.rms_norm_eps,
                              dtype=dtype)
                     if is_last_pp_rank() else None)

    def forward(self, input_ids, positions, hidden_in=None):
        ctx = get_forward_context()
        if self.embed_tokens is not None:
            hidden = self.embed_tokens(input_ids)
            mm = ctx.mm_embeds
            if mm is not None:
                idx, feats = mm
                hidden = hidden.index_copy(
                    0, idx, feats.to(hidden.dtype))
            residual = None
        else:
            hidden = hidden_in
            residual = None
        sp = ctx.sp_size
        if sp > 1:
            # Sequence parallelism: shard the residual stream rows
            # (runner pads the batch to a multiple of tp).
            from vllm_amd.parallel.state import get_tp_rank
            T = hidden.shape[0]
            assert T % sp == 0, "SP needs the batch padded to tp"
            shard = T // sp
            r = get_tp_rank()
            hidden = hidden[r * shard:(r + 1) * shard]
        for layer in self.layers[self.lo:self.hi]:
            hidden, residual = layer(positions, hidden, residual)
        if self.norm is None:
            return hidden + residual  # stage-boundary activation
        hidden, _ = self.norm(hidden, residual)
        if sp > 1:
            from vllm_amd.parallel.state import sp_all_gather_rows
            hidden = sp_all_gather_rows(hidden)
        return hidden


class LlamaForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = LlamaModel(config)
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
