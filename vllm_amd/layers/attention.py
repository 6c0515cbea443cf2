"""Attention layer: writes new K/V into the paged cache, then runs the
unified varlen attention op (prefill + decode in one call).

Role of the reference's Attention layer (layers/attention/attention.py:218)
with a single MI355X backend family instead of a backend registry.
"""

from __future__ import annotations


import torch
import torch.nn as nn

from vllm_amd import ops
from vllm_amd.worker.forward_context import get_forward_context


class Attention(nn.Module):
    def __init__(
        self,
        num_heads: int,
        head_dim: int,
        scale: float,
        num_kv_heads: int,
        layer_idx: int,
        sliding_window: int = 0,
        kv_group: str = "full",
    ):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = head_dim
        self.scale = scale
        self.num_kv_heads = num_kv_heads
        self.layer_idx = layer_idx
        self.sliding_window = sliding_window
        # "window": this layer reads/writes through the window group's
        # block table when the model has mixed sliding+global layers.
        self.kv_group = kv_group

    def forward(
        self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor
    ) -> torch.Tensor:
        """q: [T, Hq, D] or [T, Hq*D]; k/v likewise -> [T, Hq*D].

        Inputs may be strided head-slices of the fused QKV projection —
        the HIP kernels take row strides, so no contiguity copies here.
        """
        ctx = get_forward_context()
        meta = ctx.attn_metadata
        T = q.shape[0]
        if q.dim() == 2:
            q = q.view(T, self.num_heads, self.head_dim)
        if k.dim() == 2:
            k = k.view(T, self.num_kv_heads, self.head_dim)
        if v.dim() == 2:
            v = v.view(T, self.num_kv_heads, self.head_dim)

        if not ctx.kv_caches:
            # Memory-profiling run: no cache allocated; compute attention
            # over just the new tokens (worst-case activation footprint is
            # what matters, not values).
            return self._profile_attention(q, k, v, meta).reshape(T, -1)

        kv_cache = ctx.kv_caches[self.layer_idx]
        use_w = (self.kv_group == "window"
                 and meta.block_table_w is not None)
        slot_mapping = meta.slot_mapping_w if use_w else meta.slot_mapping
        block_table = meta.block_table_w if use_w else meta.block_table
        ops.reshape_and_cache(k, v, kv_cache, slot_mapping)
        out = ops.attention_unified(
            q,
            kv_cache,
            block_table,
            meta.query_start_loc,
            meta.seq_lens,
            self.scale,
            num_decodes=meta.num_decodes,
            sliding_window=self.sliding_window,
            max_seq_len=meta.max_seq_len,
            max_query_len=meta.max_query_len,
        )
        return out.reshape(T, -1)

    def _profile_attention(self, q, k, v, meta) -> torch.Tensor:
        """Cache-less varlen attention for the memory-profiling dummy run."""
        group = self.num_heads // self.num_kv_heads
        kk = k.repeat_interleave(group, dim=1)
        vv = v.repeat_interleave(group, dim=1)
        out = torch.empty_like(q)
        qs = meta.query_start_loc.tolist()
        for i in range(meta.num_reqs):
            s, e = qs[i], qs[i + 1]
            qi = q[s:e].transpose(0, 1).float()
            ki = kk[s:e].transpose(0, 1).float()
            vi = vv[s:e].transpose(0, 1).float()
            o = torch.nn.functional.scaled_dot_product_attention(
                qi, ki, vi, is_causal=True, scale=self.scale
            )
            out[s:e] = o.transpose(0, 1).to(out.dtype)
        return out
