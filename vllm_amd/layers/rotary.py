"""Rotary position embedding with host-precomputed cos/sin cache.

Per the CDNA4 guide (Appendix B): trig tables are precomputed on host —
on-device sinf/cosf turns a memory-bound op VALU-bound. The HIP kernel
just gathers cos/sin rows and rotates q/k in place.
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from vllm_amd import ops


def _compute_inv_freq(rotary_dim: int, theta: float) -> torch.Tensor:
    return 1.0 / (
        theta
        ** (torch.arange(0, rotary_dim, 2, dtype=torch.float32) / rotary_dim)
    )


def _apply_llama3_scaling(inv_freq: torch.Tensor, scaling: dict) -> torch.Tensor:
    # Llama-3.1-style rope scaling.
    factor = scaling.get("factor", 8.0)
    low_freq_factor = scaling.get("low_freq_factor", 1.0)
    high_freq_factor = scaling.get("high_freq_factor", 4.0)
    old_ctx = scaling.get("original_max_position_embeddings", 8192)
    low_wavelen = old_ctx / low_freq_factor
    high_wavelen = old_ctx / high_freq_factor
    new_freqs = []
    for f in inv_freq:
        wavelen = 2 * math.pi / f
        if wavelen < high_wavelen:
            new_freqs.append(f)
        elif wavelen > low_wavelen:
            new_freqs.append(f / factor)
        else:
            smooth = (old_ctx / wavelen - low_freq_factor) / (
                high_freq_factor - low_freq_factor
            )
            new_freqs.append((1 - smooth) * f / factor + smooth * f)
    return torch.tensor(new_freqs, dtype=inv_freq.dtype)


def _yarn_find_dim(num_rot: float, dim: int, base: float,
                   max_pos: int) -> float:
    return (dim * math.log(max_pos / (num_rot * 2 * math.pi))) / \
        (2 * math.log(base))


def _apply_yarn_scaling(inv_freq: torch.Tensor, scaling: dict, dim: int,
                        theta: float) -> tuple[torch.Tensor, float]:
    """YaRN (role of the reference's YaRNScalingRotaryEmbedding, used by
    Qwen/DeepSeek long-context configs): interpolate low-frequency
    dimensions by `factor`, keep high-frequency ones, with a linear ramp
    between the beta_fast/beta_slow rotation counts. Returns the scaled
    inv_freq and the attention mscale factor folded into cos/sin."""
    factor = scaling.get("factor", 1.0)
    old_ctx = scaling.get("original_max_position_embeddings", 4096)
    beta_fast = scaling.get("beta_fast", 32)
    beta_slow = scaling.get("beta_slow", 1)
    lo = math.floor(_yarn_find_dim(beta_fast, dim, theta, old_ctx))
    hi = math.ceil(_yarn_find_dim(beta_slow, dim, theta, old_ctx))
    lo, hi = max(lo, 0), min(hi, dim - 1)
    # ramp in half-dim index space
    idx = torch.arange(dim // 2, dtype=torch.float32)
    ramp = ((idx - lo / 2) / max((hi - lo) / 2, 0.001)).clamp(0, 1)
    extrapolation = 1 - ramp   # 1 = keep original freq (high freq dims)
    interp = inv_freq / factor
    out = interp * ramp + inv_freq * extrapolation
    # attention scaling (mscale): folded into the cos/sin cache.
    mscale_cfg = scaling.get("mscale", 1.0)
    mscale_all = scaling.get("mscale_all_dim", 0.0)

    def _ms(scale, m):
        if scale <= 1 or m == 0:
            return 1.0
        return 0.1 * m * math.log(scale) + 1.0

    attn_factor = scaling.get("attn_factor", 1.0)
    mscale = (_ms(factor, mscale_cfg) / _ms(factor, mscale_all)
              if mscale_all else
              (0.1 * math.log(factor) + 1.0 if factor > 1 else 1.0))
    return out, float(mscale * attn_factor)


class RotaryEmbedding(nn.Module):
    def __init__(
        self,
        head_dim: int,
        rotary_dim: int,
        max_position: int,
        theta: float = 10000.0,
        is_neox: bool = True,
        rope_scaling: Optional[dict] = None,
        dtype: torch.dtype = torch.float32,
    ):
        super().__init__()
        self.head_dim = head_dim
        self.rotary_dim = rotary_dim
        self.is_neox = is_neox
        inv_freq = _compute_inv_freq(rotary_dim, theta)
        mscale = 1.0
        if rope_scaling:
            rtype = rope_scaling.get("rope_type", rope_scaling.get("type"))
            if rtype == "llama3":
                inv_freq = _apply_llama3_scaling(inv_freq, rope_scaling)
            elif rtype == "yarn":
                inv_freq, mscale = _apply_yarn_scaling(
                    inv_freq, rope_scaling, rotary_dim, theta)
        t = torch.arange(max_position, dtype=torch.float32)
        freqs = torch.outer(t, inv_freq)  # [max_pos, rotary_dim/2]
        cache = torch.cat([freqs.cos(), freqs.sin()], dim=-1) * mscale
        # fp32 cache: gathered per token; negligible memory.
        self.register_buffer("cos_sin_cache", cache, persistent=False)

    def forward(
        self, positions: torch.Tensor, q: torch.Tensor, k: torch.Tensor
    ):
        """positions [T]; q [T, Hq, D], k [T, Hkv, D] — rotated in place."""
        ops.apply_rope(
            positions, q, k, self.cos_sin_cache, self.rotary_dim, self.is_neox
        )
        return q, k
