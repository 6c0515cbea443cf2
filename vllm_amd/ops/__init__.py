"""Custom op dispatch.

On a GPU (MI355X) the hand-written HIP/CDNA4 kernels in vllm_amd/csrc are
REQUIRED: if the compiled extension is missing we raise instead of
silently falling back to eager PyTorch (per-op wrappers live in
vllm_amd/ops/hip_ops.py). On CPU the plain-PyTorch reference
implementations in _torch_ref.py run — they double as the numerics
ground truth the GPU kernel tests compare against.
"""

from __future__ import annotations

import torch

from vllm_amd.ops import _torch_ref

_HIP_LIB = None
_HIP_CHECKED = False


def _load_hip():
    global _HIP_LIB, _HIP_CHECKED
    if _HIP_CHECKED:
        return _HIP_LIB
    _HIP_CHECKED = True
    try:
        from vllm_amd.ops import hip_ops

        _HIP_LIB = hip_ops
    except ImportError as e:
        _HIP_LIB = None
        _HIP_IMPORT_ERROR[0] = e
    return _HIP_LIB


_HIP_IMPORT_ERROR = [None]


def get_backend(device: torch.device):
    if device.type == "cuda":
        lib = _load_hip()
        if lib is None:
            raise RuntimeError(
                "vllm_amd HIP extension not built — refusing to run the GPU "
                "path on eager PyTorch. Run `python -m vllm_amd.build` or "
                f"__graft_entry__.build(). Import error: {_HIP_IMPORT_ERROR[0]}"
            )
        return lib
    return _torch_ref


def rms_norm(x, weight, eps):
    return get_backend(x.device).rms_norm(x, weight, eps)


def fused_add_rms_norm(x, residual, weight, eps):
    """Returns (normed, new_residual). May modify x/residual in place."""
    return get_backend(x.device).fused_add_rms_norm(x, residual, weight, eps)


def apply_rope(positions, q, k, cos_sin_cache, rotary_dim, is_neox=True):
    """In-place rotary embedding on q [T, Hq*D] and k [T, Hkv*D]."""
    return get_backend(q.device).apply_rope(
        positions, q, k, cos_sin_cache, rotary_dim, is_neox
    )


def silu_and_mul(x):
    return get_backend(x.device).silu_and_mul(x)


def gelu_and_mul(x):
    return get_backend(x.device).gelu_and_mul(x)


def reshape_and_cache(key, value, kv_cache, slot_mapping):
    return get_backend(key.device).reshape_and_cache(
        key, value, kv_cache, slot_mapping
    )


def attention_unified(
    q,
    kv_cache,
    block_table,
    query_start_loc,
    seq_lens,
    scale,
    num_decodes: int = 0,
    sliding_window: int = 0,
    max_seq_len: int = 0,
    max_query_len: int = 0,
):
    """Varlen attention over the paged KV cache (prefill + decode mixed).

    q: [num_tokens, num_heads, head_dim]; kv_cache: [2, num_blocks,
    num_kv_heads, block_size, head_dim]; returns [num_tokens, num_heads,
    head_dim]. Requests are ordered decodes-first (query_len==1).
    max_seq_len/max_query_len are CPU-known hints so the GPU path never
    syncs to size its scratch.
    """
    return get_backend(q.device).attention_unified(
        q, kv_cache, block_table, query_start_loc, seq_lens, scale,
        num_decodes, sliding_window, max_seq_len, max_query_len,
    )


def linear(x, weight, bias=None):
    """x @ weight.T + bias — tuned hipBLASLt on GPU, F.linear on CPU."""
    return get_backend(x.device).linear(x, weight, bias)


def linear_fp8(x, w_fp8, w_scale, bias=None):
    """W8A8 fp8 linear: dynamic per-token activation scales, per-channel
    weight scales. fp8 MFMA GEMM on GPU; bit-matching fp32 simulation on
    CPU (_torch_ref.linear_fp8)."""
    return get_backend(x.device).linear_fp8(x, w_fp8, w_scale, bias)


def quant_fp8_dynamic(x):
    """Per-token dynamic e4m3 quantization: [M,K] -> (fp8 [M,K], [M] f32)."""
    return get_backend(x.device).quant_fp8_dynamic(x)


# Weight quantization happens once at load — plain torch either device.
from vllm_amd.ops._torch_ref import quantize_weight_fp8  # noqa: E402,F401


def concat_and_cache_mla(c_kv, k_pe, kv_cache, slot_mapping):
    return get_backend(c_kv.device).concat_and_cache_mla(
        c_kv, k_pe, kv_cache, slot_mapping
    )


def mla_attention(q_nope, q_pe, kv_cache, block_table, query_start_loc,
                  seq_lens, scale, num_decodes=0, max_seq_len=0):
    backend = get_backend(q_nope.device)
    if backend is not _torch_ref:
        return backend.mla_attention(
            q_nope, q_pe, kv_cache, block_table, query_start_loc, seq_lens,
            scale, num_decodes=num_decodes, max_seq_len=max_seq_len,
        )
    return backend.mla_attention(
        q_nope, q_pe, kv_cache, block_table, query_start_loc, seq_lens, scale
    )


def grouped_topk(gating, topk, renormalize=True, num_groups=0,
                 topk_groups=0, scoring_func="softmax", e_score_bias=None,
                 routed_scaling_factor=1.0):
    return get_backend(gating.device).grouped_topk(
        gating, topk, renormalize, num_groups, topk_groups, scoring_func,
        e_score_bias, routed_scaling_factor
    )


def topk_softmax(gating_logits, topk, renormalize=True):
    return get_backend(gating_logits.device).topk_softmax(
        gating_logits, topk, renormalize
    )


def fused_moe(hidden, w13, w2, topk_weights, topk_ids, activation="silu",
              w13_shuf=None, w2_shuf=None):
    backend = get_backend(hidden.device)
    if backend is _torch_ref:
        return backend.fused_moe(hidden, w13, w2, topk_weights, topk_ids,
                                 activation)
    return backend.fused_moe(hidden, w13, w2, topk_weights, topk_ids,
                             activation, w13_shuf=w13_shuf,
                             w2_shuf=w2_shuf)
