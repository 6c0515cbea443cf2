"""GPU op wrappers backed by the hand-written CDNA4 HIP kernels.

Importing this module loads vllm_amd/_C.so (torch.library ops under
torch.ops.vllm_amd.*). If the extension is missing the import fails and
ops.get_backend() raises — the GPU path never silently falls back to
eager PyTorch.
"""

from __future__ import annotations

from pathlib import Path

import torch

_SO = Path(__file__).resolve().parent.parent / "_C.so"
if not _SO.exists():
    raise ImportError(
        f"vllm_amd HIP extension not found at {_SO}; "
        "run `python -m vllm_amd.build`"
    )
torch.ops.load_library(str(_SO))

_C = torch.ops.vllm_amd

# Decode flash-partition size — must match DEC_PART in attention_decode.hip.
_DEC_PART = 512


def rms_norm(x, weight, eps):
    return _C.rms_norm(x, weight, eps)


def fused_add_rms_norm(x, residual, weight, eps):
    # In-place: x <- norm(x + residual), residual <- x + residual.
    _C.fused_add_rms_norm(x, residual, weight, eps)
    return x, residual


def apply_rope(positions, q, k, cos_sin_cache, rotary_dim, is_neox=True):
    assert is_neox, "HIP rope kernel implements neox style"
    _C.rotary_embedding(positions, q, k, cos_sin_cache, rotary_dim)
    return q, k


def silu_and_mul(x):
    return _C.silu_and_mul(x)


def gelu_and_mul(x):
    return _C.gelu_and_mul(x)


def linear(x, weight, bias=None):
    # Tuned hipBLASLt GEMM (per-shape algo search on first use).
    return _C.lt_linear(x, weight, bias)


def reshape_and_cache(key, value, kv_cache, slot_mapping):
    _C.reshape_and_cache(key, value, kv_cache, slot_mapping)


def attention_unified(
    q,
    kv_cache,
    block_table,
    query_start_loc,
    seq_lens,
    scale,
    num_decodes=0,
    sliding_window=0,
    max_seq_len=0,
    max_query_len=0,
):
    """Dispatch decode rows (first num_decodes, query_len==1) to the
    partitioned VALU decode kernel and the rest to the MFMA prefill
    kernel. max_seq_len/max_query_len come from the scheduler (CPU) so
    no device sync is needed here."""
    num_tokens, num_heads, head_dim = q.shape
    # q may be a strided head-slice of the fused QKV output; the kernels
    # take its row stride. The output is always freshly contiguous.
    out = torch.empty(
        (num_tokens, num_heads, head_dim), dtype=q.dtype, device=q.device
    )
    if max_seq_len <= 0:
        max_seq_len = int(seq_lens.max().item())
    num_reqs = seq_lens.shape[0]

    if num_decodes > 0:
        q_dec = q[:num_decodes]
        num_parts = (max_seq_len + _DEC_PART - 1) // _DEC_PART
        if num_parts > 1:
            tmp_out = torch.empty(
                num_decodes, num_heads, num_parts, head_dim,
                dtype=torch.float32, device=q.device,
            )
            tmp_lse = torch.empty(
                num_decodes, num_heads, num_parts, 2,
                dtype=torch.float32, device=q.device,
            )
        else:
            tmp_out = q.new_empty(0, dtype=torch.float32)
            tmp_lse = tmp_out
        _C.paged_decode_attention(
            out[:num_decodes], q_dec, kv_cache, block_table, seq_lens,
            scale, max_seq_len, sliding_window, tmp_out, tmp_lse,
        )

    if num_decodes < num_reqs:
        if max_query_len <= 0:
            qsl = query_start_loc
            max_query_len = int((qsl[1:] - qsl[:-1]).max().item())
        _C.prefill_attention(
            out, q, kv_cache, block_table, query_start_loc, seq_lens,
            scale, num_decodes, max_query_len, sliding_window,
        )
    return out


# MoE ops: HIP grouped-GEMM kernels land with the Mixtral milestone; until
# then the documented GPU implementation is the eager composition below
# (routing + expert loop) — not a silent fallback of an existing kernel.
from vllm_amd.ops._torch_ref import fused_moe, topk_softmax  # noqa: E402,F401
