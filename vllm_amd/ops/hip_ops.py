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


def quant_fp8_dynamic(x):
    # Per-token dynamic quant (csrc/quant_fp8.hip): one workgroup/row.
    M, K = x.shape
    out = torch.empty(M, K, dtype=torch.float8_e4m3fn, device=x.device)
    scales = torch.empty(M, dtype=torch.float32, device=x.device)
    _C.dynamic_quant_fp8(out.view(torch.uint8), scales, x)
    return out, scales


def linear_fp8(x, w_fp8, w_scale, bias=None):
    """W8A8 fp8 linear: per-token activation quant -> fp8 MFMA GEMM
    (tuned hipBLASLt, raw bf16 out) -> fused rs[m]*cs[n] rescale + bias.
    ~2x bf16 GEMM rate on gfx950 and half the weight HBM traffic."""
    shape = list(x.shape)
    x2 = x.reshape(-1, shape[-1]).contiguous()
    x_fp8, x_scale = quant_fp8_dynamic(x2)
    y = _C.lt_linear_fp8(x_fp8, w_fp8)
    _C.scale_rows_cols(y, x_scale, w_scale, bias)
    shape[-1] = w_fp8.shape[0]
    return y.reshape(shape)


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
        if head_dim not in (64, 128, 256):
            # Off-template head dims take an explicit torch path for the
            # prefill rows (decode runs the HIP kernel at any supported
            # head_dim). Gemma3 (256) and OPT (64) are on the MFMA
            # kernel since the head-dim templating.
            _warn_prefill_fallback(head_dim)
            from vllm_amd.ops import _torch_ref

            qsl = query_start_loc[num_decodes:] - num_decodes
            ref_out = _torch_ref.attention_unified(
                q[num_decodes:], kv_cache, block_table[num_decodes:],
                qsl, seq_lens[num_decodes:], scale, num_decodes=0,
                sliding_window=sliding_window,
            )
            out[num_decodes:] = ref_out
            return out
        if max_query_len <= 0:
            qsl = query_start_loc
            max_query_len = int((qsl[1:] - qsl[:-1]).max().item())
        _C.prefill_attention(
            out, q, kv_cache, block_table, query_start_loc, seq_lens,
            scale, num_decodes, max_query_len, sliding_window,
        )
    return out


_PREFILL_FALLBACK_WARNED = set()


def _warn_prefill_fallback(head_dim: int) -> None:
    if head_dim not in _PREFILL_FALLBACK_WARNED:
        _PREFILL_FALLBACK_WARNED.add(head_dim)
        import logging

        logging.getLogger(__name__).warning(
            "prefill head_dim=%d: off-template (64/128/256); prefill "
            "rows use the torch path (decode stays on the HIP kernel)",
            head_dim)


# MoE routing (small [T, E] tensors — torch ops are fine here).
from vllm_amd.ops import _torch_ref  # noqa: E402
from vllm_amd.ops._torch_ref import (  # noqa: E402,F401
    concat_and_cache_mla,
    grouped_topk,
    topk_softmax,
)

_MLA_PART = 1024  # must match mla::PART in csrc/mla.hip


def mla_decode(q_nope, q_pe, kv_cache, block_table, seq_lens, scale,
               max_seq_len):
    """Absorbed-MQA MLA decode on the MFMA kernel (csrc/mla.hip);
    kv_lora_rank 512 + rope 64 latent cache."""
    T, Hq, lora = q_nope.shape
    out = torch.empty(T, Hq, lora, dtype=q_nope.dtype, device=q_nope.device)
    if max_seq_len <= 0:
        max_seq_len = int(seq_lens.max().item())
    parts = max(1, -(-max_seq_len // _MLA_PART))
    if parts > 1:
        tmp_out = torch.empty(T, Hq, parts, lora, dtype=torch.float32,
                              device=q_nope.device)
        tmp_lse = torch.empty(T, Hq, parts, 2, dtype=torch.float32,
                              device=q_nope.device)
    else:
        tmp_out = q_nope.new_empty(0, dtype=torch.float32)
        tmp_lse = tmp_out
    _C.mla_decode(out, q_nope, q_pe, kv_cache, block_table, seq_lens,
                  scale, max_seq_len, tmp_out, tmp_lse)
    return out


def mla_attention(q_nope, q_pe, kv_cache, block_table, query_start_loc,
                  seq_lens, scale, num_decodes=0, max_seq_len=0):
    """Decode rows (query_len==1, first num_decodes) on the HIP MFMA
    kernel; prefill rows on the torch composition (MLA prefill runs per
    scheduler chunk — a bounded, non-steady-state path)."""
    T = q_nope.shape[0]
    num_reqs = seq_lens.shape[0]
    use_hip = (q_nope.shape[2] == 512 and q_pe.shape[2] == 64
               and q_nope.dtype in (torch.bfloat16, torch.float16)
               and kv_cache.dtype == q_nope.dtype)
    if num_decodes > 0 and use_hip:
        out_dec = mla_decode(
            q_nope[:num_decodes].contiguous(),
            q_pe[:num_decodes].contiguous(), kv_cache,
            block_table, seq_lens[:num_decodes], scale, max_seq_len)
        if num_decodes == T:
            return out_dec
        qsl = query_start_loc[num_decodes:] - num_decodes
        out_pre = _torch_ref.mla_attention(
            q_nope[num_decodes:], q_pe[num_decodes:], kv_cache,
            block_table[num_decodes:], qsl, seq_lens[num_decodes:], scale)
        return torch.cat([out_dec, out_pre], dim=0)
    return _torch_ref.mla_attention(
        q_nope, q_pe, kv_cache, block_table, query_start_loc, seq_lens,
        scale)


_MOE_BM = 64  # must match moe::BM in csrc/moe.hip


def _moe_hip_ok(hidden, w13, w2, activation):
    if activation != "silu":
        return False
    if hidden.dtype not in (torch.bfloat16, torch.float16):
        return False
    H = hidden.shape[1]
    twoI = w13.shape[1]
    # GEMM1: K=H%64, N=2I%128; GEMM2: K=I%64, N=H%128.
    return H % 128 == 0 and twoI % 128 == 0 and (twoI // 2) % 64 == 0


def moe_shuffle_weights(w):
    """Pre-shuffle expert weights [E, N, K] into MFMA fragment-major
    order [E][N/16][K/32][64 lanes][8 elems]: each wave's B-fragment
    becomes one coalesced 16B/lane HBM read (csrc/moe.hip
    moe_gemm_shuf_kernel). One-time cost at weight load."""
    E, N, K = w.shape
    v = w.view(E, N // 16, 16, K // 32, 4, 8)
    v = v.permute(0, 1, 3, 4, 2, 5)
    return v.reshape(E, (N // 16) * (K // 32) * 512).contiguous()


def fused_moe_hip(hidden, w13, w2, topk_weights, topk_ids,
                  w13_shuf=None, w2_shuf=None):
    """Grouped-GEMM MFMA MoE (csrc/moe.hip): on-device token->expert
    sort into BM tiles, two MFMA GEMMs streaming each expert's weights,
    deterministic weighted combine. No host sync — hipGraph-capturable
    (role of the reference's fused_moe.py:299 + moe_align_sum_kernels.cu).
    """
    T, H = hidden.shape
    E, twoI, _ = w13.shape
    I = twoI // 2
    k = topk_ids.shape[1]
    total = T * k
    dev = hidden.device
    em_max = -(-(total + E * (_MOE_BM - 1)) // _MOE_BM) * _MOE_BM
    ids32 = topk_ids.to(torch.int32).contiguous()
    sorted_ids = torch.empty(em_max, dtype=torch.int32, device=dev)
    expert_tiles = torch.empty(em_max // _MOE_BM, dtype=torch.int32,
                               device=dev)
    inv_perm = torch.empty(total, dtype=torch.int32, device=dev)
    scratch = torch.empty(3 * E, dtype=torch.int32, device=dev)
    counts, fill, off = scratch[:E], scratch[E:2 * E], scratch[2 * E:]
    _C.moe_align(ids32, E, sorted_ids, expert_tiles, inv_perm, counts,
                 fill, off)
    y = torch.empty(em_max, twoI, dtype=hidden.dtype, device=dev)
    if w13_shuf is not None:
        _C.moe_gemm_shuf(hidden, w13_shuf, y, sorted_ids, expert_tiles,
                         twoI, H, k, total)
    else:
        _C.moe_gemm(hidden, w13, y, sorted_ids, expert_tiles, k, total)
    act = silu_and_mul(y)
    y2 = torch.empty(em_max, H, dtype=hidden.dtype, device=dev)
    if w2_shuf is not None:
        _C.moe_gemm_shuf(act, w2_shuf, y2, sorted_ids, expert_tiles,
                         H, I, 0, total)
    else:
        _C.moe_gemm(act, w2, y2, sorted_ids, expert_tiles, 0, total)
    out = torch.empty(T, H, dtype=hidden.dtype, device=dev)
    _C.moe_combine(out, y2, topk_weights.float().contiguous(), inv_perm)
    return out


def fused_moe(hidden, w13, w2, topk_weights, topk_ids, activation="silu",
              w13_shuf=None, w2_shuf=None):
    """MoE expert MLP on GPU. Primary path: the grouped-GEMM MFMA kernel
    (fused_moe_hip, no host sync; fragment-major weight stream when the
    layer provides pre-shuffled copies). Fallback for unsupported shapes
    or activations: tokens sorted by expert, one tuned hipBLASLt GEMM
    pair per non-empty expert (one D2H sync per call), weighted
    scatter-add combine.

    hidden: [T, H]; w13: [E, 2I, H]; w2: [E, H, I].
    """
    if _moe_hip_ok(hidden, w13, w2, activation):
        return fused_moe_hip(hidden, w13, w2, topk_weights, topk_ids,
                             w13_shuf=w13_shuf, w2_shuf=w2_shuf)
    T, H = hidden.shape
    E = w13.shape[0]
    k = topk_ids.shape[1]
    flat = topk_ids.long().flatten()
    order = torch.argsort(flat, stable=True)
    token_of = order // k
    counts = torch.bincount(flat, minlength=E).cpu().tolist()
    x = hidden[token_of].contiguous()
    out_sorted = torch.empty(T * k, H, dtype=hidden.dtype,
                             device=hidden.device)
    act = silu_and_mul if activation == "silu" else gelu_and_mul
    offset = 0
    for e in range(E):
        c = counts[e]
        if c == 0:
            continue
        seg = x[offset:offset + c]
        h = act(linear(seg, w13[e]))
        out_sorted[offset:offset + c] = linear(h, w2[e])
        offset += c
    w = topk_weights.flatten()[order].unsqueeze(1).float()
    out = torch.zeros(T, H, dtype=torch.float32, device=hidden.device)
    out.index_add_(0, token_of, out_sorted.float() * w)
    return out.to(hidden.dtype)
