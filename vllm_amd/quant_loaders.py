"""Quantized-checkpoint loaders.

Role of the reference's layers/quantization/awq.py checkpoint handling:
AWQ (4-bit, group-scaled, zero-pointed) checkpoints are DEQUANTIZED at
load into the model dtype, so any AWQ-packed model runs today; combining
with `--quantization fp8` re-quantizes the dequantized weights to W8A8
for the fp8 MFMA path. A native int4 dequant-GEMM kernel (keeping the 4×
weight-memory saving at run time) is a tracked round-2 item.

AWQ packing: qweight int32 [K, N/8] — eight 4-bit values per int32
along the OUTPUT dim in the interleaved order [0, 2, 4, 6, 1, 3, 5, 7];
qzeros int32 [K/g, N/8] same packing; scales fp16 [K/g, N]:
    w[k, n] = (q[k, n] - zero[k//g, n]) * scale[k//g, n]
and the HF Linear weight is the transpose [N, K].
"""

from __future__ import annotations

import torch

# Nibble i of the int32 holds output column base*8 + _AWQ_ORDER[i].
_AWQ_ORDER = (0, 2, 4, 6, 1, 3, 5, 7)


def _unpack_int4(packed: torch.Tensor) -> torch.Tensor:
    """int32 [R, C] -> int32 [R, C*8] in logical column order."""
    R, C = packed.shape
    out = torch.empty(R, C * 8, dtype=torch.int32)
    p = packed.to(torch.int64)  # avoid sign trouble on >>28
    for i, col in enumerate(_AWQ_ORDER):
        out[:, col::8] = ((p >> (4 * i)) & 0xF).to(torch.int32)
    return out


def dequant_awq(qweight: torch.Tensor, qzeros: torch.Tensor,
                scales: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """Returns the dense Linear weight [out_features, in_features]."""
    K = qweight.shape[0]
    q = _unpack_int4(qweight).float()               # [K, N]
    z = _unpack_int4(qzeros).float()                # [K/g, N]
    s = scales.float()                              # [K/g, N]
    group = K // z.shape[0]
    z = z.repeat_interleave(group, dim=0)           # [K, N]
    s = s.repeat_interleave(group, dim=0)
    w = (q - z) * s                                 # [K, N]
    return w.t().contiguous().to(dtype)             # [N, K]


def pack_awq(weight: torch.Tensor, group_size: int = 128,
             zero: int = 8) -> tuple[torch.Tensor, torch.Tensor,
                                     torch.Tensor]:
    """Quantize a dense [N, K] weight into AWQ tensors (tests + tooling;
    symmetric zero-point `zero`). Returns (qweight, qzeros, scales)."""
    N, K = weight.shape
    assert K % group_size == 0
    w = weight.float().t().contiguous()             # [K, N]
    wg = w.view(K // group_size, group_size, N)
    amax = wg.abs().amax(dim=1).clamp_min(1e-8)     # [K/g, N]
    scales = amax / max(zero, 15 - zero)
    q = torch.round(
        w / scales.repeat_interleave(group_size, dim=0) + zero
    ).clamp(0, 15).to(torch.int32)                  # [K, N]
    zeros = torch.full((K // group_size, N), zero, dtype=torch.int32)

    def pack(t):
        R, C = t.shape
        packed = torch.zeros(R, C // 8, dtype=torch.int64)
        for i, col in enumerate(_AWQ_ORDER):
            packed |= t[:, col::8].to(torch.int64) << (4 * i)
        # reinterpret the low 32 bits as int32
        return packed.to(torch.uint32).view(torch.int32) \
            if hasattr(torch, "uint32") else packed.to(torch.int32)

    return pack(q), pack(zeros), scales.to(torch.float16)


# GPTQ packing (non-act-order): qweight int32 [K/8, N] — eight 4-bit
# values per int32 along the INPUT dim in plain sequential order;
# qzeros packed like AWQ's along N but stored MINUS ONE; scales fp16
# [K/g, N]. Detected by qweight orientation (rows*8 == scales groups*g).
def _unpack_int4_rows(packed: torch.Tensor) -> torch.Tensor:
    """int32 [R, C] -> int32 [R*8, C] sequential along rows."""
    R, C = packed.shape
    out = torch.empty(R * 8, C, dtype=torch.int32)
    p = packed.to(torch.int64)
    for i in range(8):
        out[i::8, :] = ((p >> (4 * i)) & 0xF).to(torch.int32)
    return out


def _unpack_int4_cols_seq(packed: torch.Tensor) -> torch.Tensor:
    """int32 [R, C] -> int32 [R, C*8] sequential along columns (GPTQ
    qzeros layout)."""
    R, C = packed.shape
    out = torch.empty(R, C * 8, dtype=torch.int32)
    p = packed.to(torch.int64)
    for i in range(8):
        out[:, i::8] = ((p >> (4 * i)) & 0xF).to(torch.int32)
    return out


def dequant_gptq(qweight: torch.Tensor, qzeros: torch.Tensor,
                 scales: torch.Tensor, dtype: torch.dtype,
                 g_idx: torch.Tensor = None) -> torch.Tensor:
    """Dense Linear weight [out, in] from GPTQ tensors (act-order g_idx
    supported when provided)."""
    q = _unpack_int4_rows(qweight).float()          # [K, N]
    z = _unpack_int4_cols_seq(qzeros).float() + 1   # stored minus one
    s = scales.float()                              # [K/g, N]
    K = q.shape[0]
    group = K // s.shape[0]
    if g_idx is not None and g_idx.numel() == K:
        gi = g_idx.long()
    else:
        gi = torch.arange(K) // group
    w = (q - z[gi]) * s[gi]                         # [K, N]
    return w.t().contiguous().to(dtype)


def dequantize_awq_stream(tensors: "dict[str, torch.Tensor]",
                          dtype: torch.dtype):
    """Transform a checkpoint tensor map: every {prefix}.qweight /
    .qzeros / .scales triple becomes one {prefix}.weight dense tensor;
    other entries pass through."""
    prefixes = {n[: -len(".qweight")] for n in tensors if
                n.endswith(".qweight")}
    out = {}
    for name, t in tensors.items():
        stem = name.rsplit(".", 1)[0]
        if stem in prefixes:
            if name.endswith(".qweight"):
                scales = tensors[f"{stem}.scales"]
                qzeros = tensors[f"{stem}.qzeros"]
                # Orientation tells the formats apart: AWQ packs the
                # OUTPUT dim into int32 columns (cols*8 == N == scales
                # cols); GPTQ packs the INPUT dim into rows (cols == N).
                if t.shape[1] * 8 == scales.shape[1]:
                    out[f"{stem}.weight"] = dequant_awq(
                        t, qzeros, scales, dtype)
                elif t.shape[1] == scales.shape[1]:
                    out[f"{stem}.weight"] = dequant_gptq(
                        t, qzeros, scales, dtype,
                        tensors.get(f"{stem}.g_idx"))
                else:
                    raise ValueError(
                        f"unrecognized int4 packing for {stem}: "
                        f"qweight {tuple(t.shape)} vs scales "
                        f"{tuple(scales.shape)}")
            continue  # qzeros/scales/g_idx consumed above
        if name.endswith(".g_idx"):
            continue
        out[name] = t
    return out


def dequantize_fp8_block_stream(tensors: "dict[str, torch.Tensor]",
                                dtype: torch.dtype,
                                block: int = 128):
    """DeepSeek-V3-style block-wise fp8 checkpoints (reference
    layers/quantization/fp8.py weight_block_size=[128,128] path):
    {prefix}.weight is float8_e4m3fn [N, K] with
    {prefix}.weight_scale_inv float32 [ceil(N/128), ceil(K/128)];
    w[n, k] = fp8[n, k] * scale_inv[n//128, k//128]. Dequantized to the
    model dtype at load (the fp8 MFMA runtime path re-quantizes W8A8
    per-channel under --quantization fp8)."""
    out = {}
    for name, t in tensors.items():
        if name.endswith(".weight_scale_inv"):
            continue
        scale_name = f"{name}_scale_inv"
        if (name.endswith(".weight")
                and t.dtype == torch.float8_e4m3fn
                and scale_name in tensors):
            s = tensors[scale_name].float()
            n, k = t.shape
            s_full = s.repeat_interleave(block, 0)[:n] \
                      .repeat_interleave(block, 1)[:, :k]
            out[name] = (t.float() * s_full).to(dtype)
        else:
            out[name] = t
    return out
