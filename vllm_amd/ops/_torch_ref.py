"""Plain-PyTorch reference implementations of every custom op.

These are (a) the CPU execution path and (b) the numerics ground truth
for the HIP kernel tests (tests compare HIP output against these run in
fp32). Keep them simple and obviously correct.
"""

from __future__ import annotations

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    dtype = x.dtype
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps)
    return (out * weight.float()).to(dtype)


def fused_add_rms_norm(x, residual, weight, eps):
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rms_norm(new_residual, weight, eps), new_residual


def apply_rope(positions, q, k, cos_sin_cache, rotary_dim, is_neox=True):
    """In-place rotary embedding.

    positions: [T] int64; q: [T, Hq*D]; k: [T, Hkv*D];
    cos_sin_cache: [max_pos, rotary_dim] = concat(cos, sin) halves.
    """
    num_tokens = positions.shape[0]
    cos_sin = cos_sin_cache[positions]  # [T, rotary_dim]
    cos, sin = cos_sin.chunk(2, dim=-1)  # [T, rotary_dim/2] each

    def rotate(t: torch.Tensor) -> torch.Tensor:
        # t: [T, H, D]; rotate the first rotary_dim of D.
        rot = t[..., :rotary_dim].float()
        if is_neox:
            x1, x2 = rot.chunk(2, dim=-1)
            c = cos.unsqueeze(1).float()
            s = sin.unsqueeze(1).float()
            o1 = x1 * c - x2 * s
            o2 = x2 * c + x1 * s
            rotated = torch.cat([o1, o2], dim=-1)
        else:  # GPT-J interleaved
            x1 = rot[..., ::2]
            x2 = rot[..., 1::2]
            c = cos.unsqueeze(1).float()
            s = sin.unsqueeze(1).float()
            o1 = x1 * c - x2 * s
            o2 = x2 * c + x1 * s
            rotated = torch.stack([o1, o2], dim=-1).flatten(-2)
        t[..., :rotary_dim] = rotated.to(t.dtype)
        return t

    if q.dim() != 3 or k.dim() != 3:
        raise ValueError("apply_rope expects q/k as [T, H, D] views")
    rotate(q)
    rotate(k)
    return q, k


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    return F.silu(x[..., :d].float()).to(x.dtype) * x[..., d:]


def gelu_and_mul(x: torch.Tensor) -> torch.Tensor:
    d = x.shape[-1] // 2
    return F.gelu(x[..., :d].float(), approximate="tanh").to(x.dtype) * x[..., d:]


def reshape_and_cache(key, value, kv_cache, slot_mapping):
    """key/value: [T, Hkv, D]; kv_cache: [2, num_blocks, Hkv, block_size, D]
    (head-major inside the block — each (block, head) tile is one contiguous
    chunk for the HIP kernels); slot_mapping: [T] int64
    (block_id * block_size + offset)."""
    block_size = kv_cache.shape[3]
    block_ids = slot_mapping // block_size
    offsets = slot_mapping % block_size
    kv_cache[0, block_ids, :, offsets] = key.to(kv_cache.dtype)
    kv_cache[1, block_ids, :, offsets] = value.to(kv_cache.dtype)


def _gather_kv(kv_cache, block_table_row, ctx_len):
    """Gather [ctx_len, Hkv, D] K and V for one request from paged cache."""
    block_size = kv_cache.shape[3]
    num_blocks_needed = (ctx_len + block_size - 1) // block_size
    blocks = block_table_row[:num_blocks_needed].long()
    k = kv_cache[0, blocks]  # [nb, Hkv, bs, D]
    v = kv_cache[1, blocks]
    k = k.transpose(1, 2).reshape(-1, k.shape[1], k.shape[3])[:ctx_len]
    v = v.transpose(1, 2).reshape(-1, v.shape[1], v.shape[3])[:ctx_len]
    return k, v


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
    num_tokens, num_heads, head_dim = q.shape
    num_kv_heads = kv_cache.shape[2]
    group = num_heads // num_kv_heads
    head_dim_v = kv_cache.shape[4]
    out = q.new_empty(num_tokens, num_heads, head_dim_v)
    num_reqs = seq_lens.shape[0]
    qs = query_start_loc.tolist()
    for i in range(num_reqs):
        q_start, q_end = qs[i], qs[i + 1]
        ql = q_end - q_start
        ctx = int(seq_lens[i])
        qi = q[q_start:q_end].float()  # [ql, H, D]
        k, v = _gather_kv(kv_cache, block_table[i], ctx)
        k = k.float().repeat_interleave(group, dim=1)  # [ctx, H, D]
        v = v.float().repeat_interleave(group, dim=1)
        # scores: [H, ql, ctx]
        scores = torch.einsum("qhd,khd->hqk", qi, k) * scale
        # Causal mask: query j (global pos ctx-ql+j) sees keys <= its pos.
        kpos = torch.arange(ctx, device=q.device)
        qpos = ctx - ql + torch.arange(ql, device=q.device)
        mask = kpos.unsqueeze(0) > qpos.unsqueeze(1)  # [ql, ctx]
        if sliding_window > 0:
            mask |= kpos.unsqueeze(0) <= (qpos.unsqueeze(1) - sliding_window)
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        p = scores.softmax(dim=-1)
        o = torch.einsum("hqk,khd->qhd", p, v)
        out[q_start:q_end] = o.to(out.dtype)
    return out


def linear(x, weight, bias=None):
    return F.linear(x, weight, bias)


FP8_MAX = 448.0  # OCP e4m3fn max finite


def quant_fp8_dynamic(x):
    """Per-token (row) dynamic fp8 quantization.

    x: [M, K] -> (x_fp8 [M, K] float8_e4m3fn, scales [M] f32). Ground
    truth for the HIP kernel in csrc/quant_fp8.hip."""
    xf = x.float()
    amax = xf.abs().amax(dim=-1).clamp_min(1e-12)
    scales = amax / FP8_MAX
    q = (xf / scales.unsqueeze(1)).to(torch.float8_e4m3fn)
    return q, scales


def quantize_weight_fp8(w):
    """Per-output-channel fp8 weight quantization (done once at load).

    w: [N, K] -> (w_fp8 [N, K] float8_e4m3fn, scales [N] f32)."""
    wf = w.float()
    amax = wf.abs().amax(dim=-1).clamp_min(1e-12)
    scales = amax / FP8_MAX
    q = (wf / scales.unsqueeze(1)).to(torch.float8_e4m3fn)
    return q, scales


def linear_fp8(x, w_fp8, w_scale, bias=None):
    """W8A8 fp8 linear, CPU reference: quantize activations per-token
    exactly as the GPU path does, then compute in fp32 on dequantized
    values (same rounding points as fp8-GEMM + rs*cs rescale)."""
    shape = list(x.shape)
    x2 = x.reshape(-1, shape[-1])
    x_fp8, x_scale = quant_fp8_dynamic(x2)
    y = x_fp8.float() @ w_fp8.float().t()
    y = y * x_scale.unsqueeze(1) * w_scale.unsqueeze(0)
    if bias is not None:
        y = y + bias.float()
    shape[-1] = w_fp8.shape[0]
    return y.to(x.dtype).reshape(shape)


def topk_softmax(gating_logits, topk, renormalize=True):
    """gating_logits: [T, E] -> (topk_weights [T,k] f32, topk_ids [T,k] i32)."""
    probs = gating_logits.float().softmax(dim=-1)
    topk_weights, topk_ids = probs.topk(topk, dim=-1)
    if renormalize:
        topk_weights = topk_weights / topk_weights.sum(dim=-1, keepdim=True)
    return topk_weights, topk_ids.to(torch.int32)


def fused_moe(hidden, w13, w2, topk_weights, topk_ids, activation="silu"):
    """Reference MoE: loop over experts.

    hidden: [T, H]; w13: [E, 2I, H]; w2: [E, H, I];
    topk_weights/topk_ids: [T, k]. Returns [T, H].
    """
    T, H = hidden.shape
    E = w13.shape[0]
    out = torch.zeros(T, H, dtype=torch.float32, device=hidden.device)
    act = silu_and_mul if activation == "silu" else gelu_and_mul
    flat_ids = topk_ids.long()
    for e in range(E):
        mask = flat_ids == e  # [T, k]
        if not mask.any():
            continue
        token_idx, k_idx = mask.nonzero(as_tuple=True)
        x = hidden[token_idx]
        h = act(x @ w13[e].t())
        y = h @ w2[e].t()
        w = topk_weights[token_idx, k_idx].unsqueeze(-1).float()
        out.index_add_(0, token_idx, y.float() * w)
    return out.to(hidden.dtype)


def concat_and_cache_mla(c_kv, k_pe, kv_cache, slot_mapping):
    """MLA cache write: c_kv [T, lora], k_pe [T, rope] ->
    kv_cache [num_blocks, block_size, lora+rope]."""
    block_size = kv_cache.shape[1]
    blk = slot_mapping // block_size
    off = slot_mapping % block_size
    lora = c_kv.shape[1]
    kv_cache[blk, off, :lora] = c_kv
    kv_cache[blk, off, lora:] = k_pe


def mla_attention(
    q_nope,       # [T, Hq, d_nope]  (already absorbed: d_nope == kv_lora)
    q_pe,         # [T, Hq, d_rope]
    kv_cache,     # [num_blocks, block_size, kv_lora + d_rope]
    block_table,  # [num_reqs, max_blocks] int32
    query_start_loc,  # [num_reqs+1]
    seq_lens,     # [num_reqs]
    scale,
):
    """Attention in the compressed (absorbed) MLA space: scores =
    q_nope . c_kv + q_pe . k_pe; output is in kv_lora space [T, Hq, lora]
    (caller applies W_UV). Causal over each request's own new tokens.

    Role of the reference's MLA backends (vllm/v1/attention/backends/mla/)
    in plain torch — the numerics ground truth for the HIP MLA kernel.
    """
    T, Hq, lora = q_nope.shape
    d_rope = q_pe.shape[2]
    block_size = kv_cache.shape[1]
    out = q_nope.new_empty(T, Hq, lora)
    qs = query_start_loc.tolist()
    num_reqs = seq_lens.shape[0]
    for i in range(num_reqs):
        s, e = qs[i], qs[i + 1]
        ql = e - s
        ctx = int(seq_lens[i])
        nb = (ctx + block_size - 1) // block_size
        blocks = block_table[i, :nb].long()
        kv = kv_cache[blocks].reshape(-1, lora + d_rope)[:ctx].float()
        c_kv, k_pe = kv[:, :lora], kv[:, lora:]
        qn = q_nope[s:e].float()  # [ql, Hq, lora]
        qp = q_pe[s:e].float()
        scores = (
            torch.einsum("qhl,kl->hqk", qn, c_kv)
            + torch.einsum("qhr,kr->hqk", qp, k_pe)
        ) * scale
        kpos = torch.arange(ctx, device=qn.device)
        qpos = ctx - ql + torch.arange(ql, device=qn.device)
        mask = kpos.unsqueeze(0) > qpos.unsqueeze(1)
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        p = scores.softmax(dim=-1)
        o = torch.einsum("hqk,kl->qhl", p, c_kv)
        out[s:e] = o.to(out.dtype)
    return out


def grouped_topk(
    gating, topk, renormalize=True, num_groups=0, topk_groups=0,
    scoring_func="softmax", e_score_bias=None, routed_scaling_factor=1.0,
):
    """DeepSeek-style routing: sigmoid/softmax scoring, optional
    group-limited expert selection (n_group/topk_group), optional
    per-expert score bias (role of fused_moe grouped_topk)."""
    T, E = gating.shape
    if scoring_func == "sigmoid":
        scores = gating.float().sigmoid()
    else:
        scores = gating.float().softmax(dim=-1)
    select = scores if e_score_bias is None else scores + e_score_bias
    if num_groups > 0 and topk_groups > 0:
        g = select.view(T, num_groups, E // num_groups)
        # Group score: sum of top-2 experts inside each group.
        group_score = g.topk(min(2, g.shape[-1]), dim=-1).values.sum(-1)
        keep = group_score.topk(topk_groups, dim=-1).indices
        mask = torch.zeros(T, num_groups, dtype=torch.bool,
                           device=gating.device)
        mask.scatter_(1, keep, True)
        select = select.masked_fill(
            ~mask.unsqueeze(-1).expand_as(g).reshape(T, E), float("-inf")
        )
    topk_ids = select.topk(topk, dim=-1).indices
    topk_weights = scores.gather(1, topk_ids)
    if renormalize:
        topk_weights = topk_weights / topk_weights.sum(-1, keepdim=True)
    topk_weights = topk_weights * routed_scaling_factor
    return topk_weights, topk_ids.to(torch.int32)
