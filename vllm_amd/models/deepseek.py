"""DeepSeek-V3-style decoder: MLA attention + sigmoid group-routed MoE
with shared experts (BASELINE config 5).

Structure mirrors the reference's deepseek_v2.py (vllm/model_executor/
models/deepseek_v2.py) and mla.py (model_executor/layers/mla.py:36):

- MLA: per-token KV compressed to kv_lora_rank (512) + rope dim (64) —
  the paged cache stores 576 values/token/layer regardless of head
  count, which is what makes 128-head attention fit 288 GB HBM3E.
- Decode/prefill run in the ABSORBED space: q_nope is folded through
  W_UK so attention works directly on the compressed cache; W_UV is
  applied after (no per-token KV decompression).
- MoE: sigmoid scoring, group-limited top-k (n_group/topk_group),
  routed_scaling_factor, plus dense shared experts; leading
  first_dense_layers use a dense MLP.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd import ops
from vllm_amd.config import ModelConfig, ModelSpec
from vllm_amd.layers.activation import SiluAndMul
from vllm_amd.layers.embedding import ParallelLMHead, VocabParallelEmbedding
from vllm_amd.layers.fused_moe import FusedMoE
from vllm_amd.layers.layernorm import RMSNorm
from vllm_amd.layers.linear import (
    ColumnParallelLinear,
    MergedColumnParallelLinear,
    ReplicatedLinear,
    RowParallelLinear,
)
from vllm_amd.layers.rotary import RotaryEmbedding
from vllm_amd.parallel.state import get_tp_world_size
from vllm_amd.worker.forward_context import get_forward_context


class DeepseekMLAAttention(nn.Module):

    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position):
        super().__init__()
        tp = get_tp_world_size()
        assert spec.num_heads % tp == 0
        self.layer_idx = layer_idx
        self.num_heads = spec.num_heads // tp
        self.qk_nope = spec.qk_nope_head_dim
        self.qk_rope = spec.qk_rope_head_dim
        self.v_head_dim = spec.v_head_dim
        self.kv_lora = spec.kv_lora_rank
        self.qk_head_dim = self.qk_nope + self.qk_rope
        self.scale = self.qk_head_dim**-0.5

        if spec.q_lora_rank > 0:
            self.q_a_proj = ReplicatedLinear(spec.hidden_size,
                                             spec.q_lora_rank, dtype=dtype)
            self.q_a_layernorm = RMSNorm(spec.q_lora_rank, spec.rms_norm_eps,
                                         dtype=dtype)
            self.q_b_proj = ColumnParallelLinear(
                spec.q_lora_rank, spec.num_heads * self.qk_head_dim,
                dtype=dtype,
            )
        else:
            self.q_a_proj = None
            self.q_proj = ColumnParallelLinear(
                spec.hidden_size, spec.num_heads * self.qk_head_dim,
                dtype=dtype,
            )
        # Compressed KV projection (replicated; output is per-token).
        self.kv_a_proj_with_mqa = ReplicatedLinear(
            spec.hidden_size, self.kv_lora + self.qk_rope, dtype=dtype
        )
        self.kv_a_layernorm = RMSNorm(self.kv_lora, spec.rms_norm_eps,
                                      dtype=dtype)
        # Decompression weights, stored pre-split per local head for the
        # absorbed compute: W_UK [h, nope, lora], W_UV [h, lora, v].
        self.w_uk = nn.Parameter(
            torch.empty(self.num_heads, self.qk_nope, self.kv_lora,
                        dtype=dtype), requires_grad=False)
        self.w_uv = nn.Parameter(
            torch.empty(self.num_heads, self.kv_lora, self.v_head_dim,
                        dtype=dtype), requires_grad=False)
        self.o_proj = RowParallelLinear(
            spec.num_heads * self.v_head_dim, spec.hidden_size, dtype=dtype
        )
        self.rotary_emb = RotaryEmbedding(
            self.qk_rope, self.qk_rope, max_position, theta=spec.rope_theta,
            rope_scaling=spec.rope_scaling,
        )

    def load_kv_b_proj(self, kv_b_weight: torch.Tensor) -> None:
        """kv_b_proj.weight [Hq*(nope+v), lora] (full) -> absorbed splits
        for this rank's heads."""
        tp = get_tp_world_size()
        from vllm_amd.parallel.state import get_tp_rank

        full_heads = kv_b_weight.shape[0] // (self.qk_nope + self.v_head_dim)
        w = kv_b_weight.view(full_heads, self.qk_nope + self.v_head_dim,
                             self.kv_lora)
        w = w.chunk(tp, dim=0)[get_tp_rank()]
        self.w_uk.data.copy_(w[:, : self.qk_nope])
        self.w_uv.data.copy_(w[:, self.qk_nope:].transpose(1, 2))

    def forward(self, positions, hidden):
        T = hidden.shape[0]
        if self.q_a_proj is not None:
            q = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(hidden)))
        else:
            q = self.q_proj(hidden)
        q = q.view(T, self.num_heads, self.qk_head_dim)
        q_nope, q_pe = q[..., : self.qk_nope], q[..., self.qk_nope:]

        kv_a = self.kv_a_proj_with_mqa(hidden)
        c_kv = self.kv_a_layernorm(kv_a[:, : self.kv_lora].contiguous())
        k_pe = kv_a[:, self.kv_lora:].unsqueeze(1)  # [T, 1, rope]

        q_pe = q_pe.contiguous()
        k_pe = k_pe.contiguous()
        self.rotary_emb(positions, q_pe, k_pe)

        # Absorb W_UK: q' [T, h, lora].
        q_absorbed = torch.einsum(
            "thn,hnl->thl", q_nope.float(), self.w_uk.float()
        ).to(hidden.dtype)

        ctx = get_forward_context()
        meta = ctx.attn_metadata
        if not ctx.kv_caches:
            o_c = self._profile_attention(q_absorbed, q_pe, c_kv,
                                          k_pe.squeeze(1), meta)
        else:
            kv_cache = ctx.kv_caches[self.layer_idx]
            ops.concat_and_cache_mla(c_kv, k_pe.squeeze(1), kv_cache,
                                     meta.slot_mapping)
            o_c = ops.mla_attention(
                q_absorbed, q_pe, kv_cache, meta.block_table,
                meta.query_start_loc, meta.seq_lens, self.scale,
                num_decodes=meta.num_decodes, max_seq_len=meta.max_seq_len,
            )
        # Un-absorb W_UV: [T, h, v].
        out = torch.einsum("thl,hlv->thv", o_c.float(),
                           self.w_uv.float()).to(hidden.dtype)
        return self.o_proj(out.reshape(T, -1))

    def _profile_attention(self, q_absorbed, q_pe, c_kv, k_pe, meta):
        """Cache-less MLA attention over the step's own tokens (memory
        profiling run)."""
        out = q_absorbed.new_empty(q_absorbed.shape)
        qs = meta.query_start_loc.tolist()
        for i in range(meta.num_reqs):
            s, e = qs[i], qs[i + 1]
            qn = q_absorbed[s:e].float()
            qp = q_pe[s:e].float()
            ck = c_kv[s:e].float()
            kp = k_pe[s:e].float()
            ql = e - s
            scores = (
                torch.einsum("qhl,kl->hqk", qn, ck)
                + torch.einsum("qhr,kr->hqk", qp, kp)
            ) * self.scale
            pos = torch.arange(ql, device=qn.device)
            mask = pos.unsqueeze(0) > pos.unsqueeze(1)
            scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
            p = scores.softmax(dim=-1)
            out[s:e] = torch.einsum("hqk,kl->qhl", p, ck).to(out.dtype)
        return out


class DeepseekDenseMLP(nn.Module):
    def __init__(self, hidden: int, intermediate: int, dtype):
        super().__init__()
        self.gate_up_proj = MergedColumnParallelLinear(
            hidden, [intermediate, intermediate], dtype=dtype
        )
        self.down_proj = RowParallelLinear(intermediate, hidden, dtype=dtype)
        self.act_fn = SiluAndMul()

    def forward(self, x):
        return self.down_proj(self.act_fn(self.gate_up_proj(x)))


class DeepseekMoE(nn.Module):
    """Routed experts (sigmoid + group-limited top-k) + shared experts."""

    def __init__(self, spec: ModelSpec, dtype, ep: bool = False,
                 eplb_window: int = 0):
        super().__init__()
        self.spec = spec
        self.moe = FusedMoE(
            num_experts=spec.num_experts,
            top_k=spec.num_experts_per_tok,
            hidden_size=spec.hidden_size,
            intermediate_size=spec.moe_intermediate_size,
            renormalize=spec.norm_topk_prob,
            dtype=dtype,
            enable_expert_parallel=ep,
            eplb_window=eplb_window,
        )
        # DeepSeek-V3 aux-loss-free balancing bias (inference: applied to
        # selection only).
        self.e_score_correction_bias = nn.Parameter(
            torch.zeros(spec.num_experts, dtype=torch.float32),
            requires_grad=False,
        )
        self.shared_experts = (
            DeepseekDenseMLP(
                spec.hidden_size,
                spec.moe_intermediate_size * spec.num_shared_experts,
                dtype,
            )
            if spec.num_shared_experts > 0
            else None
        )

    def forward(self, hidden):
        spec = self.spec
        router_logits = self.moe.gate(hidden)
        topk_weights, topk_ids = ops.grouped_topk(
            router_logits, spec.num_experts_per_tok,
            renormalize=spec.norm_topk_prob,
            num_groups=spec.n_group, topk_groups=spec.topk_group,
            scoring_func=spec.scoring_func,
            e_score_bias=self.e_score_correction_bias,
            routed_scaling_factor=spec.routed_scaling_factor,
        )
        out = self.moe.run_experts(hidden, topk_weights, topk_ids)
        if self.shared_experts is not None:
            out = out + self.shared_experts(hidden)
        from vllm_amd.parallel.state import (
            get_tp_world_size, tensor_model_parallel_all_reduce)

        if get_tp_world_size() > 1:
            out = tensor_model_parallel_all_reduce(out)
        return out


class DeepseekDecoderLayer(nn.Module):
    def __init__(self, spec: ModelSpec, layer_idx: int, dtype, max_position,
                 ep: bool = False, eplb_window: int = 0):
        super().__init__()
        self.self_attn = DeepseekMLAAttention(spec, layer_idx, dtype,
                                              max_position)
        if layer_idx < spec.first_dense_layers:
            self.mlp = DeepseekDenseMLP(spec.hidden_size,
                                        spec.intermediate_size, dtype)
        else:
            self.mlp = DeepseekMoE(spec, dtype, ep=ep,
                                   eplb_window=eplb_window)
        self.input_layernorm = RMSNorm(spec.hidden_size, spec.rms_norm_eps,
                                       dtype=dtype)
        self.post_attention_layernorm = RMSNorm(
            spec.hidden_size, spec.rms_norm_eps, dtype=dtype
        )

    def forward(self, positions, hidden, residual):
        if residual is None:
            residual = hidden
            hidden = self.input_layernorm(hidden)
        else:
            hidden, residual = self.input_layernorm(hidden, residual)
        hidden = self.self_attn(positions, hidden)
        hidden, residual = self.post_attention_layernorm(hidden, residual)
        hidden = self.mlp(hidden)
        return hidden, residual


class DeepseekModel(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        dtype = config.torch_dtype
        self.embed_tokens = VocabParallelEmbedding(
            spec.vocab_size, spec.hidden_size, dtype=dtype
        )
        self.layers = nn.ModuleList([
            DeepseekDecoderLayer(spec, i, dtype, config.max_model_len,
                                 ep=config.enable_expert_parallel,
                                eplb_window=config.eplb_window)
            for i in range(spec.num_layers)
        ])
        self.norm = RMSNorm(spec.hidden_size, spec.rms_norm_eps, dtype=dtype)

    def forward(self, input_ids, positions):
        hidden = self.embed_tokens(input_ids)
        residual = None
        for layer in self.layers:
            hidden, residual = layer(positions, hidden, residual)
        hidden, _ = self.norm(hidden, residual)
        return hidden


class DeepseekForCausalLM(nn.Module):
    def __init__(self, config: ModelConfig):
        super().__init__()
        spec = config.spec
        self.config = config
        self.model = DeepseekModel(config)
        self.lm_head = ParallelLMHead(
            spec.vocab_size, spec.hidden_size, dtype=config.torch_dtype
        )

    def forward(self, input_ids, positions):
        return self.model(input_ids, positions)

    def compute_logits(self, hidden):
        return self.lm_head.compute_logits(hidden)
