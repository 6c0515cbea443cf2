"""Fused Mixture-of-Experts layer (role of the reference's
vllm/model_executor/layers/fused_moe/layer.py FusedMoE).

Two sharding modes over the same forward:

- TP (default): every rank holds all experts with the intermediate dim
  sharded 1/tp (gate/up halves sharded separately inside w13).
- EP (enable_expert_parallel): experts sharded across ranks at full
  intermediate width; tokens are replicated in SPMD TP serving, so each
  rank computes its local experts for every token and the combine is
  the same RCCL all-reduce (the naive AgRs manager of the reference's
  all2all.py — DeepEP-style dispatch/combine lands later).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from vllm_amd import ops
from vllm_amd.layers.linear import ReplicatedLinear
from vllm_amd.parallel.state import (
    get_tp_rank,
    get_tp_world_size,
    tensor_model_parallel_all_reduce,
)


class FusedMoE(nn.Module):

    def __init__(
        self,
        num_experts: int,
        top_k: int,
        hidden_size: int,
        intermediate_size: int,
        renormalize: bool = True,
        activation: str = "silu",
        dtype: torch.dtype = None,
        enable_expert_parallel: bool = False,
    ):
        super().__init__()
        from vllm_amd.config import EngineConfig  # noqa: F401 (doc only)
        from vllm_amd.parallel.state import get_ep_group

        tp = get_tp_world_size()
        self.ep_size = 1
        self.ep_rank = 0
        if enable_expert_parallel and tp > 1:
            ep = get_ep_group()
            self.ep_size = ep.world_size
            self.ep_rank = ep.rank_in_group
        if self.ep_size > 1:
            assert num_experts % self.ep_size == 0, (
                f"{num_experts} experts not divisible by ep {self.ep_size}")
            self.num_local_experts = num_experts // self.ep_size
            self.expert_lo = self.ep_rank * self.num_local_experts
            self.i_shard = intermediate_size  # full width per local expert
        else:
            assert intermediate_size % tp == 0, (
                f"moe intermediate {intermediate_size} not divisible by "
                f"tp {tp}")
            self.num_local_experts = num_experts
            self.expert_lo = 0
            self.i_shard = intermediate_size // tp
        self.num_experts = num_experts
        self.top_k = top_k
        self.renormalize = renormalize
        self.activation = activation
        self.gate = ReplicatedLinear(hidden_size, num_experts, bias=False,
                                     dtype=dtype)
        # TP: [E, 2*I/tp, H]; EP: [E/ep, 2*I, H].
        self.w13 = nn.Parameter(
            torch.empty(self.num_local_experts, 2 * self.i_shard,
                        hidden_size, dtype=dtype),
            requires_grad=False,
        )
        self.w2 = nn.Parameter(
            torch.empty(self.num_local_experts, hidden_size, self.i_shard,
                        dtype=dtype),
            requires_grad=False,
        )
        # Fragment-major weight copies for the HIP grouped GEMM (built
        # lazily on the first GPU forward; freed by invalidate_shuffled
        # when weights change). Costs a second copy of the expert
        # weights — the MI355X trade: 288 GB HBM buys coalesced,
        # barrier-free weight streams. VLLM_AMD_MOE_SHUF=0 disables.
        self._w13_shuf = None
        self._w2_shuf = None

    def refresh_shuffled(self) -> None:
        """Call after in-place weight updates (RL update_weights):
        re-shuffles INTO THE EXISTING storage so captured hipGraphs that
        reference the fragment-major tensors stay valid."""
        if self._w13_shuf is None:
            return
        from vllm_amd.ops import get_backend
        backend = get_backend(self.w13.device)
        self._w13_shuf.copy_(backend.moe_shuffle_weights(self.w13.data))
        self._w2_shuf.copy_(backend.moe_shuffle_weights(self.w2.data))

    def _maybe_shuffled(self, hidden):
        import os

        # Fragment-major streaming is opt-in (VLLM_AMD_MOE_SHUF=1): on
        # MI355X the staged-LDS grouped GEMM measured faster at decode
        # batch >= 256 (the barrier-free stream wins only at tiny
        # batches; see profiles/r02_summary.md).
        if (not hidden.is_cuda
                or os.environ.get("VLLM_AMD_MOE_SHUF", "0") != "1"):
            return None, None
        if self._w13_shuf is None:
            from vllm_amd.ops import get_backend
            backend = get_backend(hidden.device)
            shuffle = getattr(backend, "moe_shuffle_weights", None)
            if shuffle is None or not backend._moe_hip_ok(
                    hidden, self.w13, self.w2, self.activation):
                return None, None
            self._w13_shuf = shuffle(self.w13.data)
            self._w2_shuf = shuffle(self.w2.data)
        return self._w13_shuf, self._w2_shuf

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        router_logits = self.gate(hidden)
        topk_weights, topk_ids = ops.topk_softmax(
            router_logits, self.top_k, renormalize=self.renormalize
        )
        out = self.run_experts(hidden, topk_weights, topk_ids)
        if get_tp_world_size() > 1:
            out = tensor_model_parallel_all_reduce(out)
        return out

    def run_experts(self, hidden, topk_weights, topk_ids) -> torch.Tensor:
        """Local expert computation. EP mode: activations are replicated
        across ranks (SPMD TP serving), so each rank COMPACTS to the
        (token, slot) pairs routed to its local experts — no wasted
        expert FLOPs on foreign slots — and the caller's all-reduce
        completes the combine. (A DeepEP-style a2a dispatch only pays
        once activations are DP/sequence-sharded — a round-2 item
        together with DP attention for MoE.)"""
        if self.ep_size > 1:
            lo = self.expert_lo
            hi = lo + self.num_local_experts
            mask = (topk_ids >= lo) & (topk_ids < hi)
            slot_tok, slot_k = mask.nonzero(as_tuple=True)
            if slot_tok.numel() == 0:
                return torch.zeros_like(hidden)
            sel_ids = (topk_ids[slot_tok, slot_k] - lo).unsqueeze(1)
            sel_w = topk_weights[slot_tok, slot_k].unsqueeze(1)
            w13s, w2s = self._maybe_shuffled(hidden)
            y = ops.fused_moe(
                hidden[slot_tok], self.w13, self.w2, sel_w, sel_ids,
                activation=self.activation, w13_shuf=w13s, w2_shuf=w2s,
            )
            out = torch.zeros(hidden.shape, dtype=torch.float32,
                              device=hidden.device)
            out.index_add_(0, slot_tok, y.float())
            return out.to(hidden.dtype)
        w13s, w2s = self._maybe_shuffled(hidden)
        return ops.fused_moe(
            hidden, self.w13, self.w2, topk_weights, topk_ids,
            activation=self.activation, w13_shuf=w13s, w2_shuf=w2s,
        )

    def load_full_weights(self, w1_full, w3_full, w2_full) -> None:
        """Shard full expert weights onto this rank: w1/w3 [E, I, H]
        (gate / up), w2 [E, H, I]."""
        i = self.i_shard
        if self.ep_size > 1:
            lo, hi = self.expert_lo, self.expert_lo + self.num_local_experts
            self.w13.data[:, :i] = w1_full[lo:hi]
            self.w13.data[:, i:] = w3_full[lo:hi]
            self.w2.data.copy_(w2_full[lo:hi])
            return
        tp, r = get_tp_world_size(), get_tp_rank()
        self.w13.data[:, :i] = w1_full[:, r * i:(r + 1) * i]
        self.w13.data[:, i:] = w3_full[:, r * i:(r + 1) * i]
        self.w2.data.copy_(w2_full[:, :, r * i:(r + 1) * i])
