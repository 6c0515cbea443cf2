"""Fused Mixture-of-Experts layer (role of the reference's
vllm/model_executor/layers/fused_moe/layer.py FusedMoE).

TP sharding: each rank holds the full expert set with the intermediate
dim sharded 1/tp (gate and up halves sharded separately inside w13);
forward ends in one RCCL all-reduce over xGMI. EP (experts sharded
across ranks + all-to-all dispatch) layers on later (all2all.py role).
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
    ):
        super().__init__()
        tp = get_tp_world_size()
        assert intermediate_size % tp == 0, (
            f"moe intermediate {intermediate_size} not divisible by tp {tp}"
        )
        self.num_experts = num_experts
        self.top_k = top_k
        self.renormalize = renormalize
        self.activation = activation
        self.i_shard = intermediate_size // tp
        self.gate = ReplicatedLinear(hidden_size, num_experts, bias=False,
                                     dtype=dtype)
        # [E, 2*I/tp, H]: gate rows then up rows, both sharded.
        self.w13 = nn.Parameter(
            torch.empty(num_experts, 2 * self.i_shard, hidden_size,
                        dtype=dtype),
            requires_grad=False,
        )
        # [E, H, I/tp]
        self.w2 = nn.Parameter(
            torch.empty(num_experts, hidden_size, self.i_shard, dtype=dtype),
            requires_grad=False,
        )

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        router_logits = self.gate(hidden)
        topk_weights, topk_ids = ops.topk_softmax(
            router_logits, self.top_k, renormalize=self.renormalize
        )
        out = ops.fused_moe(
            hidden, self.w13, self.w2, topk_weights, topk_ids,
            activation=self.activation,
        )
        if get_tp_world_size() > 1:
            out = tensor_model_parallel_all_reduce(out)
        return out

    def load_full_weights(self, w1_full, w3_full, w2_full) -> None:
        """Shard full expert weights onto this rank: w1/w3 [E, I, H]
        (gate / up), w2 [E, H, I]."""
        tp, r = get_tp_world_size(), get_tp_rank()
        i = self.i_shard
        self.w13.data[:, :i] = w1_full[:, r * i:(r + 1) * i]
        self.w13.data[:, i:] = w3_full[:, r * i:(r + 1) * i]
        self.w2.data.copy_(w2_full[:, :, r * i:(r + 1) * i])
