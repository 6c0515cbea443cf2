"""Vocab-parallel embedding and logits head.

Reference roles: vocab_parallel_embedding.py (all-reduce after masked
local lookup when TP>1) and logits_processor.py:92 (all-gather of vocab
shards after the LM-head GEMM).
"""

from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from vllm_amd.parallel.state import (
    get_tp_rank,
    get_tp_world_size,
    tensor_model_parallel_all_gather,
    tensor_model_parallel_all_reduce,
)


class VocabParallelEmbedding(nn.Module):
    def __init__(self, num_embeddings: int, embedding_dim: int, dtype=None):
        super().__init__()
        tp = get_tp_world_size()
        rank = get_tp_rank()
        self.num_embeddings = num_embeddings
        # Pad vocab so it divides tp.
        self.num_embeddings_padded = (num_embeddings + tp - 1) // tp * tp
        self.num_embeddings_per_partition = self.num_embeddings_padded // tp
        self.vocab_start = rank * self.num_embeddings_per_partition
        self.vocab_end = self.vocab_start + self.num_embeddings_per_partition
        self.weight = nn.Parameter(
            torch.empty(
                self.num_embeddings_per_partition, embedding_dim, dtype=dtype
            ),
            requires_grad=False,
        )

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        if get_tp_world_size() == 1:
            return F.embedding(input_ids, self.weight)
        mask = (input_ids >= self.vocab_start) & (input_ids < self.vocab_end)
        local_ids = (input_ids - self.vocab_start).clamp_(min=0)
        local_ids[~mask] = 0
        emb = F.embedding(local_ids, self.weight)
        emb[~mask] = 0.0
        return tensor_model_parallel_all_reduce(emb)

    def load_weight(self, full_weight: torch.Tensor) -> None:
        tp = get_tp_world_size()
        if tp == 1:
            self.weight.data[: full_weight.shape[0]].copy_(full_weight)
            return
        pad = self.num_embeddings_padded - full_weight.shape[0]
        if pad:
            full_weight = torch.cat(
                [full_weight,
                 full_weight.new_zeros(pad, full_weight.shape[1])], dim=0
            )
        self.weight.data.copy_(full_weight.chunk(tp, dim=0)[get_tp_rank()])

    def dummy_shard_shapes(self) -> dict:
        return {"weight": (self.num_embeddings, self.weight.shape[1])}

    def dummy_shard(self, pname: str, full: torch.Tensor) -> torch.Tensor:
        tp = get_tp_world_size()
        if tp == 1:
            return full
        pad = self.num_embeddings_padded - full.shape[0]
        if pad:
            full = torch.cat(
                [full, full.new_zeros(pad, full.shape[1])], dim=0)
        return full.chunk(tp, dim=0)[get_tp_rank()]


class ParallelLMHead(VocabParallelEmbedding):
    """LM head sharing the vocab-parallel sharding; logits are gathered."""

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        from vllm_amd import ops
        logits = ops.linear(hidden, self.weight, None)
        if get_tp_world_size() > 1:
            logits = tensor_model_parallel_all_gather(logits, dim=-1)
            logits = logits[..., : self.num_embeddings]
        return logits
