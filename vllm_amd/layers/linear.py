"""TP-sharded linear layers.

Role of the reference's vllm/model_executor/layers/linear.py (Column/
Merged/QKV/RowParallel): weights are sharded at construction time; a
RowParallelLinear forward ends in one RCCL all-reduce over xGMI — the
dominant collective of TP decode (2 per decoder layer).

On GPU, GEMMs go through the tuned hipBLASLt op (ops.linear — per-shape
algo search, csrc/gemm_hipblaslt.cpp); on CPU, torch F.linear.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

from vllm_amd import ops

from vllm_amd.parallel.state import (
    get_tp_rank,
    get_tp_world_size,
    tensor_model_parallel_all_gather,
    tensor_model_parallel_all_reduce,
)


def _run_gemm(module, x, bias):
    """Dispatch to the fp8 W8A8 path when this layer has been quantized
    (registry.quantize_model_fp8), else the bf16/fp16 tuned GEMM."""
    w8 = getattr(module, "weight_fp8", None)
    if w8 is not None:
        return ops.linear_fp8(x, w8, module.weight_scale, bias)
    return ops.linear(x, module.weight, bias)


def _quantize_fp8(module):
    """Convert this layer's weight to e4m3 + per-channel scales and free
    the wide copy (half the weight HBM traffic and footprint)."""
    w8, scale = ops.quantize_weight_fp8(module.weight.data)
    module.register_buffer("weight_fp8", w8)
    module.register_buffer("weight_scale", scale)
    module.weight.data = module.weight.data.new_empty(0)


def _maybe_apply_lora(module, x, y):
    """Add per-request LoRA deltas when the forward context carries
    adapter ids and this layer has registered slices (lora.py)."""
    slices = getattr(module, "lora_slices", None)
    if slices is None:
        return y
    from vllm_amd.worker.forward_context import get_forward_context

    try:
        ctx = get_forward_context()
    except Exception:  # outside a model step (e.g. unit use)
        return y
    if ctx is None or ctx.lora_manager is None or ctx.lora_ids is None:
        return y
    from vllm_amd.lora import apply_lora_slices

    x2 = x.reshape(-1, x.shape[-1])
    y2 = y.reshape(-1, y.shape[-1])
    apply_lora_slices(x2, y2, ctx.lora_ids, ctx.lora_manager, slices)
    return y


class ReplicatedLinear(nn.Module):
    def __init__(self, input_size, output_size, bias=False, dtype=None):
        super().__init__()
        self.weight = nn.Parameter(
            torch.empty(output_size, input_size, dtype=dtype),
            requires_grad=False,
        )
        self.bias = (
            nn.Parameter(torch.empty(output_size, dtype=dtype),
                         requires_grad=False)
            if bias
            else None
        )

    def forward(self, x):
        return _run_gemm(self, x, self.bias)

    quantize_fp8 = _quantize_fp8


class ColumnParallelLinear(nn.Module):
    """Output-dim sharded: Y_local = X @ W_local^T. No communication in
    forward (unless gather_output)."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        gather_output: bool = False,
        dtype: Optional[torch.dtype] = None,
    ):
        super().__init__()
        tp = get_tp_world_size()
        assert output_size % tp == 0, f"{output_size} % tp={tp}"
        self.input_size = input_size
        self.output_size = output_size
        self.output_size_per_partition = output_size // tp
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.output_size_per_partition, input_size,
                        dtype=dtype),
            requires_grad=False,
        )
        self.bias = (
            nn.Parameter(
                torch.empty(self.output_size_per_partition, dtype=dtype),
                requires_grad=False,
            )
            if bias
            else None
        )

    def forward(self, x):
        y = _run_gemm(self, x, self.bias)
        y = _maybe_apply_lora(self, x, y)
        if self.gather_output:
            y = tensor_model_parallel_all_gather(y, dim=-1)
        return y

    quantize_fp8 = _quantize_fp8

    def load_weight(self, full_weight: torch.Tensor) -> None:
        tp, rank = get_tp_world_size(), get_tp_rank()
        shard = full_weight.chunk(tp, dim=0)[rank]
        self.weight.data.copy_(shard)

    def load_bias(self, full_bias: torch.Tensor) -> None:
        tp, rank = get_tp_world_size(), get_tp_rank()
        self.bias.data.copy_(full_bias.chunk(tp, dim=0)[rank])

    # Dummy-init partition invariance: the initializer generates the
    # FULL-shape tensor (name-seeded) and each rank copies its shard
    # through the same mapping the checkpoint loaders use, so tp=N
    # dummy weights are exact slices of the tp=1 weights.
    def dummy_shard_shapes(self) -> dict:
        d = {"weight": (self.output_size, self.input_size)}
        if self.bias is not None:
            d["bias"] = (self.output_size,)
        return d

    def dummy_shard(self, pname: str, full: torch.Tensor) -> torch.Tensor:
        tp, rank = get_tp_world_size(), get_tp_rank()
        return full.chunk(tp, dim=0)[rank]


class MergedColumnParallelLinear(ColumnParallelLinear):
    """Several column-parallel projections fused into one GEMM
    (e.g. gate_proj + up_proj). Each sub-output is sharded separately."""

    def __init__(self, input_size, output_sizes: list[int], bias=False,
                 dtype=None):
        super().__init__(input_size, sum(output_sizes), bias=bias, dtype=dtype)
        self.output_sizes = output_sizes

    def load_sub_weight(self, idx: int, full_weight: torch.Tensor) -> None:
        tp, rank = get_tp_world_size(), get_tp_rank()
        offset = sum(self.output_sizes[:idx]) // tp
        size = self.output_sizes[idx] // tp
        shard = full_weight.chunk(tp, dim=0)[rank]
        self.weight.data[offset : offset + size].copy_(shard)

    def dummy_shard(self, pname: str, full: torch.Tensor) -> torch.Tensor:
        tp, rank = get_tp_world_size(), get_tp_rank()
        parts = full.split(self.output_sizes, dim=0)
        return torch.cat([p.chunk(tp, dim=0)[rank] for p in parts],
                         dim=0)


class QKVParallelLinear(ColumnParallelLinear):
    """Fused QKV projection, head-aware sharding. KV heads are replicated
    when tp > num_kv_heads."""

    def __init__(
        self,
        hidden_size: int,
        head_dim: int,
        num_heads: int,
        num_kv_heads: int,
        bias: bool = False,
        dtype: Optional[torch.dtype] = None,
    ):
        tp = get_tp_world_size()
        self.hidden_size = hidden_size
        self.head_dim = head_dim
        self.total_num_heads = num_heads
        self.total_num_kv_heads = num_kv_heads
        self.num_heads = num_heads // tp
        if num_kv_heads >= tp:
            assert num_kv_heads % tp == 0
            self.num_kv_heads = num_kv_heads // tp
            self.num_kv_head_replicas = 1
        else:
            assert tp % num_kv_heads == 0
            self.num_kv_heads = 1
            self.num_kv_head_replicas = tp // num_kv_heads
        output_size = (
            (self.num_heads + 2 * self.num_kv_heads) * head_dim * tp
        )
        super().__init__(hidden_size, output_size, bias=bias, dtype=dtype)
        # Override: output partition is exactly heads for THIS rank.
        self.q_size = self.num_heads * head_dim
        self.kv_size = self.num_kv_heads * head_dim

    def split_qkv(self, qkv: torch.Tensor):
        return qkv.split([self.q_size, self.kv_size, self.kv_size], dim=-1)

    def load_qkv(self, q_w, k_w, v_w, q_b=None, k_b=None, v_b=None) -> None:
        tp, rank = get_tp_world_size(), get_tp_rank()
        q_shard = q_w.chunk(tp, dim=0)[rank]
        kv_rank = rank // self.num_kv_head_replicas
        kv_chunks = max(self.total_num_kv_heads // max(self.num_kv_heads, 1), 1)
        k_shard = k_w.chunk(kv_chunks, dim=0)[kv_rank % kv_chunks]
        v_shard = v_w.chunk(kv_chunks, dim=0)[kv_rank % kv_chunks]
        self.weight.data.copy_(torch.cat([q_shard, k_shard, v_shard], dim=0))
        if q_b is not None and self.bias is not None:
            qb = q_b.chunk(tp, dim=0)[rank]
            kb = k_b.chunk(kv_chunks, dim=0)[kv_rank % kv_chunks]
            vb = v_b.chunk(kv_chunks, dim=0)[kv_rank % kv_chunks]
            self.bias.data.copy_(torch.cat([qb, kb, vb], dim=0))

    def dummy_shard_shapes(self) -> dict:
        rows = (self.total_num_heads
                + 2 * self.total_num_kv_heads) * self.head_dim
        d = {"weight": (rows, self.hidden_size)}
        if self.bias is not None:
            d["bias"] = (rows,)
        return d

    def dummy_shard(self, pname: str, full: torch.Tensor) -> torch.Tensor:
        q_rows = self.total_num_heads * self.head_dim
        kv_rows = self.total_num_kv_heads * self.head_dim
        q, k, v = full.split([q_rows, kv_rows, kv_rows], dim=0)
        tp, rank = get_tp_world_size(), get_tp_rank()
        kv_chunks = max(
            self.total_num_kv_heads // max(self.num_kv_heads, 1), 1)
        kv_rank = (rank // self.num_kv_head_replicas) % kv_chunks
        return torch.cat([q.chunk(tp, dim=0)[rank],
                          k.chunk(kv_chunks, dim=0)[kv_rank],
                          v.chunk(kv_chunks, dim=0)[kv_rank]], dim=0)


class RowParallelLinear(nn.Module):
    """Input-dim sharded: Y = sum_ranks(X_local @ W_local^T) via all-reduce.
    The all-reduce is the per-layer TP collective (reference linear.py:1767)."""

    def __init__(
        self,
        input_size: int,
        output_size: int,
        bias: bool = False,
        reduce_results: bool = True,
        dtype: Optional[torch.dtype] = None,
    ):
        super().__init__()
        tp = get_tp_world_size()
        assert input_size % tp == 0
        self.input_size = input_size
        self.input_size_per_partition = input_size // tp
        self.output_size = output_size
        self.reduce_results = reduce_results
        self.weight = nn.Parameter(
            torch.empty(output_size, self.input_size_per_partition,
                        dtype=dtype),
            requires_grad=False,
        )
        # Bias is added post-reduce on rank 0's contribution only.
        self.bias = (
            nn.Parameter(torch.empty(output_size, dtype=dtype),
                         requires_grad=False)
            if bias
            else None
        )

    def forward(self, x, reduce_results=None):
        y = _run_gemm(self, x, None)
        y = _maybe_apply_lora(self, x, y)
        reduce = (self.reduce_results if reduce_results is None
                  else reduce_results)
        if reduce:
            y = tensor_model_parallel_all_reduce(y)
        if self.bias is not None and (reduce or get_tp_rank() == 0):
            # Partial (sequence-parallel) outputs add the bias once:
            # rank 0's contribution carries it through the later
            # reduce-scatter.
            y = y + self.bias
        return y

    def load_weight(self, full_weight: torch.Tensor) -> None:
        tp, rank = get_tp_world_size(), get_tp_rank()
        shard = full_weight.chunk(tp, dim=1)[rank]
        self.weight.data.copy_(shard)

    def load_bias(self, full_bias: torch.Tensor) -> None:
        self.bias.data.copy_(full_bias)

    def dummy_shard_shapes(self) -> dict:
        d = {"weight": (self.output_size, self.input_size)}
        if self.bias is not None:
            d["bias"] = (self.output_size,)
        return d

    def dummy_shard(self, pname: str, full: torch.Tensor) -> torch.Tensor:
        if pname == "bias":  # bias applied once post-reduce, unsharded
            return full
        tp, rank = get_tp_world_size(), get_tp_rank()
        return full.chunk(tp, dim=1)[rank]

    quantize_fp8 = _quantize_fp8
