"""Multi-LoRA serving (role of the reference's vllm/lora/): per-request
low-rank adapters applied inside the parallel linear layers.

Torch-composed segment matmuls (rows grouped by adapter; rank r is
small so the A/B GEMMs are cheap), TP-aware: column-parallel layers take
this rank's rows of B, row-parallel layers this rank's columns of A
(partial deltas sum in the layer's all-reduce). HF PEFT checkpoint
layout (adapter_config.json + adapter_model.safetensors) or in-memory
tensors. Adapters are served under their registered name: an OpenAI
request whose `model` equals an adapter name runs with that adapter.
"""

from __future__ import annotations

import json
import os
from typing import Optional

import torch


class LoRAAdapter:
    """One adapter: module_key -> (A [r, in], B [out, r]); fused-layer
    slices are resolved by the owning linear layer."""

    def __init__(self, name: str, rank: int, scaling: float,
                 weights: dict[str, tuple[torch.Tensor, torch.Tensor]]):
        self.name = name
        self.rank = rank
        self.scaling = scaling
        self.weights = weights  # e.g. "model.layers.0.self_attn.q_proj"

    @classmethod
    def from_path(cls, name: str, path: str, dtype, device) -> "LoRAAdapter":
        from safetensors import safe_open

        with open(os.path.join(path, "adapter_config.json")) as f:
            cfg = json.load(f)
        rank = cfg["r"]
        scaling = cfg.get("lora_alpha", rank) / rank
        pairs: dict[str, dict[str, torch.Tensor]] = {}
        st = os.path.join(path, "adapter_model.safetensors")
        with safe_open(st, framework="pt", device="cpu") as sf:
            for key in sf.keys():
                # base_model.model.model.layers.N.self_attn.q_proj.lora_A.weight
                t = sf.get_tensor(key).to(dtype)
                core = key.replace("base_model.model.", "")
                core = core.replace(".weight", "")
                if core.endswith(".lora_A"):
                    pairs.setdefault(core[: -len(".lora_A")], {})["A"] = t
                elif core.endswith(".lora_B"):
                    pairs.setdefault(core[: -len(".lora_B")], {})["B"] = t
        weights = {}
        for mod, ab in pairs.items():
            if "A" in ab and "B" in ab:
                weights[mod] = (ab["A"].to(device), ab["B"].to(device))
        return cls(name, rank, scaling, weights)


class LoRAManager:
    """Registered adapters, integer ids (0 = base model / no adapter)."""

    def __init__(self):
        self.adapters: list[Optional[LoRAAdapter]] = [None]
        self.by_name: dict[str, int] = {}

    def register(self, adapter: LoRAAdapter) -> int:
        lora_id = len(self.adapters)
        self.adapters.append(adapter)
        self.by_name[adapter.name] = lora_id
        return lora_id

    def id_of(self, name: Optional[str]) -> int:
        if name is None:
            return 0
        if name not in self.by_name:
            raise ValueError(f"unknown LoRA adapter {name!r}; "
                             f"registered: {sorted(self.by_name)}")
        return self.by_name[name]

    def __len__(self):
        return len(self.adapters) - 1

    def get(self, lora_id: int) -> Optional[LoRAAdapter]:
        return self.adapters[lora_id]


def apply_lora_slices(
    x: torch.Tensor,
    out: torch.Tensor,
    lora_ids: torch.Tensor,  # [T] int32/int64; 0 = none
    manager: LoRAManager,
    # (module_key, out_off_local, out_len, b_off_full, a_off, a_len):
    # column-parallel layers shard B rows (b_off_full selects this
    # rank's slice of the FULL adapter B); row-parallel layers shard A
    # columns (a_off/a_len select this rank's input slice — the partial
    # deltas sum in the layer's existing all-reduce).
    slices: list[tuple],
) -> None:
    """out[rows, off:off+len] += scaling * (x A^T) B^T per adapter, per
    fused-output slice. Rows are grouped per adapter id (few adapters
    per batch). TP-aware: adapters store FULL A/B; shards are views."""
    unique = torch.unique(lora_ids)
    for lid_t in unique:
        lid = int(lid_t)
        if lid == 0:
            continue
        adapter = manager.get(lid)
        rows = (lora_ids == lid_t).nonzero(as_tuple=True)[0]
        xi = x[rows]
        for key, off, length, b_off, a_off, a_len in slices:
            ab = adapter.weights.get(key)
            if ab is None:
                continue
            A, B = ab
            if a_len is not None:
                A = A[:, a_off:a_off + a_len]
            B = B[b_off:b_off + length]
            delta = (xi @ A.t()) @ B.t() * adapter.scaling
            out[rows, off:off + length] += delta.to(out.dtype)


def attach_lora_metadata(model: torch.nn.Module) -> None:
    """Walk the model and record, per parallel linear, the HF module keys
    and fused-output slices LoRA deltas apply to."""
    from vllm_amd.layers.linear import (
        ColumnParallelLinear, MergedColumnParallelLinear,
        QKVParallelLinear, RowParallelLinear,
    )

    from vllm_amd.parallel.state import get_tp_rank

    rank = get_tp_rank()
    for name, mod in model.named_modules():
        if isinstance(mod, QKVParallelLinear):
            base = name.rsplit(".", 1)[0]
            qs = mod.num_heads * mod.head_dim      # per-rank sizes
            ks = mod.num_kv_heads * mod.head_dim
            kv_rank = rank // mod.num_kv_head_replicas
            kv_chunks = max(mod.total_num_kv_heads
                            // max(mod.num_kv_heads, 1), 1)
            mod.lora_slices = [
                (f"{base}.q_proj", 0, qs, rank * qs, None, None),
                (f"{base}.k_proj", qs, ks,
                 (kv_rank % kv_chunks) * ks, None, None),
                (f"{base}.v_proj", qs + ks, ks,
                 (kv_rank % kv_chunks) * ks, None, None),
            ]
        elif isinstance(mod, MergedColumnParallelLinear):
            base = name.rsplit(".", 1)[0]
            half = mod.weight.shape[0] // 2        # per-rank half
            mod.lora_slices = [
                (f"{base}.gate_proj", 0, half, rank * half, None, None),
                (f"{base}.up_proj", half, half, rank * half, None, None),
            ]
        elif isinstance(mod, RowParallelLinear):
            in_len = mod.input_size_per_partition
            mod.lora_slices = [
                (name, 0, mod.weight.shape[0], 0, rank * in_len, in_len)]
        elif isinstance(mod, ColumnParallelLinear):
            out_len = mod.weight.shape[0]
            mod.lora_slices = [
                (name, 0, out_len, rank * out_len, None, None)]
