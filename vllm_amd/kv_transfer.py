"""Disaggregated prefill/decode KV transfer primitive.

Role of the reference's kv_transfer / connector layer (P-D
disaggregation): a PREFILL engine computes a prompt's KV once and
exports it; a DECODE engine imports the blocks as content-hashed
prefix-cache entries and generation continues from a full prefix hit —
no recompute beyond the mandatory last token.

The payload is engine-layout-agnostic on the control side (token ids +
per-block tensors); the import path is the same materialize-as-cached-
free-block trick the host-offload tier uses, so the scheduler, block
accounting and kernels need no new states. Transport is the caller's
choice (same process here; bytes over any fabric in a deployment — the
tensors are contiguous and self-describing).
"""

from __future__ import annotations

from dataclasses import dataclass

import torch

from vllm_amd.core.kv_cache_utils import hash_request_tokens


@dataclass
class KVHandoff:
    """One prompt's transferable KV: full blocks only (the tail partial
    block is recomputed by the decode side's chunked prefill)."""

    token_ids: list[int]
    block_size: int
    num_full_blocks: int
    # Per layer: [num_full_blocks, ...block dims] (standard cache
    # [2, H, BS, D] per block; MLA [BS, per_tok]).
    layers: list[torch.Tensor]

    def num_tokens(self) -> int:
        return self.num_full_blocks * self.block_size


def export_prefix_kv(engine_core, token_ids: list[int]) -> KVHandoff:
    """Export the KV of `token_ids`' full blocks from an engine whose
    prefix cache holds them (i.e. after a prefill of that prompt)."""
    sched = engine_core.scheduler
    mgr = sched.kv_cache_manager
    runner = engine_core.worker.runner
    bs = mgr.block_size
    hashes = hash_request_tokens(bs, list(token_ids))
    block_ids = []
    for h in hashes:
        blk = mgr.block_pool.get_cached_block(h)
        if blk is None:
            break
        block_ids.append(blk.block_id)
    if not block_ids:
        raise ValueError("no cached KV for this prompt on the prefill "
                         "engine (run it through prefill first, with "
                         "prefix caching enabled)")
    mla = runner.spec.is_mla
    idx = torch.tensor(block_ids, device=runner.device)
    layers = []
    for cache in runner.kv_caches:
        if mla:
            layers.append(cache.index_select(0, idx).cpu())
        else:
            # cache [2, N, H, BS, D] -> [n_blocks, 2, H, BS, D]
            layers.append(
                cache.index_select(1, idx).transpose(0, 1).cpu())
    return KVHandoff(token_ids=list(token_ids), block_size=bs,
                     num_full_blocks=len(block_ids), layers=layers)


def import_prefix_kv(engine_core, handoff: KVHandoff) -> int:
    """Install a handoff into this engine's prefix cache. Returns the
    number of tokens now cached. A subsequent request with the same
    prompt prefix hits the cache and decodes without recomputing."""
    sched = engine_core.scheduler
    mgr = sched.kv_cache_manager
    runner = engine_core.worker.runner
    if handoff.block_size != mgr.block_size:
        raise ValueError(
            f"block_size mismatch: handoff {handoff.block_size} vs "
            f"engine {mgr.block_size}")
    hashes = hash_request_tokens(mgr.block_size, handoff.token_ids)
    n = min(handoff.num_full_blocks, len(hashes))
    pool = mgr.block_pool
    if pool.get_num_free_blocks() < n:
        raise ValueError("decode engine KV pool too small for handoff")
    blocks = pool.get_new_blocks(n)
    mla = runner.spec.is_mla
    dst = torch.tensor([b.block_id for b in blocks],
                       device=runner.device)
    for cache, src in zip(runner.kv_caches, handoff.layers):
        src_dev = src[:n].to(runner.device, dtype=cache.dtype)
        if mla:
            cache.index_copy_(0, dst, src_dev)
        else:
            cache.index_copy_(1, dst, src_dev.transpose(0, 1))
    # Register as cached free blocks (ref 0, hash set): exactly how a
    # host-tier hit or a finished request's blocks look to the pool.
    for blk, h in zip(blocks, hashes[:n]):
        blk.block_hash = h
        pool.cached_block_hash_to_block[h.value] = blk
    pool.free_blocks(blocks)
    if runner.device.type == "cuda":
        torch.cuda.synchronize(runner.device)
    return n * mgr.block_size
