"""Ref-counted block pool with prefix cache (role of vllm/v1/core/block_pool.py:143)."""

from __future__ import annotations

from typing import Optional

from vllm_amd.core.kv_cache_utils import (
    BlockHash,
    FreeKVCacheBlockQueue,
    KVCacheBlock,
)


class BlockPool:
    """Owns all KV cache blocks of the GPU.

    Blocks are ref-counted. Free blocks live in an LRU queue but may keep
    their content hash, making them a prefix cache: a lookup hit revives a
    free block (touch) instead of recomputing its KV.
    """

    def __init__(self, num_gpu_blocks: int, enable_caching: bool = True,
                 on_evict=None) -> None:
        assert num_gpu_blocks > 0
        self.num_gpu_blocks = num_gpu_blocks
        self.enable_caching = enable_caching
        # Called as on_evict(block_hash, block_id) when a content-cached
        # free block is about to be reused (KV offload hook).
        self.on_evict = on_evict
        self.blocks: list[KVCacheBlock] = [
            KVCacheBlock(block_id=i) for i in range(num_gpu_blocks)
        ]
        self.free_block_queue = FreeKVCacheBlockQueue(self.blocks)
        # hash value -> block holding that content (at most one; the latest
        # fully-cached block wins).
        self.cached_block_hash_to_block: dict[int, KVCacheBlock] = {}

    def get_num_free_blocks(self) -> int:
        return self.free_block_queue.num_free_blocks

    def get_usage(self) -> float:
        return 1.0 - self.get_num_free_blocks() / self.num_gpu_blocks

    def get_cached_block(self, block_hash: BlockHash) -> Optional[KVCacheBlock]:
        block = self.cached_block_hash_to_block.get(block_hash.value)
        if block is None:
            return None
        # Guard against int-hash collisions.
        if block.block_hash is None or block.block_hash.token_ids != block_hash.token_ids:
            return None
        return block

    def touch(self, blocks: list[KVCacheBlock]) -> None:
        """Revive cache-hit blocks: bump refs, pull free ones off the queue."""
        for block in blocks:
            if block.ref_cnt == 0:
                self.free_block_queue.remove(block)
            block.ref_cnt += 1

    def get_new_blocks(self, num_blocks: int) -> list[KVCacheBlock]:
        if num_blocks > self.get_num_free_blocks():
            raise ValueError("Cannot get more blocks than free blocks")
        out: list[KVCacheBlock] = []
        for _ in range(num_blocks):
            block = self.free_block_queue.popleft()
            # Evict stale cache entry if this block was a cached free block.
            if block.block_hash is not None:
                cached = self.cached_block_hash_to_block.get(block.block_hash.value)
                if cached is block:
                    del self.cached_block_hash_to_block[block.block_hash.value]
                    if self.on_evict is not None:
                        self.on_evict(block.block_hash, block.block_id)
                block.reset_hash()
            block.ref_cnt = 1
            out.append(block)
        return out

    def cache_full_blocks(
        self,
        blocks: list[KVCacheBlock],
        block_hashes: list[BlockHash],
        num_cached_blocks: int,
        num_full_blocks: int,
    ) -> None:
        """Register content hashes for blocks [num_cached, num_full) of a request."""
        if not self.enable_caching:
            return
        for i in range(num_cached_blocks, num_full_blocks):
            block = blocks[i]
            if block.block_hash is not None:
                continue  # already cached (e.g. shared prefix block)
            h = block_hashes[i]
            block.block_hash = h
            self.cached_block_hash_to_block[h.value] = block

    def uncache(self, block: KVCacheBlock) -> None:
        """Drop a block's prefix-cache registration WITHOUT eviction
        side effects (no host-tier save). Used by sliding-window
        reclaim: a window-reclaimed block's content is semantically dead
        — a later prefix hit on it would resume from KV the window
        model can no longer extend (and the memory may be reused)."""
        if block.block_hash is None:
            return
        cached = self.cached_block_hash_to_block.get(block.block_hash.value)
        if cached is block:
            del self.cached_block_hash_to_block[block.block_hash.value]
        block.reset_hash()

    def free_blocks(self, ordered_blocks: list[KVCacheBlock]) -> None:
        """Deref blocks; zero-ref blocks go to the free queue in the given
        order (callers pass eviction-preference order: tail blocks first)."""
        for block in ordered_blocks:
            block.ref_cnt -= 1
            assert block.ref_cnt >= 0, f"double free of {block}"
            if block.ref_cnt == 0:
                self.free_block_queue.append(block)

    def reset_prefix_cache(self) -> bool:
        if self.get_num_free_blocks() != self.num_gpu_blocks:
            return False  # in-use blocks present; refuse
        self.cached_block_hash_to_block.clear()
        for block in self.blocks:
            block.reset_hash()
        return True
