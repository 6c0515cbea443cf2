"""KV cache block hashing & free-list structures.

Role of the reference's vllm/v1/core/kv_cache_utils.py (hash_block_tokens
:576, FreeKVCacheBlockQueue :184): content-addressed block hashing for
prefix caching and an O(1) LRU free queue with middle-removal.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import NamedTuple, Optional


class BlockHash(NamedTuple):
    """Hash of one full KV block: chains parent hash with the block's tokens
    so a block is only reusable when its whole prefix matches."""

    value: int
    token_ids: tuple[int, ...]  # kept for collision double-check


# Sentinel parent hash for the first block of a sequence.
NONE_HASH = 616101987686463245


def hash_block_tokens(
    parent_hash: Optional[int], token_ids: tuple[int, ...]
) -> BlockHash:
    parent = parent_hash if parent_hash is not None else NONE_HASH
    return BlockHash(hash((parent, token_ids)), token_ids)


def hash_request_tokens(
    block_size: int, token_ids: list[int], start_block: int = 0,
    prior_hashes: Optional[list[BlockHash]] = None,
    salt: Optional[int] = None,
) -> list[BlockHash]:
    """Hash all *full* blocks of a token stream, reusing prior prefix hashes.

    `salt` distinguishes KV contents beyond token ids (LoRA adapter id —
    an adapter's k/v projections change the cached values; role of the
    reference's block-hash extra_keys, kv_cache_utils.py:576).
    Returns the complete list of full-block hashes (prior + new).
    """
    hashes: list[BlockHash] = list(prior_hashes) if prior_hashes else []
    num_full_blocks = len(token_ids) // block_size
    parent = hashes[-1].value if hashes else (
        -salt if salt else None)
    for i in range(len(hashes), num_full_blocks):
        block_tokens = tuple(token_ids[i * block_size : (i + 1) * block_size])
        h = hash_block_tokens(parent, block_tokens)
        hashes.append(h)
        parent = h.value
    return hashes


@dataclass
class KVCacheBlock:
    """One page of the paged KV cache (CPU-side accounting record)."""

    block_id: int
    ref_cnt: int = 0
    # Content hash when this block holds a complete, reusable block.
    block_hash: Optional[BlockHash] = None
    # Doubly-linked free list pointers.
    prev_free_block: Optional["KVCacheBlock"] = None
    next_free_block: Optional["KVCacheBlock"] = None

    def reset_hash(self) -> None:
        self.block_hash = None

    def __repr__(self) -> str:
        return f"KVCacheBlock(id={self.block_id}, ref={self.ref_cnt})"


class FreeKVCacheBlockQueue:
    """Doubly-linked LRU list of free blocks.

    popleft() evicts the least-recently-freed block; remove() supports
    O(1) extraction when a cached free block gets a prefix-cache hit.
    Blocks freed with their hash intact stay lookup-able until evicted.
    """

    def __init__(self, blocks: list[KVCacheBlock]) -> None:
        self.num_free_blocks = 0
        # Sentinel head/tail simplify edge cases.
        self._head = KVCacheBlock(block_id=-1)
        self._tail = KVCacheBlock(block_id=-2)
        self._head.next_free_block = self._tail
        self._tail.prev_free_block = self._head
        for b in blocks:
            self.append(b)

    def popleft(self) -> KVCacheBlock:
        first = self._head.next_free_block
        if first is self._tail:
            raise ValueError("No free blocks available")
        self.remove(first)
        return first

    def remove(self, block: KVCacheBlock) -> None:
        assert block.prev_free_block is not None, f"{block} not in free list"
        prev, nxt = block.prev_free_block, block.next_free_block
        prev.next_free_block = nxt
        nxt.prev_free_block = prev
        block.prev_free_block = None
        block.next_free_block = None
        self.num_free_blocks -= 1

    def append(self, block: KVCacheBlock) -> None:
        last = self._tail.prev_free_block
        last.next_free_block = block
        block.prev_free_block = last
        block.next_free_block = self._tail
        self._tail.prev_free_block = block
        self.num_free_blocks += 1

    def get_all_free_blocks(self) -> list[KVCacheBlock]:
        out = []
        cur = self._head.next_free_block
        while cur is not self._tail:
            out.append(cur)
            cur = cur.next_free_block
        return out
