"""Per-request paged KV block accounting (role of vllm/v1/core/kv_cache_manager.py:118).

Single KV-cache group (full attention) in v1; the structure leaves room
for hybrid groups (sliding window / MLA) later.
"""

from __future__ import annotations

from collections import defaultdict
from typing import Optional

from vllm_amd.core.block_pool import BlockPool
from vllm_amd.core.host_kv_pool import HostKVPool
from vllm_amd.core.kv_cache_utils import KVCacheBlock, hash_request_tokens
from vllm_amd.request import Request


class KVCacheManager:

    def __init__(
        self,
        num_gpu_blocks: int,
        block_size: int,
        enable_caching: bool = True,
        num_host_blocks: int = 0,
        sliding_window: int = 0,
        mixed_window: int = 0,
    ) -> None:
        self.block_size = block_size
        # Mixed sliding+global models run two KV groups; prefix caching
        # across groups needs hybrid hash alignment (reference
        # kv_cache_coordinator.py HybridKVCacheCoordinator) — out of
        # scope, so caching is disabled for them.
        self.mixed_window = mixed_window
        if mixed_window > 0 and enable_caching:
            import logging
            logging.getLogger(__name__).info(
                "prefix caching disabled: mixed sliding+global KV groups")
            enable_caching = False
        self.enable_caching = enable_caching
        # Host (CPU RAM) tier for evicted prefix blocks: GPU evictions
        # are saved D2H, host hits are restored H2D (core/host_kv_pool.py).
        self.host_pool = (HostKVPool(num_host_blocks)
                          if num_host_blocks > 0 and enable_caching
                          else None)
        self.pending_swap_ops: list[tuple[str, int, int]] = []
        self.num_host_hits = 0
        self.num_host_saves = 0
        self.block_pool = BlockPool(
            num_gpu_blocks, enable_caching,
            on_evict=self._on_evict if self.host_pool else None)
        self.req_to_blocks: dict[str, list[KVCacheBlock]] = defaultdict(list)
        # How many blocks of each request are already content-cached.
        self.num_cached_blocks: dict[str, int] = defaultdict(int)
        # Uniform sliding-window models (EVERY attention layer windowed,
        # e.g. Mistral): blocks that slid fully out of the window are
        # returned to the pool mid-request. The runner's block table
        # keeps the stale ids (never dereferenced — the kernels compute
        # t_begin from the window), so only the MEMORY is reclaimed.
        # Mixed-pattern models (Gemma3 global layers) need per-group
        # tables — tracked for round 2.
        self.sliding_window = sliding_window
        # req_id -> number of leading blocks already returned to the pool.
        self.num_reclaimed: dict[str, int] = defaultdict(int)
        # Window-group structures (mixed models): a second block list per
        # request, positionally aligned with the full-group list (stale
        # leading ids stay in the table after reclaim; only the memory is
        # returned). Role of the reference's per-group block tables
        # (kv_cache_coordinator.py:60, SlidingWindowManager:878).
        self.req_to_blocks_w: dict[str, list[KVCacheBlock]] = \
            defaultdict(list)
        self.num_reclaimed_w: dict[str, int] = defaultdict(int)

    def _on_evict(self, block_hash, block_id: int) -> None:
        slot = self.host_pool.put(block_hash)
        if slot is not None:
            self.num_host_saves += 1
            self.pending_swap_ops.append(("out", block_id, slot))

    def take_swap_ops(self) -> list[tuple[str, int, int]]:
        ops, self.pending_swap_ops = self.pending_swap_ops, []
        if self.host_pool is not None:
            self.host_pool.end_round()
        return ops

    @property
    def usage(self) -> float:
        return self.block_pool.get_usage()

    def get_computed_blocks(
        self, request: Request
    ) -> tuple[list[KVCacheBlock], int]:
        """Prefix-cache lookup: longest chain of cached full blocks matching
        this request's prompt. Returns (blocks, num_computed_tokens)."""
        if not self.enable_caching:
            return [], 0
        request.block_hashes = hash_request_tokens(
            self.block_size,
            request.all_token_ids,
            prior_hashes=request.block_hashes,
            salt=((getattr(request, "lora_id", 0)
                   ^ getattr(request, "mm_hash", 0)) or None),
        )
        computed: list[KVCacheBlock] = []
        for h in request.block_hashes:
            block = self.block_pool.get_cached_block(h)
            if block is None and self.host_pool is not None:
                block = self._materialize_host_hit(h)
            if block is None:
                break
            # Hold a ref while the chain walk continues: a later
            # _materialize_host_hit allocates a GPU block, and without
            # the ref it could evict — or reuse as its H2D destination —
            # a block already collected here (ref 0 in the free queue),
            # silently corrupting the reported prefix.
            self.block_pool.touch([block])
            computed.append(block)
        num_computed = len(computed) * self.block_size
        # Never report the full prompt as computed: at least the last token
        # must be recomputed to produce logits.
        if num_computed >= request.num_tokens:
            self.block_pool.free_blocks([computed.pop()])
            num_computed -= self.block_size
        # Release the walk refs (blocks re-enter the free queue MRU-end);
        # allocate_slots re-touches the hits it commits immediately after.
        for b in computed:
            self.block_pool.free_blocks([b])
        return computed, num_computed

    def allocate_slots(
        self,
        request: Request,
        num_new_tokens: int,
        new_computed_blocks: Optional[list[KVCacheBlock]] = None,
    ) -> Optional[list[KVCacheBlock]]:
        """Allocate blocks so the request can hold
        num_computed_tokens + len(new_computed_blocks)*bs + num_new_tokens.

        Returns newly allocated blocks (excluding cache hits), or None if
        the pool cannot satisfy the request (caller should preempt).
        """
        assert num_new_tokens > 0
        new_computed_blocks = new_computed_blocks or []
        req_blocks = self.req_to_blocks[request.request_id]

        num_computed_tokens = request.num_computed_tokens + len(
            new_computed_blocks
        ) * self.block_size
        total_tokens = num_computed_tokens + num_new_tokens
        num_required_blocks = (
            total_tokens + self.block_size - 1
        ) // self.block_size
        num_new_blocks = (
            num_required_blocks - len(req_blocks) - len(new_computed_blocks)
        )

        # Sliding window: return leading blocks whose every token lies
        # outside the window of ALL queries from this chunk onward. The
        # bound uses the FIRST new token of this allocation, not the
        # last: earlier query rows of a prefill chunk still read (and
        # write) positions inside their own windows, and a block freed
        # here can be handed to another request scheduled in the SAME
        # round — basing reclaim on total_tokens corrupted KV under pool
        # pressure (caught by test_mixed_model_preemption_resume). Runs
        # BEFORE the capacity check so this request's own stale blocks
        # fund its new allocation; reclaimed blocks are UNCACHED (a
        # prefix hit on window-reclaimed KV cannot be extended). One
        # block of margin absorbs spec-decode rollbacks.
        chunk_start = total_tokens - num_new_tokens
        if self.sliding_window > 0:
            reclaim_below = (
                chunk_start - self.sliding_window - self.block_size
            ) // self.block_size
            done = self.num_reclaimed[request.request_id]
            if reclaim_below > done:
                stale = req_blocks[done:reclaim_below]
                for b in stale:
                    self.block_pool.uncache(b)
                self.block_pool.free_blocks(list(reversed(stale)))
                self.num_reclaimed[request.request_id] = reclaim_below
        if self.mixed_window > 0:
            # Same reclaim rule, applied to the WINDOW group only; global
            # layers keep their full-length blocks.
            reclaim_below = (
                chunk_start - self.mixed_window - self.block_size
            ) // self.block_size
            done = self.num_reclaimed_w[request.request_id]
            if reclaim_below > done:
                blocks_w = self.req_to_blocks_w[request.request_id]
                stale = blocks_w[done:reclaim_below]
                for b in stale:
                    self.block_pool.uncache(b)
                self.block_pool.free_blocks(list(reversed(stale)))
                self.num_reclaimed_w[request.request_id] = reclaim_below

        need = max(num_new_blocks, 0)
        if self.mixed_window > 0:
            need *= 2  # window group allocates in lockstep
        # Committing ref-0 prefix hits (touch) PULLS them off the free
        # queue, so they must be counted against the free pool alongside
        # the new blocks — counting them as "free" let the check pass
        # and get_new_blocks raise under pressure (caught by the soak
        # test).
        need += sum(1 for b in new_computed_blocks if b.ref_cnt == 0)
        if need > self.block_pool.get_num_free_blocks():
            return None

        # Commit cache hits (bump refs) only after we know allocation fits.
        if new_computed_blocks:
            self.block_pool.touch(new_computed_blocks)
            req_blocks.extend(new_computed_blocks)
            request.num_cached_tokens = (
                len(new_computed_blocks) * self.block_size
            )
            # Hit blocks are already content-cached.
            self.num_cached_blocks[request.request_id] = len(req_blocks)

        if num_new_blocks <= 0:
            new_blocks: list[KVCacheBlock] = []
        else:
            new_blocks = self.block_pool.get_new_blocks(num_new_blocks)
            req_blocks.extend(new_blocks)
        if self.mixed_window > 0 and num_new_blocks > 0:
            self.req_to_blocks_w[request.request_id].extend(
                self.block_pool.get_new_blocks(num_new_blocks))

        # Content-cache the blocks that become full after this step.
        if self.enable_caching:
            num_full_after = total_tokens // self.block_size
            request.block_hashes = hash_request_tokens(
                self.block_size,
                request.all_token_ids,
                prior_hashes=request.block_hashes,
                salt=((getattr(request, "lora_id", 0)
                   ^ getattr(request, "mm_hash", 0)) or None),
            )
            # Only blocks whose tokens are all known can be hashed; with
            # chunked prefill total_tokens <= num_tokens so this holds.
            num_hashable = min(num_full_after, len(request.block_hashes))
            cached = self.num_cached_blocks[request.request_id]
            # Never (re)hash a reclaimed block: the pool may have handed
            # it to another owner; tagging it would poison the cache.
            cached = max(cached, self.num_reclaimed[request.request_id])
            if num_hashable > cached:
                self.block_pool.cache_full_blocks(
                    req_blocks,
                    request.block_hashes,
                    cached,
                    num_hashable,
                )
                self.num_cached_blocks[request.request_id] = num_hashable
        return new_blocks

    def _materialize_host_hit(self, h) -> Optional[KVCacheBlock]:
        """Host-tier prefix hit: bring the block back as a CACHED FREE
        GPU block (ref 0, hash registered) plus an H2D swap op — from
        here on it behaves exactly like a normal prefix-cache entry, so
        admission/touch/eviction logic is untouched."""
        slot = self.host_pool.lookup(h)
        if slot is None:
            return None
        if self.block_pool.get_num_free_blocks() == 0:
            return None
        [block] = self.block_pool.get_new_blocks(1)
        block.block_hash = h
        self.block_pool.cached_block_hash_to_block[h.value] = block
        self.block_pool.free_blocks([block])  # ref 0, stays cached
        self.host_pool.in_flight.add(slot)
        self.num_host_hits += 1
        self.pending_swap_ops.append(("in", block.block_id, slot))
        return block

    def free(self, request: Request) -> None:
        blocks = self.req_to_blocks.pop(request.request_id, [])
        self.num_cached_blocks.pop(request.request_id, None)
        reclaimed = self.num_reclaimed.pop(request.request_id, 0)
        # Free in reverse so the tail blocks (least useful as prefix cache)
        # are evicted first (LRU queue order). Skip blocks the sliding
        # window already returned.
        self.block_pool.free_blocks(list(reversed(blocks[reclaimed:])))
        blocks_w = self.req_to_blocks_w.pop(request.request_id, [])
        if blocks_w:
            reclaimed_w = self.num_reclaimed_w.pop(request.request_id, 0)
            self.block_pool.free_blocks(
                list(reversed(blocks_w[reclaimed_w:])))

    def release_trailing(self, request: Request, n: int) -> None:
        """Undo the last `n` blocks appended by allocate_slots (fast-path
        bail rollback): without this the manager keeps blocks the runner
        was never told about, and later boundary checks think the
        request already owns them — the runner then writes through a
        stale zero entry in its block table."""
        if n <= 0:
            return
        blocks = self.req_to_blocks[request.request_id]
        tail = blocks[len(blocks) - n:]
        del blocks[len(blocks) - n:]
        self.block_pool.free_blocks(list(reversed(tail)))

    def get_block_ids(self, request_id: str) -> list[int]:
        return [b.block_id for b in self.req_to_blocks[request_id]]

    def get_block_ids_w(self, request_id: str) -> Optional[list[int]]:
        if self.mixed_window <= 0:
            return None
        return [b.block_id for b in self.req_to_blocks_w[request_id]]

    def last_w_block_ids(self, request_id: str,
                         n: int) -> Optional[list[int]]:
        """Ids of the n window-group blocks the latest allocate_slots
        appended (positionally aligned with its returned full-group
        blocks)."""
        if self.mixed_window <= 0:
            return None
        blocks = self.req_to_blocks_w[request_id]
        return [b.block_id for b in blocks[len(blocks) - n:]] if n else []

    def reset_prefix_cache(self) -> bool:
        if self.host_pool is not None:
            self.host_pool.clear()
            self.pending_swap_ops.clear()
        return self.block_pool.reset_prefix_cache()
