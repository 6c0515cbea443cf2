"""Unit tests for BlockPool / KVCacheManager / prefix caching.

Mirrors the reference's CPU-only control-plane test pattern
(tests/v1/core/test_prefix_caching.py etc.).
"""

import pytest

from vllm_amd.core.block_pool import BlockPool
from vllm_amd.core.kv_cache_manager import KVCacheManager
from vllm_amd.core.kv_cache_utils import (
    FreeKVCacheBlockQueue,
    KVCacheBlock,
    hash_block_tokens,
    hash_request_tokens,
)
from vllm_amd.request import Request
from vllm_amd.sampling_params import SamplingParams


def make_request(req_id, tokens, max_tokens=16):
    return Request(
        request_id=req_id,
        prompt_token_ids=tokens,
        sampling_params=SamplingParams(max_tokens=max_tokens),
        eos_token_id=None,
    )


class TestHashing:
    def test_chain(self):
        h1 = hash_block_tokens(None, (1, 2, 3))
        h2 = hash_block_tokens(h1.value, (4, 5, 6))
        h1b = hash_block_tokens(None, (1, 2, 3))
        assert h1 == h1b
        assert h1 != h2
        # Different parent -> different hash even for same tokens
        h3 = hash_block_tokens(h2.value, (1, 2, 3))
        assert h3 != h1

    def test_request_tokens_incremental(self):
        toks = list(range(40))
        hashes = hash_request_tokens(16, toks)
        assert len(hashes) == 2  # 40 // 16
        more = hash_request_tokens(16, toks + [40, 41, 42, 43, 44, 45, 46, 47],
                                   prior_hashes=hashes)
        assert len(more) == 3
        assert more[:2] == hashes


class TestFreeQueue:
    def test_lru_order(self):
        blocks = [KVCacheBlock(block_id=i) for i in range(4)]
        q = FreeKVCacheBlockQueue(blocks)
        assert q.num_free_blocks == 4
        assert q.popleft().block_id == 0
        q.append(blocks[0])
        assert [b.block_id for b in q.get_all_free_blocks()] == [1, 2, 3, 0]

    def test_middle_removal(self):
        blocks = [KVCacheBlock(block_id=i) for i in range(4)]
        q = FreeKVCacheBlockQueue(blocks)
        q.remove(blocks[2])
        assert [b.block_id for b in q.get_all_free_blocks()] == [0, 1, 3]
        assert q.num_free_blocks == 3


class TestBlockPool:
    def test_alloc_free(self):
        pool = BlockPool(num_gpu_blocks=8)
        blocks = pool.get_new_blocks(3)
        assert pool.get_num_free_blocks() == 5
        assert all(b.ref_cnt == 1 for b in blocks)
        pool.free_blocks(blocks)
        assert pool.get_num_free_blocks() == 8

    def test_eviction_clears_cache(self):
        pool = BlockPool(num_gpu_blocks=2)
        blocks = pool.get_new_blocks(1)
        h = hash_block_tokens(None, tuple(range(16)))
        pool.cache_full_blocks(blocks, [h], 0, 1)
        assert pool.get_cached_block(h) is blocks[0]
        pool.free_blocks(blocks)
        # Still cached while free.
        assert pool.get_cached_block(h) is blocks[0]
        # Allocate all blocks: the cached free block gets evicted.
        pool.get_new_blocks(2)
        assert pool.get_cached_block(h) is None


class TestKVCacheManager:
    def test_basic_allocate(self):
        mgr = KVCacheManager(num_gpu_blocks=16, block_size=16)
        req = make_request("r1", list(range(40)))
        blocks = mgr.allocate_slots(req, 40)
        assert blocks is not None and len(blocks) == 3  # ceil(40/16)
        mgr.free(req)
        assert mgr.block_pool.get_num_free_blocks() == 16

    def test_allocation_failure(self):
        mgr = KVCacheManager(num_gpu_blocks=2, block_size=16)
        req = make_request("r1", list(range(100)))
        assert mgr.allocate_slots(req, 100) is None

    def test_prefix_cache_hit(self):
        mgr = KVCacheManager(num_gpu_blocks=16, block_size=16)
        prompt = list(range(48))
        req1 = make_request("r1", prompt)
        blocks = mgr.allocate_slots(req1, 48)
        assert len(blocks) == 3
        req1.num_computed_tokens = 48

        # Same prompt: 2 full blocks hit (48 tokens = 3 full blocks, but the
        # last token must be recomputed -> at most 32 reported computed).
        req2 = make_request("r2", list(prompt))
        computed, num = mgr.get_computed_blocks(req2)
        assert num == 32
        assert [b.block_id for b in computed] == [b.block_id for b in blocks[:2]]

        new_blocks = mgr.allocate_slots(req2, 16, computed)
        assert new_blocks is not None
        # Hit blocks now shared.
        assert computed[0].ref_cnt == 2

        mgr.free(req1)
        mgr.free(req2)
        assert mgr.block_pool.get_num_free_blocks() == 16

    def test_prefix_cache_survives_free(self):
        mgr = KVCacheManager(num_gpu_blocks=16, block_size=16)
        prompt = list(range(32))
        req1 = make_request("r1", prompt)
        mgr.allocate_slots(req1, 32)
        mgr.free(req1)
        # Blocks are free but content-cached.
        req2 = make_request("r2", list(prompt) + [99])
        computed, num = mgr.get_computed_blocks(req2)
        assert num == 32

    def test_never_full_prompt_cached(self):
        mgr = KVCacheManager(num_gpu_blocks=16, block_size=16)
        prompt = list(range(32))
        req1 = make_request("r1", prompt)
        mgr.allocate_slots(req1, 32)
        mgr.free(req1)
        req2 = make_request("r2", list(prompt))
        computed, num = mgr.get_computed_blocks(req2)
        # 32-token prompt, 2 blocks cached, but must leave >=1 token to run.
        assert num == 16

    def test_disable_caching(self):
        mgr = KVCacheManager(num_gpu_blocks=16, block_size=16,
                             enable_caching=False)
        prompt = list(range(32))
        req1 = make_request("r1", prompt)
        mgr.allocate_slots(req1, 32)
        mgr.free(req1)
        req2 = make_request("r2", list(prompt))
        computed, num = mgr.get_computed_blocks(req2)
        assert num == 0 and not computed
