"""KV cache CPU-offload tier tests (reference surface: vllm kv_offload /
CPU offloading connector; ours is prefix-cache-integrated — see
core/host_kv_pool.py).

The e2e test forces GPU-pool eviction with a tiny pool, then re-sends the
evicted prompt: with offload enabled the prefix must be restored from the
host tier (cache hit + identical tokens); the copies are real tensor
copies on CPU too, so correctness is end-to-end observable."""

import pytest

from vllm_amd.core.host_kv_pool import HostKVPool
from vllm_amd.core.kv_cache_utils import BlockHash


def _h(v, toks):
    return BlockHash(value=v, token_ids=tuple(toks))


def test_host_pool_lru_and_collision_guard():
    pool = HostKVPool(2)
    s1 = pool.put(_h(1, [1, 2]))
    s2 = pool.put(_h(2, [3, 4]))
    assert {s1, s2} == {0, 1}
    pool.end_round()
    # same int hash, different tokens -> collision guard rejects
    assert pool.lookup(_h(1, [9, 9])) is None
    assert pool.lookup(_h(1, [1, 2])) == s1  # also makes 1 MRU
    # full pool: inserting a third evicts LRU (hash 2)
    s3 = pool.put(_h(3, [5, 6]))
    assert s3 == s2
    pool.end_round()
    assert pool.lookup(_h(2, [3, 4])) is None
    assert pool.lookup(_h(1, [1, 2])) == s1
    assert pool.lookup(_h(3, [5, 6])) == s3


def test_host_pool_in_flight_not_evicted():
    pool = HostKVPool(1)
    s1 = pool.put(_h(1, [1]))
    # slot is in flight this round: a second put must refuse
    assert pool.put(_h(2, [2])) is None
    pool.end_round()
    assert pool.put(_h(2, [2])) == s1


def _gen(llm, prompt, n=6):
    from vllm_amd.sampling_params import SamplingParams

    p = SamplingParams(temperature=0.0, max_tokens=n, ignore_eos=True)
    [out] = llm.generate([prompt], p)
    return out.outputs[0].token_ids, out.outputs[0]


def test_offload_restores_evicted_prefix():
    from vllm_amd.entrypoints.llm import LLM

    kw = dict(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=12, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=2)
    prompt_a = [(7 * j) % 900 + 3 for j in range(47)]    # 3 blocks
    # B spans ~11 of the 12 blocks, forcing eviction of A's cached blocks.
    prompt_b = [(11 * j) % 900 + 3 for j in range(170)]

    # Baseline tokens without offload.
    llm = LLM(**kw)
    base_a, _ = _gen(llm, prompt_a)
    llm.shutdown()

    llm = LLM(cpu_offload_gb=0.001, **kw)  # plenty of host slots
    a1, _ = _gen(llm, prompt_a)
    sched = llm.engine.engine_core.scheduler
    mgr = sched.kv_cache_manager
    assert mgr.host_pool is not None
    # Evict A's cached blocks by filling the 12-block pool with B.
    _gen(llm, prompt_b)
    assert len(mgr.host_pool.entries) > 0  # evictions landed in host tier
    # Re-send A: prefix restored from the host tier.
    q0, h0 = sched.prefix_cache_queries, sched.prefix_cache_hits
    a2, out2 = _gen(llm, prompt_a)
    assert sched.prefix_cache_hits > h0  # counted as a prefix hit
    assert mgr.num_host_hits > 0  # restored from the host tier, not GPU
    assert mgr.num_host_saves > 0
    llm.shutdown()
    assert a1 == base_a
    assert a2 == base_a  # host-restored KV gives identical decode


def test_offload_disabled_by_default():
    from vllm_amd.entrypoints.llm import LLM

    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=32, max_model_len=128,
              max_num_batched_tokens=128, max_num_seqs=2)
    assert llm.engine.engine_core.scheduler.kv_cache_manager.host_pool \
        is None
    llm.shutdown()
