"""Sliding-window KV block reclaim (uniform-window models).

Every-layer-windowed models (Mistral) return blocks that slid fully out
of the window to the pool mid-request: steady-state KV memory is
window-bound, not context-bound. Mixed-pattern models (Gemma3) are
excluded (they need per-group tables — tracked for round 2)."""

from vllm_amd.core.kv_cache_manager import KVCacheManager
from vllm_amd.request import Request
from vllm_amd.sampling_params import SamplingParams


def _req(rid, n_tokens):
    return Request(request_id=rid,
                   prompt_token_ids=list(range(3, 3 + n_tokens)),
                   sampling_params=SamplingParams(max_tokens=4))


def test_manager_reclaims_out_of_window_blocks():
    m = KVCacheManager(num_gpu_blocks=32, block_size=16,
                       enable_caching=True, sliding_window=16)
    r = _req("a", 200)
    free0 = m.block_pool.get_num_free_blocks()
    blocks = m.allocate_slots(r, 200)
    assert blocks is not None
    r.num_computed_tokens = 200
    # Reclaim is bounded by the CHUNK START (the first new token's
    # window), not the chunk end: the 200-token prefill chunk itself
    # still reads its early blocks, so nothing is reclaimed during it.
    assert m.num_reclaimed["a"] == 0
    # The next allocation (chunk start 200) reclaims blocks below
    # (200 - window 16 - margin 16) // 16 = 10.
    m.allocate_slots(r, 8)
    r.num_computed_tokens = 208
    assert m.num_reclaimed["a"] == 10
    used = free0 - m.block_pool.get_num_free_blocks()
    assert used == 13 - 10  # 13 allocated, 10 reclaimed
    # decode steps keep the window moving
    m.allocate_slots(r, 8)
    r.num_computed_tokens = 216
    assert m.num_reclaimed["a"] == 11
    # request end: remaining blocks freed exactly once (no double free)
    m.free(r)
    assert m.block_pool.get_num_free_blocks() == free0


def test_manager_no_reclaim_without_window():
    m = KVCacheManager(num_gpu_blocks=32, block_size=16,
                       enable_caching=True)
    r = _req("a", 200)
    m.allocate_slots(r, 200)
    assert m.num_reclaimed["a"] == 0


def test_e2e_window_bound_memory():
    """A pool far smaller than the total sequence serves a long windowed
    generation (impossible without reclaim), and reclaim does not change
    the sampled tokens vs a big-pool run."""
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    prompt = [(5 * j) % 900 + 3 for j in range(32)]
    p = SamplingParams(temperature=0.0, max_tokens=120, ignore_eos=True)

    big = LLM(model="tiny-swa", dtype="fp32", device="cpu", block_size=16,
              num_gpu_blocks=64, max_model_len=512,
              max_num_batched_tokens=128, max_num_seqs=2)
    [ref] = big.generate([prompt], p)
    big.shutdown()

    # 8 blocks = 128 token-slots < 152 total tokens: only works because
    # out-of-window blocks are returned mid-request.
    small = LLM(model="tiny-swa", dtype="fp32", device="cpu",
                block_size=16, num_gpu_blocks=8, max_model_len=512,
                max_num_batched_tokens=128, max_num_seqs=2)
    [out] = small.generate([prompt], p)
    sched = small.engine.engine_core.scheduler
    assert sched.kv_cache_manager.sliding_window == 16
    small.shutdown()
    assert len(out.outputs[0].token_ids) == 120
    assert out.outputs[0].token_ids == ref.outputs[0].token_ids


def test_e2e_gpu_geometry_on_cpu():
    """Same shapes as the GPU reclaim test (block 64, window 64, 3-block
    pool) on the CPU reference path — de-risks the @gpu variant."""
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    prompt = [(5 * j) % 900 + 3 for j in range(80)]
    p = SamplingParams(temperature=0.0, max_tokens=200, ignore_eos=True)
    kw = dict(model="tiny-swa-128", dtype="fp32", device="cpu",
              max_model_len=512, max_num_batched_tokens=512,
              max_num_seqs=2, block_size=64)
    big = LLM(num_gpu_blocks=64, **kw)
    [ref] = big.generate([prompt], p)
    big.shutdown()
    small = LLM(num_gpu_blocks=3, **kw)
    [out] = small.generate([prompt], p)
    small.shutdown()
    assert out.outputs[0].token_ids == ref.outputs[0].token_ids
