"""Hybrid KV groups (mixed sliding-window + global models, Gemma3
pattern): window layers run on their own block table whose out-of-window
blocks are reclaimed, while global layers keep full-length KV (role of
the reference's per-group block tables / HybridKVCacheCoordinator,
vllm/v1/core/kv_cache_coordinator.py:60,
single_type_kv_cache_manager.py:878 SlidingWindowManager)."""

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


def _llm(**kw):
    return LLM(model="tiny-gemma3", dtype="fp32", device="cpu",
               block_size=16, num_gpu_blocks=kw.pop("blocks", 256),
               max_model_len=512,
               max_num_batched_tokens=kw.pop("mnbt", 512),
               max_num_seqs=4, **kw)


def test_mixed_model_has_two_groups():
    llm = _llm()
    mgr = llm.engine.engine_core.scheduler.kv_cache_manager
    assert mgr.mixed_window == 8
    assert not mgr.enable_caching  # hybrid prefix caching out of scope
    llm.shutdown()


def test_window_group_reclaims_blocks():
    """Steady-state window-group KV is bounded by the window: after a
    long decode the W group holds ~1-2 live blocks while the full group
    holds ceil(len/16)."""
    llm = _llm()
    sched = llm.engine.engine_core.scheduler
    mgr = sched.kv_cache_manager

    reclaimed = {}
    orig = mgr.block_pool.free_blocks

    prompt = list(range(3, 40))  # 37 tokens
    params = SamplingParams(temperature=0.0, max_tokens=120,
                            ignore_eos=True)
    [out] = llm.generate([prompt], params)
    assert len(out.outputs[0].token_ids) == 120
    # Window reclaim counters advanced (the request is freed by now, so
    # inspect the manager's bookkeeping via a fresh request mid-flight).
    total_len = 37 + 120  # 157 tokens -> 10 full-group blocks
    # Run again and snapshot live state right before the request ends.
    req_blocks = {}

    class Probe:
        def __call__(self):
            for rid, blocks in mgr.req_to_blocks.items():
                w = mgr.req_to_blocks_w.get(rid, [])
                req_blocks[rid] = (len(blocks),
                                   mgr.num_reclaimed_w.get(rid, 0), len(w))

    probe = Probe()
    orig_free = mgr.free

    def free_probe(request):
        probe()
        return orig_free(request)

    mgr.free = free_probe
    llm.generate([prompt], params)
    llm.shutdown()
    assert req_blocks, "probe never fired"
    (nf, nrw, nw), = req_blocks.values()
    assert nf == -(-total_len // 16)      # full group: full-length KV
    assert nw == nf                        # aligned table lengths
    # window 8 + one block margin: all but the last ~2 blocks reclaimed.
    assert nrw >= nf - 2, f"window group only reclaimed {nrw} of {nf}"


def test_mixed_outputs_batched_vs_single():
    """Block reuse across the two groups must not corrupt outputs:
    batched generation matches one-at-a-time generation exactly."""
    prompts = [list(range(3, 30)), list(range(50, 95)),
               [7, 8, 9] * 11]
    params = SamplingParams(temperature=0.0, max_tokens=16,
                            ignore_eos=True)
    llm = _llm()
    batched = [o.outputs[0].token_ids for o in llm.generate(prompts, params)]
    llm.shutdown()
    singles = []
    for p in prompts:
        llm = _llm()
        singles.append(
            llm.generate([p], params)[0].outputs[0].token_ids)
        llm.shutdown()
    assert batched == singles


def test_mixed_chunked_prefill_invariance():
    """Chunked prefill writes window-layer KV through the W table in
    pieces; output must match whole-prompt prefill."""
    prompt = list(range(3, 120))
    params = SamplingParams(temperature=0.0, max_tokens=12,
                            ignore_eos=True)
    big = _llm()
    whole = big.generate([prompt], params)[0].outputs[0].token_ids
    big.shutdown()
    small = _llm(mnbt=32)
    chunked = small.generate([prompt], params)[0].outputs[0].token_ids
    small.shutdown()
    assert whole == chunked


def test_mixed_llama_arch_uses_two_groups():
    """tiny-mistral (llama arch with the Gemma3-style mixed pattern)
    also routes its window layers through the W group; outputs stay
    consistent between batched and single runs."""
    kw = dict(model="tiny-mistral", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=2)
    llm = LLM(**kw)
    mgr = llm.engine.engine_core.scheduler.kv_cache_manager
    assert mgr.mixed_window == 8 and mgr.sliding_window == 0
    prompts = [list(range(3, 40)), [5, 6, 7] * 9]
    params = SamplingParams(temperature=0.0, max_tokens=20,
                            ignore_eos=True)
    batched = [o.outputs[0].token_ids for o in llm.generate(prompts, params)]
    llm.shutdown()
    for p, expect in zip(prompts, batched):
        llm = LLM(**kw)
        got = llm.generate([p], params)[0].outputs[0].token_ids
        llm.shutdown()
        assert got == expect


def test_mixed_model_preemption_resume():
    """Preemption under memory pressure with two KV groups: the victim's
    blocks (both groups) are freed and the resumed request re-prefills
    through fresh W-group tables — outputs must match an uncontended
    run."""
    params = SamplingParams(temperature=0.0, max_tokens=24,
                            ignore_eos=True)
    prompts = [list(range(3, 60)), list(range(100, 170))]

    # Uncontended reference outputs.
    llm = _llm(blocks=256)
    ref = [o.outputs[0].token_ids for o in llm.generate(prompts, params)]
    llm.shutdown()

    # Tiny pool: two groups * two requests force preemption mid-decode.
    llm = _llm(blocks=16, mnbt=128)
    sched = llm.engine.engine_core.scheduler
    got = [o.outputs[0].token_ids for o in llm.generate(prompts, params)]
    preempted = sched.num_preemptions_total
    llm.shutdown()
    assert got == ref
    assert preempted > 0, "pool was large enough that nothing preempted"
