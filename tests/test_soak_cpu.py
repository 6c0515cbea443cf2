"""Randomized engine soak (CPU): many requests with mixed sampling
params, chunked prefill, prefix caching + host offload, priority
scheduling, spec decode and a pool small enough to force preemption —
all at once. Asserts (a) every request completes with the requested
token count, (b) the whole run is bit-deterministic across two
executions (catches cross-feature state leaks the targeted tests
miss — the round-2 window-reclaim corruption was exactly this class)."""

import numpy as np

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


def _run(seed: int):
    rng = np.random.default_rng(seed)
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=30, max_model_len=256,
              max_num_batched_tokens=96, max_num_seqs=6,
              enable_prefix_caching=True, cpu_offload_gb=0.001,
              scheduling_policy="priority", num_speculative_tokens=2)
    prompts = []
    params = []
    shared = [int(x) for x in rng.integers(3, 900, size=24)]
    for i in range(28):
        plen = int(rng.integers(4, 120))
        if i % 3 == 0:
            p = shared + [int(x) for x in
                          rng.integers(3, 900, size=max(1, plen - 24))]
        else:
            p = [int(x) for x in rng.integers(3, 900, size=plen)]
        prompts.append(p[:200])
        temp = float(rng.choice([0.0, 0.0, 1.0]))
        params.append(SamplingParams(
            temperature=temp,
            seed=int(rng.integers(0, 2**31)) if temp > 0 else None,
            top_p=float(rng.choice([1.0, 0.9])),
            top_k=int(rng.choice([0, 20])),
            repetition_penalty=float(rng.choice([1.0, 1.1])),
            max_tokens=int(rng.integers(1, 60)),
            ignore_eos=True,
            priority=int(rng.integers(0, 3)),
        ))
    outs = llm.generate(prompts, params)
    sched = llm.engine.engine_core.scheduler
    stats = dict(preempts=sched.num_preemptions_total,
                 host_hits=sched.kv_cache_manager.num_host_hits,
                 cache_hits=sched.prefix_cache_hits,
                 drafted=sched.spec_stats_drafted)
    llm.shutdown()
    toks = [o.outputs[0].token_ids for o in outs]
    return toks, stats, params


def test_soak_deterministic_under_pressure():
    toks1, stats1, params = _run(1234)
    toks2, stats2, _ = _run(1234)
    assert toks1 == toks2, "run-to-run divergence under pool pressure"
    for t, p in zip(toks1, params):
        assert len(t) == p.max_tokens
    # The soak must actually exercise the pressure paths.
    assert stats1["preempts"] > 0 or stats1["host_hits"] > 0, stats1
    assert stats1["cache_hits"] > 0, stats1
    assert stats1["drafted"] > 0, stats1


def test_soak_mixed_window_model():
    """Same soak shape on the hybrid-KV model (two block-table groups,
    window reclaim, preemption) — prefix caching auto-disabled there."""
    def run(seed):
        rng = np.random.default_rng(seed)
        llm = LLM(model="tiny-gemma3", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=36, max_model_len=256,
                  max_num_batched_tokens=96, max_num_seqs=4,
                  scheduling_policy="priority")
        prompts = [[int(x) for x in rng.integers(3, 900,
                                                 size=int(rng.integers(4, 100)))]
                   for _ in range(16)]
        params = [SamplingParams(
            temperature=float(rng.choice([0.0, 1.0])),
            seed=int(rng.integers(0, 2**31)),
            max_tokens=int(rng.integers(1, 50)), ignore_eos=True,
            priority=int(rng.integers(0, 3))) for _ in range(16)]
        outs = llm.generate(prompts, params)
        sched = llm.engine.engine_core.scheduler
        pre = sched.num_preemptions_total
        llm.shutdown()
        return [o.outputs[0].token_ids for o in outs], pre, params

    t1, pre1, params = run(77)
    t2, _, _ = run(77)
    assert t1 == t2
    for t, p in zip(t1, params):
        assert len(t) == p.max_tokens
    assert pre1 > 0, "no pool pressure exercised"


def test_soak_jamba_hybrid_state():
    """Soak shape on the attention+SSM hybrid: paged KV and recurrent
    state rows under preemption and chunked prefill. Bit-determinism
    across runs catches state-row leaks (a finished request's recycled
    row bleeding into a new request) that targeted tests can miss."""
    def run(seed):
        rng = np.random.default_rng(seed)
        llm = LLM(model="tiny-jamba", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=24, max_model_len=256,
                  max_num_batched_tokens=96, max_num_seqs=4,
                  scheduling_policy="priority")
        prompts = [[int(x) for x in
                    rng.integers(3, 900, size=int(rng.integers(4, 100)))]
                   for _ in range(16)]
        params = [SamplingParams(
            temperature=float(rng.choice([0.0, 1.0])),
            seed=int(rng.integers(0, 2**31)),
            max_tokens=int(rng.integers(1, 50)), ignore_eos=True,
            priority=int(rng.integers(0, 3))) for _ in range(16)]
        outs = llm.generate(prompts, params)
        sched = llm.engine.engine_core.scheduler
        pre = sched.num_preemptions_total
        llm.shutdown()
        return [o.outputs[0].token_ids for o in outs], pre, params

    t1, pre1, params = run(99)
    t2, _, _ = run(99)
    assert t1 == t2
    for t, p in zip(t1, params):
        assert len(t) == p.max_tokens
    assert pre1 > 0, "no pool pressure exercised"


def test_soak_moe_model():
    """Soak shape on the MoE model (expert routing + spec decode +
    preemption + prefix caching together)."""
    def run(seed):
        rng = np.random.default_rng(seed)
        llm = LLM(model="tiny-mixtral", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=16, max_model_len=256,
                  max_num_batched_tokens=96, max_num_seqs=4,
                  scheduling_policy="priority",
                  num_speculative_tokens=2)
        shared = [int(x) for x in rng.integers(3, 900, size=20)]
        prompts = []
        for i in range(14):
            plen = int(rng.integers(4, 90))
            p = (shared + [int(x) for x in
                           rng.integers(3, 900, size=max(1, plen - 20))]
                 if i % 3 == 0 else
                 [int(x) for x in rng.integers(3, 900, size=plen)])
            prompts.append(p[:180])
        params = [SamplingParams(
            temperature=float(rng.choice([0.0, 1.0])),
            seed=int(rng.integers(0, 2**31)),
            max_tokens=int(rng.integers(20, 60)), ignore_eos=True,
            priority=int(rng.integers(0, 3))) for _ in range(14)]
        outs = llm.generate(prompts, params)
        pre = llm.engine.engine_core.scheduler.num_preemptions_total
        llm.shutdown()
        return [o.outputs[0].token_ids for o in outs], pre, params

    t1, pre1, params = run(55)
    t2, _, _ = run(55)
    assert t1 == t2
    for t, p in zip(t1, params):
        assert len(t) == p.max_tokens
    assert pre1 > 0, "no pool pressure exercised"
