"""Jamba-style hybrid (attention + mamba) on CPU: paged KV and SSM
state rows coexist in one model (reference jamba.py +
HybridKVCacheCoordinator with FullAttentionManager + MambaManager)."""

import numpy as np
import pytest

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


def _llm(**kw):
    return LLM(model="tiny-jamba", dtype="fp32", device="cpu",
               block_size=16, num_gpu_blocks=kw.pop("blocks", 64),
               max_model_len=256,
               max_num_batched_tokens=kw.pop("mnbt", 256),
               max_num_seqs=4, **kw)


def _prompt(seed, n=24):
    rng = np.random.default_rng(seed)
    return rng.integers(10, 900, size=n).tolist()


GREEDY = SamplingParams(max_tokens=8, temperature=0.0, ignore_eos=True,
                        logprobs=1)


def _gen(llm, prompt, params=GREEDY):
    o = llm.generate([{"prompt_token_ids": list(prompt)}], params)[0]
    return o.outputs[0].token_ids


def test_layer_pattern():
    from vllm_amd.config import get_model_spec

    spec = get_model_spec("tiny-jamba")
    assert [spec.is_attn_layer(i) for i in range(4)] == [
        False, True, False, True]


def test_long_range_history_via_attention():
    """Unlike pure mamba (where early-token influence decays through the
    recurrence), the attention layers read the FULL paged KV: perturbing
    the first prompt token must move the final logits."""
    llm = _llm()

    def lp_of(prompt):
        o = llm.generate([{"prompt_token_ids": prompt}], GREEDY)[0]
        out = o.outputs[0]
        v = out.logprobs[0][out.token_ids[0]]
        return float(getattr(v, "logprob", v))

    p = _prompt(0)
    lp1 = lp_of(p)
    p2 = list(p)
    p2[0] = (p2[0] + 13) % 900 + 10
    lp2 = lp_of(p2)
    llm.shutdown()
    assert lp1 != lp2


def test_chunked_prefill_invariance():
    """Chunking must carry BOTH cache kinds: paged KV for the attention
    layers and conv/SSM state for the mamba layers."""
    p = _prompt(1)
    big = _llm()
    whole = _gen(big, p)
    big.shutdown()
    small = _llm(mnbt=8)
    chunked = _gen(small, p)
    small.shutdown()
    assert whole == chunked


def test_decode_matches_prefill():
    p = _prompt(2)
    llm = _llm()
    toks = _gen(llm, p)
    follow = _gen(llm, p + toks[:5],
                  SamplingParams(max_tokens=1, temperature=0.0,
                                 ignore_eos=True))
    llm.shutdown()
    assert follow[0] == toks[5]


def test_batched_isolation_and_preemption():
    prompts = [_prompt(10 + i, n=40) for i in range(4)]
    params = SamplingParams(max_tokens=16, temperature=0.0,
                            ignore_eos=True)
    calm = _llm()
    want = [o.outputs[0].token_ids for o in calm.generate(
        [{"prompt_token_ids": p} for p in prompts], params)]
    solo = _gen(calm, prompts[0], params)
    calm.shutdown()
    assert want[0] == solo  # batched == solo (state + KV isolation)
    tight = _llm(blocks=16)
    got = [o.outputs[0].token_ids for o in tight.generate(
        [{"prompt_token_ids": p} for p in prompts], params)]
    tight.shutdown()
    assert got == want  # preemption recomputes both cache kinds


def test_gates():
    llm = _llm()
    mgr = llm.engine.engine_core.scheduler.kv_cache_manager
    assert not mgr.enable_caching
    llm.shutdown()
    with pytest.raises(ValueError, match="speculative"):
        _llm(num_speculative_tokens=3)


@pytest.mark.gpu
def test_jamba_gpu_smoke():
    # head_dim-128 / block-64 geometry: the attention layers run the
    # HIP paged kernels, the mamba layers the torch SSM path.
    llm = LLM(model="tiny-jamba-128", dtype="bf16", device="cuda",
              block_size=64, num_gpu_blocks=64, max_model_len=2048,
              max_num_batched_tokens=2048, max_num_seqs=4)
    p = _prompt(42)
    a = _gen(llm, p)
    b = _gen(llm, p)
    llm.shutdown()
    assert len(a) == 8
    assert a == b
