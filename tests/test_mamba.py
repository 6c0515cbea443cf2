"""Mamba (SSM) family on CPU: recurrent state carried across chunked
prefill, decode recurrence consistent with the prefill scan, per-request
state isolation, preemption recompute, and the engine gates (prefix
caching off, no spec decode). Reference behavior:
vllm/model_executor/models/mamba.py + MambaManager state handling."""

import numpy as np
import pytest

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


def _llm(**kw):
    return LLM(model="tiny-mamba", dtype="fp32", device="cpu",
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


def test_generates_and_depends_on_history():
    llm = _llm()
    a = _gen(llm, _prompt(0))
    assert len(a) == 8
    # Perturb ONE early prompt token: the recurrent state must carry the
    # change to the end of the sequence (logprob of the first sampled
    # token moves).
    p = _prompt(0)

    def lp_of(prompt):
        o = llm.generate([{"prompt_token_ids": prompt}], GREEDY)[0]
        out = o.outputs[0]
        v = out.logprobs[0][out.token_ids[0]]
        return float(getattr(v, "logprob", v))

    lp1 = lp_of(p)
    # Perturb a RECENT prompt token (inside the conv lookback and with
    # little state decay): the logits must move. (A very early token's
    # influence decays as prod(dA) — ~1e-7 after 20 steps with this
    # random init — so early-token sensitivity is not assertable.)
    p2 = list(p)
    p2[-3] = (p2[-3] + 13) % 900 + 10
    lp2 = lp_of(p2)
    llm.shutdown()
    assert lp1 != lp2


def test_chunked_prefill_state_carry():
    """mnbt=8 forces the 24-token prompt through 3 chunks: conv and SSM
    state must flow across chunk boundaries to match the whole-prompt
    run (same fp op order per token -> exact equality in fp32)."""
    p = _prompt(1)
    big = _llm()
    whole = _gen(big, p)
    big.shutdown()
    small = _llm(mnbt=8)
    chunked = _gen(small, p)
    small.shutdown()
    assert whole == chunked


def test_decode_matches_prefill_scan():
    """Cross-validate the two state paths: tokens produced by the O(1)
    decode recurrence must match what the prefill scan predicts when the
    same context arrives as a prompt."""
    p = _prompt(2)
    llm = _llm()
    toks = _gen(llm, p)
    # Ask for 1 token from prompt+first 5 generated: its argmax comes
    # from the scan path and must agree with generated token 6.
    follow = _gen(llm, p + toks[:5],
                  SamplingParams(max_tokens=1, temperature=0.0,
                                 ignore_eos=True))
    llm.shutdown()
    assert follow[0] == toks[5]


def test_request_state_isolation():
    """Two concurrent requests must keep separate state rows: batched
    results equal solo results."""
    pa, pb = _prompt(3), _prompt(4, n=17)
    llm = _llm()
    solo_a = _gen(llm, pa)
    solo_b = _gen(llm, pb)
    outs = llm.generate([{"prompt_token_ids": pa},
                         {"prompt_token_ids": pb}], GREEDY)
    llm.shutdown()
    assert outs[0].outputs[0].token_ids == solo_a
    assert outs[1].outputs[0].token_ids == solo_b


def test_preemption_recomputes_state():
    """Pool pressure forces preemption; the victim resumes by zeroing
    its state row and re-scanning from position 0, so outputs match an
    unpressured run."""
    prompts = [_prompt(10 + i, n=40) for i in range(4)]
    params = SamplingParams(max_tokens=16, temperature=0.0,
                            ignore_eos=True)
    calm = _llm()
    want = [llm_out.outputs[0].token_ids for llm_out in calm.generate(
        [{"prompt_token_ids": p} for p in prompts], params)]
    calm.shutdown()
    tight = _llm(blocks=16)  # 256 tokens of KV accounting for 4*56 demand
    got = [o.outputs[0].token_ids for o in tight.generate(
        [{"prompt_token_ids": p} for p in prompts], params)]
    tight.shutdown()
    assert got == want


def test_prefix_caching_forced_off_and_spec_gated():
    llm = _llm()
    assert not llm.engine.engine_core.scheduler.kv_cache_manager.\
        enable_caching
    llm.shutdown()
    with pytest.raises(ValueError, match="speculative"):
        _llm(num_speculative_tokens=3)


def _mixer():
    import torch

    from vllm_amd.config import get_model_spec
    from vllm_amd.models.mamba import MambaMixer

    spec = get_model_spec("tiny-mamba")
    torch.manual_seed(7)
    m = MambaMixer(spec, cache_idx=0, dtype=torch.float32)
    with torch.no_grad():
        for p in m.parameters():
            p.uniform_(-0.5, 0.5)
        m.A_log.uniform_(-1.0, 0.5)  # keep the recurrence contractive
        m.D.uniform_(0.5, 1.5)
    return m


def test_mixer_ssm_step_equals_scan():
    """The O(1) decode recurrence composed T times must equal the
    prefill scan from the same initial state (same op per token)."""
    import torch

    m = _mixer()
    T = 12
    xc = torch.randn(T, m.d_inner)
    h0 = torch.randn(m.d_inner, m.d_state)
    y_scan, h_scan = m._ssm(xc, h0.clone(), step=False)
    h = h0.clone().unsqueeze(0)
    ys = []
    for t in range(T):
        y_t, h = m._ssm(xc[t:t + 1], h, step=True)
        ys.append(y_t)
    assert torch.allclose(torch.cat(ys), y_scan, rtol=1e-4,
                          atol=1e-3)
    assert torch.allclose(h[0], h_scan, rtol=1e-4, atol=1e-4)


def test_mixer_conv_decode_matches_conv1d():
    """Step-wise causal conv through the rolling cache must match plain
    torch F.conv1d (independent fp32 reference) over the sequence."""
    import torch
    import torch.nn.functional as F

    m = _mixer()
    T = 9
    seq = torch.randn(T, m.d_inner)
    ref = F.conv1d(seq.t().unsqueeze(0),
                   m.conv_weight.detach().unsqueeze(1),
                   m.conv_bias.detach(),
                   padding=m.d_conv - 1,
                   groups=m.d_inner)[0][:, :T].t()
    ref = F.silu(ref)
    cache = torch.zeros(1, m.d_inner, m.d_conv - 1)
    rows = torch.tensor([0])
    got = torch.cat([m._conv_decode(seq[t:t + 1], rows, cache)
                     for t in range(T)])
    assert torch.allclose(got, ref, atol=1e-5)


@pytest.mark.gpu
def test_mamba_gpu_smoke():
    """Eager SSM path on MI355X: generates, and is deterministic across
    identical runs (prefix caching is off, so both runs chunk alike)."""
    llm = LLM(model="tiny-mamba", dtype="bf16", device="cuda",
              block_size=16, num_gpu_blocks=64, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=4)
    p = _prompt(42)
    a = _gen(llm, p)
    b = _gen(llm, p)
    llm.shutdown()
    assert len(a) == 8
    assert a == b
