"""Speculative decoding (ngram prompt-lookup) tests: spec-on greedy
output must equal spec-off output token-for-token, and drafts must be
accepted on repetitive text."""

import pytest

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams
from vllm_amd.spec_decode.ngram import NgramProposer


def test_ngram_proposer_basic():
    p = NgramProposer(min_n=2, max_n=3, k=3)
    # "5 6 7 8" repeats: suffix [7, 8] seen before, followed by 9, 5, 6.
    toks = [5, 6, 7, 8, 9, 5, 6, 7, 8]
    assert p.propose(toks) == [9, 5, 6]
    assert p.propose([1, 2, 3]) is None


def _generate(spec_tokens: int):
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=spec_tokens)
    # Strongly repetitive prompt so n-gram lookup fires.
    prompt = [7, 8, 9, 10] * 12
    outs = llm.generate(
        [prompt, list(range(30, 60))],
        SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True),
    )
    stats = None
    sched = llm.engine.engine_core.scheduler
    stats = (sched.spec_stats_drafted, sched.spec_stats_accepted)
    llm.shutdown()
    return [o.outputs[0].token_ids for o in outs], stats


def test_spec_decode_matches_baseline():
    base, _ = _generate(0)
    spec, _ = _generate(4)
    assert base == spec
    assert all(len(t) == 24 for t in spec)


def test_scheduler_spec_rollback():
    """Scheduler-level: draft scheduling, acceptance accounting and KV
    rollback of rejected positions (reference update_from_output:1679)."""
    from tests.test_scheduler import create_scheduler, make_request
    from vllm_amd.core.sched_output import ModelRunnerOutput

    sched = create_scheduler()
    class _NoneProposer:
        model_based = False

        def propose(self, toks):
            return None

    sched.spec_proposer = _NoneProposer()  # enable the propose path
    req = make_request("r1", num_tokens=16, max_tokens=32)
    req.sampling_params.ignore_eos = True
    sched.add_request(req)

    # Prefill + first token.
    out = sched.schedule()
    sched.update_from_output(out, ModelRunnerOutput(
        req_ids=["r1"], sampled_token_ids=[[7]]))
    # Manually attach drafts (as the ngram proposer would).
    req.spec_token_ids = [100, 101, 102]
    out = sched.schedule()
    assert out.num_scheduled_tokens["r1"] == 4  # 1 real + 3 drafts
    assert out.scheduled_spec_decode_tokens["r1"] == [100, 101, 102]
    computed_after_sched = req.num_computed_tokens
    # Runner accepts draft 100 but rejects 101: returns 3 tokens
    # (s0=100 accepted? semantics: runner returns [s0, s1] when first
    # draft matched, second did not).
    sched.update_from_output(out, ModelRunnerOutput(
        req_ids=["r1"], sampled_token_ids=[[100, 55]]))
    # 4 scheduled, 2 kept -> 2 rejected positions rolled back.
    assert req.num_computed_tokens == computed_after_sched - 2
    assert req.all_token_ids[-2:] == [100, 55]
    assert sched.spec_stats_drafted == 3
    assert sched.spec_stats_accepted == 1


def test_spec_decode_accepts_on_repetition():
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=4)
    # Single-token loop prompt: the model likely repeats; even if not,
    # the engine must stay correct (count check only).
    outs = llm.generate(
        [[3, 4] * 20],
        SamplingParams(temperature=0.0, max_tokens=16, ignore_eos=True),
    )
    llm.shutdown()
    assert len(outs[0].outputs[0].token_ids) == 16


def test_medusa_heads_propose_shapes():
    import torch

    from vllm_amd.spec_decode.medusa import MedusaHeads

    heads = MedusaHeads(hidden_size=64, vocab_size=97, k=3,
                        dtype=torch.float32)
    heads.init_dummy(seed=0)
    h = torch.randn(5, 64)
    d = heads.propose(h)
    assert d.shape == (5, 3)
    assert (d >= 0).all() and (d < 97).all()
    # deterministic
    assert torch.equal(d, heads.propose(h))
    heads2 = MedusaHeads(hidden_size=64, vocab_size=97, k=3,
                         dtype=torch.float32)
    heads2.init_dummy(seed=0)
    assert torch.equal(d, heads2.propose(h))


def _generate_medusa(spec_tokens: int):
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=spec_tokens,
              spec_decode_method="medusa")
    prompt = [7, 8, 9, 10] * 12
    outs = llm.generate(
        [prompt, list(range(30, 60))],
        SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True),
    )
    sched = llm.engine.engine_core.scheduler
    stats = (sched.spec_stats_drafted, sched.spec_stats_accepted)
    llm.shutdown()
    return [o.outputs[0].token_ids for o in outs], stats


def test_medusa_matches_baseline():
    """Random (untrained) medusa heads: almost every draft is rejected,
    but greedy in-place verification must keep the output EXACTLY equal
    to the non-speculative run — correctness never depends on head
    quality."""
    base, _ = _generate(0)
    med, stats = _generate_medusa(3)
    assert base == med
    assert all(len(t) == 24 for t in med)
    assert stats[0] > 0  # drafts were actually proposed and verified


def test_dynamic_speculation_length():
    """Poor draft acceptance shrinks the speculation length; outputs stay
    exact (drafts are only ever rejected work, never wrong tokens)."""
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=4, spec_decode_method="medusa")
    # Random-head medusa drafts are ~never accepted -> k must shrink.
    prompts = [[(i * 17 + j) % 900 + 3 for j in range(24)]
               for i in range(3)]
    p = SamplingParams(temperature=0.0, max_tokens=40, ignore_eos=True)
    outs = llm.generate(prompts, p)
    sched = llm.engine.engine_core.scheduler
    assert sched.spec_k < 4
    assert all(len(o.outputs[0].token_ids) == 40 for o in outs)
    # equality with non-speculative run
    llm2 = LLM(model="tiny-llama", dtype="fp32", device="cpu",
               block_size=16, num_gpu_blocks=256, max_model_len=512,
               max_num_batched_tokens=512, max_num_seqs=4)
    base = llm2.generate(prompts, p)
    llm2.shutdown()
    llm.shutdown()
    for a, b in zip(outs, base):
        assert a.outputs[0].token_ids == b.outputs[0].token_ids


# ---------------------------------------------------------------- EAGLE

def test_eagle_draft_chunk_shapes_and_determinism():
    import torch

    from vllm_amd.spec_decode.eagle import EagleDraft, _ReqKV

    d = EagleDraft(hidden_size=64, num_heads=4, num_kv_heads=2,
                   intermediate_size=128, dtype=torch.float32)
    d.init_dummy(seed=0)
    kv = _ReqKV(64, 2, 16, torch.float32, torch.device("cpu"))
    inp = torch.randn(5, 128)
    g = d.forward_chunk(inp, 0, kv)
    assert g.shape == (5, 64)
    assert kv.len == 5
    # Incremental single-slot continuation matches a fresh full pass.
    inp2 = torch.randn(1, 128)
    g2 = d.forward_chunk(inp2, 5, kv)
    kv_b = _ReqKV(64, 2, 16, torch.float32, torch.device("cpu"))
    g_full = d.forward_chunk(torch.cat([inp, inp2]), 0, kv_b)
    assert torch.allclose(g2[-1], g_full[-1], atol=1e-5)
    # Rollback: rewriting slot 5 after truncation reuses the slot.
    kv.len = 5
    g3 = d.forward_chunk(inp2, 5, kv)
    assert torch.allclose(g2, g3)


def _generate_eagle(spec_tokens: int):
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=spec_tokens,
              spec_decode_method="eagle")
    prompt = [7, 8, 9, 10] * 12
    outs = llm.generate(
        [prompt, list(range(30, 60))],
        SamplingParams(temperature=0.0, max_tokens=24, ignore_eos=True),
    )
    sched = llm.engine.engine_core.scheduler
    stats = (sched.spec_stats_drafted, sched.spec_stats_accepted)
    llm.shutdown()
    return [o.outputs[0].token_ids for o in outs], stats


def test_eagle_matches_baseline():
    """Random (untrained) EAGLE draft: drafts are ~all rejected, but
    greedy in-place verification keeps outputs EXACTLY equal to the
    non-speculative run — correctness never depends on draft quality."""
    base, _ = _generate(0)
    egl, stats = _generate_eagle(3)
    assert base == egl
    assert all(len(t) == 24 for t in egl)
    assert stats[0] > 0  # drafts were actually produced and verified


class _JunkProposer:
    """Deterministic arbitrary proposer: rejection sampling must keep
    the output distribution EXACTLY regardless of draft quality — bad
    drafts only cost acceptance rate."""

    model_based = False

    def propose(self, toks):
        t = toks[-1]
        return [t, (t * 7 + 13) % 900 + 3, (t * 3 + 5) % 900 + 3]


def _generate_sampled(spec_tokens: int, seed: int = 1234):
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=spec_tokens)
    if spec_tokens > 0:
        llm.engine.engine_core.scheduler.spec_proposer = _JunkProposer()
    prompt = [7, 8, 9, 10] * 12
    outs = llm.generate(
        [prompt, list(range(30, 60))],
        SamplingParams(temperature=1.0, seed=seed, max_tokens=24,
                       ignore_eos=True),
    )
    sched = llm.engine.engine_core.scheduler
    stats = (sched.spec_stats_drafted, sched.spec_stats_accepted)
    llm.shutdown()
    return [o.outputs[0].token_ids for o in outs], stats


def test_sampled_spec_decode_preserves_distribution():
    """Rejection sampling for temperature>0: with a fixed seed the
    per-position target samples are identical with and without spec
    decode, so outputs must match token-for-token — the strongest form
    of the distribution-preservation property (reference
    rejection_sampler.py:38 semantics for one-hot drafts)."""
    base, _ = _generate_sampled(0)
    spec, (drafted, accepted) = _generate_sampled(4)
    assert base == spec
    assert all(len(t) == 24 for t in spec)
    assert drafted > 0  # drafts were actually scheduled at temp>0


def test_sampled_spec_decode_accepts_drafts():
    """Low-temperature sampled run on repetitive text: acceptance must
    actually happen (not just scheduling) for temp>0 traffic."""
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=4)
    prompt = [3, 4, 5, 6] * 14
    llm.generate([prompt], SamplingParams(
        temperature=0.05, seed=7, max_tokens=24, ignore_eos=True))
    sched = llm.engine.engine_core.scheduler
    assert sched.spec_stats_drafted > 0
    assert sched.spec_stats_accepted > 0
    llm.shutdown()


def test_draft_model_spec_decode_matches_baseline():
    """Draft-model proposer with the TARGET model as its own draft:
    greedy outputs must equal the non-spec engine, and acceptance must
    be near-perfect (the draft argmaxes the same distribution)."""
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=3, spec_decode_method="draft",
              speculative_model="tiny-llama")
    prompts = [[7, 8, 9, 10] * 12, list(range(30, 60))]
    outs = llm.generate(prompts, SamplingParams(
        temperature=0.0, max_tokens=24, ignore_eos=True))
    spec = [o.outputs[0].token_ids for o in outs]
    sched = llm.engine.engine_core.scheduler
    drafted, accepted = sched.spec_stats_drafted, sched.spec_stats_accepted
    llm.shutdown()

    base_llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
                   block_size=16, num_gpu_blocks=256, max_model_len=512,
                   max_num_batched_tokens=512, max_num_seqs=4)
    base = [o.outputs[0].token_ids for o in base_llm.generate(
        prompts, SamplingParams(temperature=0.0, max_tokens=24,
                                ignore_eos=True))]
    base_llm.shutdown()
    assert spec == base
    assert drafted > 0
    # Identical draft and target: nearly every draft position accepted.
    assert accepted / drafted > 0.9, (accepted, drafted)


def test_draft_model_different_draft_still_exact():
    """A DIFFERENT (random-init) draft model cannot change outputs —
    only the acceptance rate."""
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=256, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4,
              num_speculative_tokens=3, spec_decode_method="draft",
              speculative_model="tiny-qwen3")
    prompts = [[5, 6, 7] * 10]
    outs = llm.generate(prompts, SamplingParams(
        temperature=0.0, max_tokens=16, ignore_eos=True))
    spec = outs[0].outputs[0].token_ids
    llm.shutdown()

    base_llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
                   block_size=16, num_gpu_blocks=256, max_model_len=512,
                   max_num_batched_tokens=512, max_num_seqs=4)
    base = base_llm.generate(prompts, SamplingParams(
        temperature=0.0, max_tokens=16,
        ignore_eos=True))[0].outputs[0].token_ids
    base_llm.shutdown()
    assert spec == base
