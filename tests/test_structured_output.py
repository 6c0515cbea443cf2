"""Structured output (guided_choice) tests: the constrained greedy
output must be exactly one of the allowed strings."""

from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams
from vllm_amd.structured_output import CompiledGrammar


def test_trie_grammar_mechanics():
    g = CompiledGrammar([[5, 6], [5, 7, 8], [9]], eos_token_id=0)
    s = g.initial_state()
    assert g.allowed_tokens(s) == {5, 9}
    s = g.advance(s, 5)
    assert g.allowed_tokens(s) == {6, 7}
    s2 = g.advance(s, 6)
    assert g.allowed_tokens(s2) == {0}  # terminal -> eos only
    assert g.advance(s, 99) is None


def test_guided_choice_e2e():
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=4)
    choices = ["yes", "no", "maybe not"]
    tok = llm.engine.tokenizer
    params = SamplingParams(temperature=0.0, max_tokens=16,
                            guided_choice=choices)
    outs = llm.generate(["is this constrained?", "answer please"],
                        [params, params])
    llm.shutdown()
    bos = getattr(tok.tokenizer, "bos_token_id", None)
    allowed_seqs = []
    for c in choices:
        ids = tok.encode(c)
        if bos is not None and ids and ids[0] == bos:
            ids = ids[1:]
        allowed_seqs.append(ids)
    for o in outs:
        toks = o.outputs[0].token_ids
        assert o.outputs[0].finish_reason == "stop"
        # Generated tokens (minus the final eos if present) must equal one
        # of the choice token sequences exactly.
        body = toks[:-1] if toks and toks[-1] == tok.eos_token_id else toks
        assert any(body == seq for seq in allowed_seqs), (body, allowed_seqs)


def test_guided_choice_mixed_batch():
    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=256,
              max_num_batched_tokens=256, max_num_seqs=4)
    g = SamplingParams(temperature=0.0, max_tokens=8,
                       guided_choice=["alpha", "beta"])
    free = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
    outs = llm.generate([[1, 2, 3], [4, 5, 6]], [g, free])
    llm.shutdown()
    assert outs[0].outputs[0].finish_reason == "stop"
    assert len(outs[1].outputs[0].token_ids) == 8
