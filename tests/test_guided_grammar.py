"""guided_grammar (GBNF-style EBNF -> regex -> lazy-DFA masks)."""

import re

import pytest

from vllm_amd.guided_grammar import GrammarError, grammar_to_regex

GRAMMAR = r'''
# yes/no or a small number
root   ::= answer | number
answer ::= "yes" | "no"
number ::= [1-9] [0-9]* ("." [0-9]+)?
'''


def test_grammar_to_regex_matches():
    pattern = grammar_to_regex(GRAMMAR)
    for ok in ["yes", "no", "7", "42", "3.14"]:
        assert re.fullmatch(pattern, ok), ok
    for bad in ["maybe", "042", "", "7."]:
        assert not re.fullmatch(pattern, bad), bad


def test_grammar_errors():
    with pytest.raises(GrammarError):
        grammar_to_regex('start ::= "x"')  # no root
    with pytest.raises(GrammarError):
        grammar_to_regex('root ::= a\na ::= "x" a | "y"')  # recursion
    with pytest.raises(GrammarError):
        grammar_to_regex('root ::= undefined_rule')
    with pytest.raises(GrammarError):
        grammar_to_regex('root ::= "unterminated')


def test_guided_grammar_e2e():
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=128,
              max_num_batched_tokens=128, max_num_seqs=2)
    [out] = llm.generate(
        [[5, 9, 13, 17]],
        SamplingParams(temperature=0.0, max_tokens=16,
                       guided_grammar=GRAMMAR))
    llm.shutdown()
    text = out.outputs[0].text
    assert re.fullmatch(grammar_to_regex(GRAMMAR), text), text


def test_guided_grammar_api():
    from fastapi.testclient import TestClient

    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-llama", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=128,
                      max_num_batched_tokens=64, max_num_seqs=2)
    app, state = make_server(args)
    with TestClient(app) as c:
        r = c.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "choose: ",
            "max_tokens": 12, "temperature": 0.0,
            "guided_grammar": 'root ::= "aa" | "bb"'})
        assert r.status_code == 200, r.text
        assert r.json()["choices"][0]["text"] in ("aa", "bb")
    state.engine.shutdown()
