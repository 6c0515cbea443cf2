"""Regex / JSON-schema constrained decoding: FSM unit tests against
python re, schema→regex compilation, and end-to-end guided generation
through the engine (byte-level mock tokenizer)."""

import json
import re

import pytest

from vllm_amd.guided_json import any_json_regex, schema_to_regex
from vllm_amd.guided_regex import RegexFSM, RegexGrammar


CASES = [
    (r"abc", ["abc"], ["ab", "abcd", ""]),
    (r"a*b+c?", ["b", "aaabbc", "abbb"], ["a", "c", "ac"]),
    (r"(foo|bar)+", ["foo", "barfoo"], ["fo", "foobaz", ""]),
    (r"[a-c]{2,3}", ["ab", "abc", "ccc"], ["a", "abcd", "ad"]),
    (r"-?[0-9]+(\.[0-9]+)?", ["1", "-2.5", "007"], ["-", "1.", ".5"]),
    (r"[^x]+", ["abc", "yz"], ["axb", "x", ""]),
    (r"\d{3}-\d{4}", ["555-1234"], ["5551234", "55-1234"]),
    (r"a{2}", ["aa"], ["a", "aaa"]),
    (r"a{2,}", ["aa", "aaaa"], ["a"]),
    (r"(ab)*", ["", "abab"], ["aba"]),
    (r"\w+@\w+\.(com|org)", ["a_1@b.com"], ["a@b.net", "@b.com"]),
    (r'"[^"]*"', ['""', '"hi"'], ['"', 'hi']),
]


@pytest.mark.parametrize("pattern,good,bad", CASES)
def test_fsm_matches_re(pattern, good, bad):
    fsm = RegexFSM(pattern)
    for s in good:
        assert re.fullmatch(pattern, s), f"test bug: {pattern} {s}"
        assert fsm.fullmatch(s), f"{pattern} should match {s!r}"
    for s in bad:
        assert not re.fullmatch(pattern, s), f"test bug: {pattern} {s}"
        assert not fsm.fullmatch(s), f"{pattern} should reject {s!r}"


def test_schema_regex_accepts_valid_json():
    schema = {
        "type": "object",
        "properties": {
            "name": {"type": "string"},
            "age": {"type": "integer"},
            "tags": {"type": "array", "items": {"type": "string"}},
            "role": {"enum": ["admin", "user"]},
        },
        "required": ["name", "age"],
    }
    pattern = schema_to_regex(schema)
    fsm = RegexFSM(pattern)
    ok = '{"name":"bo","age":3,"tags":["x","y"],"role":"user"}'
    assert fsm.fullmatch(ok)
    assert fsm.fullmatch(
        '{"name": "bo", "age": -1, "tags": [], "role": "admin"}')
    assert not fsm.fullmatch('{"age":3}')
    assert not fsm.fullmatch('{"name":"bo","age":"x","tags":[],'
                             '"role":"user"}')


def test_schema_regex_nested_and_refs():
    schema = {
        "$defs": {"pt": {"type": "object", "properties": {
            "x": {"type": "number"}, "y": {"type": "number"}}}},
        "type": "object",
        "properties": {
            "a": {"$ref": "#/$defs/pt"},
            "ok": {"type": "boolean"},
            "opt": {"anyOf": [{"type": "null"}, {"type": "integer"}]},
        },
    }
    fsm = RegexFSM(schema_to_regex(schema))
    assert fsm.fullmatch('{"a":{"x":1.5,"y":-2},"ok":true,"opt":null}')
    assert fsm.fullmatch('{"a": {"x": 1e3, "y": 0}, "ok": false, "opt": 7}')
    assert not fsm.fullmatch('{"a":{},"ok":true,"opt":null}')


def test_any_json_regex():
    fsm = RegexFSM(any_json_regex(depth=2))
    assert fsm.fullmatch('{"k": [1, "two", {"three": null}], "b": true}')
    assert not fsm.fullmatch('[1,2]')
    assert not fsm.fullmatch('{"k": }')


class _ByteTok:
    """Standalone byte tokenizer for grammar unit tests."""

    vocab_size = 258
    bos_token_id = 256
    eos_token_id = 257

    def decode(self, ids, skip_special_tokens=True):
        return bytes(i for i in ids if i < 256).decode(
            "utf-8", errors="replace")


def test_regex_grammar_masks_and_advance():
    tok = _ByteTok()
    g = RegexGrammar(r"(yes|no)!", tok, eos_token_id=257)
    s = g.initial_state()
    allowed = g.allowed_tokens(s)
    assert allowed == {ord("y"), ord("n")}
    s = g.advance(s, ord("y"))
    assert g.allowed_tokens(s) == {ord("e")}
    s = g.advance(s, ord("e"))
    s = g.advance(s, ord("s"))
    assert g.allowed_tokens(s) == {ord("!")}
    s = g.advance(s, ord("!"))
    assert g.allowed_tokens(s) == {257}  # accepting → only EOS
    assert g.is_exhausted(s)
    assert g.advance(s, ord("x")) is None


def test_e2e_guided_regex_and_json():
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
              block_size=16, num_gpu_blocks=128, max_model_len=512,
              max_num_batched_tokens=512, max_num_seqs=4)
    pattern = r"[0-9]{2}-[0-9]{2}"
    out = llm.generate(
        [[5, 6, 7]],
        SamplingParams(temperature=0.0, max_tokens=20,
                       guided_regex=pattern),
    )[0]
    text = out.outputs[0].text
    assert re.fullmatch(pattern, text), text

    # Bias EOS/closers so the (random-weight) model closes the document
    # at the first accepting state instead of growing unbounded values;
    # bounded value types make termination structural.
    bias = {257: 50.0, ord("}"): 20.0, ord('"'): 10.0}
    schema = {"type": "object", "properties": {
        "a": {"type": "boolean"}, "b": {"enum": ["x", "y"]},
        "c": {"type": "string", "maxLength": 3}}}
    out = llm.generate(
        [[9, 8, 7]],
        SamplingParams(temperature=0.0, max_tokens=60,
                       guided_json=schema, logit_bias=bias),
    )[0]
    assert out.outputs[0].finish_reason == "stop", out.outputs[0].text
    doc = json.loads(out.outputs[0].text)
    assert isinstance(doc["a"], bool) and doc["b"] in ("x", "y")
    assert len(doc["c"]) <= 3

    out = llm.generate(
        [[1, 2]],
        SamplingParams(temperature=0.0, max_tokens=80,
                       guided_json_object=True, logit_bias=bias),
    )[0]
    llm.shutdown()
    assert out.outputs[0].finish_reason == "stop", out.outputs[0].text
    assert isinstance(json.loads(out.outputs[0].text), dict)
