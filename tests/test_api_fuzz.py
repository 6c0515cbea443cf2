"""Property-based fuzz of the OpenAI surface: arbitrary (often invalid)
request bodies must produce 2xx/4xx — never a 500 — and must not wedge
the engine (health stays green). Derandomized so CI runs are stable."""

import pytest
from fastapi.testclient import TestClient
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.openai.api_server import make_server


@pytest.fixture(scope="module")
def client():
    args = EngineArgs(
        model="tiny-llama", dtype="fp32", device="cpu", block_size=16,
        num_gpu_blocks=64, max_model_len=128,
        max_num_batched_tokens=128, max_num_seqs=2)
    app, state = make_server(args, served_model_name="tiny-llama")
    with TestClient(app, raise_server_exceptions=False) as c:
        yield c
    state.engine.shutdown()


_scalar = st.one_of(
    st.none(), st.booleans(),
    st.integers(min_value=-10, max_value=10**6),
    st.floats(allow_nan=False, allow_infinity=False,
              min_value=-100, max_value=100),
    st.text(max_size=8),
    st.lists(st.integers(min_value=-5, max_value=2000), max_size=5),
)

_completion_fields = st.dictionaries(
    st.sampled_from([
        "prompt", "max_tokens", "n", "best_of", "temperature", "top_p",
        "top_k", "min_p", "seed", "stop", "stop_token_ids", "logprobs",
        "prompt_logprobs", "echo", "stream", "logit_bias", "min_tokens",
        "repetition_penalty", "presence_penalty", "frequency_penalty",
        "guided_regex", "guided_choice", "bad_words", "ignore_eos",
        "truncate_prompt_tokens", "allowed_token_ids", "priority",
    ]),
    _scalar, max_size=6)

_chat_fields = st.dictionaries(
    st.sampled_from([
        "messages", "max_tokens", "n", "temperature", "top_p", "seed",
        "stop", "stream", "logprobs", "top_logprobs", "tools",
        "tool_choice", "response_format", "guided_json", "min_tokens",
    ]),
    _scalar, max_size=5)


def _check(client, url, body):
    r = client.post(url, json=body)
    assert r.status_code < 500, (url, body, r.status_code, r.text[:300])
    assert client.get("/health").status_code == 200, (url, body)


@settings(max_examples=80, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(extra=_completion_fields)
def test_completions_never_500(client, extra):
    body = {"model": "tiny-llama", "max_tokens": 2, **extra}
    _check(client, "/v1/completions", body)


@settings(max_examples=60, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(extra=_chat_fields)
def test_chat_never_500(client, extra):
    body = {"model": "tiny-llama", "max_tokens": 2,
            "messages": [{"role": "user", "content": "hi"}], **extra}
    _check(client, "/v1/chat/completions", body)


@settings(max_examples=30, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(body=st.dictionaries(st.text(max_size=10), _scalar, max_size=4))
def test_misc_endpoints_never_500(client, body):
    for url in ("/v1/embeddings", "/v1/responses", "/tokenize",
                "/v1/audio/transcriptions"):
        _check(client, url, body)


def test_extreme_logprob_counts_and_empty_allowed(client):
    """Regression: logprobs/prompt_logprobs beyond the vocab crashed
    the topk mid-step (poisoning the loop); an empty allowed_token_ids
    would mask every token. Huge counts now mean full-vocab, empty
    allowed sets 400."""
    for body, want in (
        ({"prompt": "x", "max_tokens": 2, "logprobs": 10**6}, 200),
        ({"prompt": "x", "max_tokens": 2,
          "prompt_logprobs": 10**6}, 200),
        ({"prompt": "x", "max_tokens": 2,
          "allowed_token_ids": []}, 400),
    ):
        r = client.post("/v1/completions",
                        json={"model": "tiny-llama", **body})
        assert r.status_code == want, (body, r.status_code, r.text[:200])
        ok = client.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "ok", "max_tokens": 2,
            "ignore_eos": True})
        assert ok.status_code == 200, (body, ok.text[:200])


def test_poison_requests_do_not_kill_the_engine(client):
    """Regression: an out-of-vocab logit_bias crashed the sampler's
    index_add_ mid-step and poisoned the engine loop for every later
    request; malformed guided_json schemas 500'd. Both must 400 and the
    NEXT normal request must still succeed."""
    for body in (
        {"prompt": "x", "logit_bias": {"99999999": 5.0},
         "max_tokens": 2},
        {"prompt": "x", "logit_bias": {"-3": 1.0}, "max_tokens": 2},
        {"prompt": "x", "guided_json":
            {"type": "array", "items": {"$ref": "#/nope"}},
         "max_tokens": 2},
        {"prompt": "x", "temperature": 0.0, "top_k": -3,
         "max_tokens": 2},
    ):
        r = client.post("/v1/completions",
                        json={"model": "tiny-llama", **body})
        assert r.status_code == 400, (body, r.status_code, r.text[:200])
        ok = client.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "ok", "max_tokens": 2,
            "ignore_eos": True})
        assert ok.status_code == 200, (body, ok.text[:200])


_nasty = st.one_of(
    st.dictionaries(st.sampled_from(["-1", "0", "1023", "1024",
                                     "99999999"]),
                    st.floats(-50, 50, allow_nan=False), max_size=3),
    st.recursive(
        st.dictionaries(st.sampled_from(["type", "items", "$ref",
                                         "properties", "enum"]),
                        st.sampled_from(["array", "object", "#/x",
                                         "integer", "zzz"]), max_size=3),
        lambda inner: st.dictionaries(
            st.sampled_from(["items", "properties"]), inner, max_size=2),
        max_leaves=4),
)


@settings(max_examples=40, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(val=_nasty, field=st.sampled_from(["logit_bias", "guided_json"]))
def test_sampler_and_grammar_payloads_never_500(client, val, field):
    body = {"model": "tiny-llama", "prompt": "x", "max_tokens": 2,
            field: val}
    _check(client, "/v1/completions", body)


def test_detokenize_out_of_range_ids(client):
    r = client.post("/detokenize", json={"tokens": [-5, 99999999]})
    assert r.status_code == 400
    r = client.post("/detokenize", json={"tokens": [72, 105]})
    assert r.status_code == 200


_anthropic_content = st.one_of(
    st.text(max_size=6),
    st.lists(st.dictionaries(
        st.sampled_from(["type", "text", "source", "content", "id"]),
        st.one_of(st.text(max_size=6), st.none(),
                  st.dictionaries(st.sampled_from(
                      ["type", "data", "media_type"]),
                      st.text(max_size=8), max_size=3)),
        max_size=3), max_size=2),
)


@settings(max_examples=40, deadline=None, derandomize=True,
          suppress_health_check=[HealthCheck.function_scoped_fixture])
@given(content=_anthropic_content,
       role=st.sampled_from(["user", "assistant", "tool", "zzz"]))
def test_anthropic_messages_never_500(client, content, role):
    body = {"model": "tiny-llama", "max_tokens": 2,
            "messages": [{"role": role, "content": content}]}
    _check(client, "/v1/messages", body)
