"""OpenAI API server tests on CPU (tiny model, mock tokenizer).

Covers completions (non-stream + stream), chat completions, models,
tokenize/detokenize, health and metrics — the surface of
vllm/entrypoints/openai (reference tests/entrypoints/openai pattern).
"""

import json

import pytest
from fastapi.testclient import TestClient

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.openai.api_server import make_server


@pytest.fixture(scope="module")
def client():
    args = EngineArgs(
        model="tiny-llama",
        dtype="fp32",
        device="cpu",
        block_size=16,
        num_gpu_blocks=256,
        max_model_len=512,
        max_num_batched_tokens=512,
        max_num_seqs=8,
    )
    app, state = make_server(args, served_model_name="tiny-llama")
    with TestClient(app) as c:
        yield c
    state.engine.shutdown()


def test_health_and_version(client):
    assert client.get("/health").status_code == 200
    assert "version" in client.get("/version").json()


def test_models(client):
    data = client.get("/v1/models").json()
    assert data["object"] == "list"
    assert data["data"][0]["id"] == "tiny-llama"


def test_completions(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": "hello world this is a test",
        "max_tokens": 8,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["object"] == "text_completion"
    assert data["usage"]["completion_tokens"] == 8
    assert data["choices"][0]["finish_reason"] == "length"


def test_completions_token_ids_prompt(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": [5, 6, 7, 8, 9],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["prompt_tokens"] == 5


def test_completions_stream(client):
    with client.stream("POST", "/v1/completions", json={
        "model": "tiny-llama",
        "prompt": "stream me",
        "max_tokens": 6,
        "temperature": 0.0,
        "ignore_eos": True,
        "stream": True,
    }) as r:
        assert r.status_code == 200
        chunks = []
        for line in r.iter_lines():
            if line.startswith("data: "):
                payload = line[6:]
                if payload == "[DONE]":
                    break
                chunks.append(json.loads(payload))
        assert len(chunks) >= 1
        assert chunks[-1]["choices"][0]["finish_reason"] == "length"


def test_chat_completions(client):
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [
            {"role": "system", "content": "You are a test."},
            {"role": "user", "content": "Say something."},
        ],
        "max_tokens": 5,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["object"] == "chat.completion"
    assert data["choices"][0]["message"]["role"] == "assistant"
    assert data["usage"]["completion_tokens"] == 5


def test_chat_stream(client):
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
        "stream": True,
    }) as r:
        got_role = False
        got_done = False
        for line in r.iter_lines():
            if line.startswith("data: "):
                payload = line[6:]
                if payload == "[DONE]":
                    got_done = True
                    break
                d = json.loads(payload)
                if d["choices"][0]["delta"].get("role") == "assistant":
                    got_role = True
        assert got_role and got_done


def test_tokenize_roundtrip(client):
    r = client.post("/tokenize", json={"prompt": "round trip"})
    toks = r.json()["tokens"]
    assert r.json()["count"] == len(toks) > 0
    r2 = client.post("/detokenize", json={"tokens": toks})
    assert isinstance(r2.json()["prompt"], str)


def test_metrics(client):
    text = client.get("/metrics").text
    assert "vllm_amd:num_requests_total" in text
    assert "vllm_amd:kv_blocks_free" in text


def test_stop_string(client):
    # Force a stop string that can't appear -> finishes by length.
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": "check stops",
        "max_tokens": 4,
        "temperature": 0.0,
        "stop": ["never"],
        "ignore_eos": True,
    })
    assert r.json()["choices"][0]["finish_reason"] == "length"


def test_concurrent_requests(client):
    import concurrent.futures as cf

    def one(i):
        r = client.post("/v1/completions", json={
            "model": "tiny-llama",
            "prompt": f"request number {i}",
            "max_tokens": 6,
            "temperature": 0.0,
            "ignore_eos": True,
        })
        return r.json()["usage"]["completion_tokens"]

    with cf.ThreadPoolExecutor(4) as ex:
        results = list(ex.map(one, range(8)))
    assert results == [6] * 8


def test_bench_serving_tool():
    """The serving benchmark tool runs end-to-end on CPU."""
    import subprocess
    import sys

    r = subprocess.run(
        [sys.executable, "benchmarks/bench_serving.py", "--num-prompts", "4",
         "--qps", "100", "--input-len", "16", "--output-len", "4"],
        capture_output=True, text=True, timeout=240,
    )
    assert r.returncode == 0, r.stderr[-800:]
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert out["output_tokens_per_s"] > 0
    assert out["ttft_ms"]["p50"] > 0


def test_anthropic_messages(client):
    r = client.post("/v1/messages", json={
        "model": "tiny-llama",
        "max_tokens": 6,
        "messages": [{"role": "user", "content": "hello"}],
        "temperature": 0.0,
    })
    assert r.status_code == 200, r.text
    d = r.json()
    assert d["type"] == "message" and d["role"] == "assistant"
    assert d["usage"]["output_tokens"] >= 1
    assert d["content"][0]["type"] == "text"


def test_anthropic_messages_stream(client):
    with client.stream("POST", "/v1/messages", json={
        "model": "tiny-llama",
        "max_tokens": 4,
        "messages": [{"role": "user", "content": "hi"}],
        "temperature": 0.0,
        "stream": True,
    }) as r:
        events = [ln.split(": ", 1)[1] for ln in r.iter_lines()
                  if ln.startswith("event: ")]
    assert events[0] == "message_start"
    assert "message_stop" in events


def test_chat_response_format_json_schema(client):
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "Emit JSON."}],
        "max_tokens": 60,
        "temperature": 0.0,
        "logit_bias": {"257": 50.0, "125": 20.0, "34": 10.0},
        "response_format": {"type": "json_schema", "json_schema": {
            "name": "out", "schema": {"type": "object", "properties": {
                "ok": {"type": "boolean"},
                "kind": {"enum": ["a", "b"]}}}}},
    })
    assert r.status_code == 200, r.text
    import json as _json
    doc = _json.loads(r.json()["choices"][0]["message"]["content"])
    assert isinstance(doc["ok"], bool) and doc["kind"] in ("a", "b")


def test_completions_guided_regex(client):
    import re as _re
    pattern = r"[ab]{3}-[0-9]{2}"
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": "match: ",
        "max_tokens": 20,
        "temperature": 0.0,
        "guided_regex": pattern,
    })
    assert r.status_code == 200, r.text
    assert _re.fullmatch(pattern, r.json()["choices"][0]["text"])


def test_sleep_wake_endpoints(client):
    assert client.get("/is_sleeping").json() == {"is_sleeping": False}
    r = client.post("/sleep?level=1")
    assert r.status_code == 200
    assert client.get("/is_sleeping").json() == {"is_sleeping": True}
    assert client.post("/wake_up").status_code == 200
    assert client.get("/is_sleeping").json() == {"is_sleeping": False}
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "after wake", "max_tokens": 4,
        "temperature": 0.0, "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["completion_tokens"] == 4


def test_metrics_histograms(client):
    # non-stream + stream requests populate the latency histograms
    client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "hist test", "max_tokens": 4,
        "temperature": 0.0, "ignore_eos": True})
    with client.stream("POST", "/v1/completions", json={
            "model": "tiny-llama", "prompt": "hist stream", "max_tokens": 4,
            "temperature": 0.0, "ignore_eos": True, "stream": True}) as r:
        for _ in r.iter_lines():
            pass
    text = client.get("/metrics").text
    assert "vllm_amd:e2e_request_latency_seconds_count" in text
    assert "vllm_amd:time_to_first_token_seconds_bucket" in text
    assert "vllm_amd:time_per_output_token_seconds_sum" in text
    assert "vllm_amd:request_generation_tokens_count" in text
    assert "vllm_amd:prefix_cache_queries_total" in text
    assert "vllm_amd:num_preemptions_total" in text
    # counts are cumulative and > 0
    for line in text.splitlines():
        if line.startswith("vllm_amd:e2e_request_latency_seconds_count"):
            assert float(line.split()[-1]) >= 2


def test_run_batch_jsonl(tmp_path):
    """Offline OpenAI batch runner: JSONL in -> JSONL out through the
    in-process ASGI app (reference: entrypoints/openai/run_batch.py)."""
    import subprocess
    import sys

    inp = tmp_path / "in.jsonl"
    outp = tmp_path / "out.jsonl"
    lines = [
        {"custom_id": "a", "method": "POST", "url": "/v1/completions",
         "body": {"model": "tiny-llama", "prompt": "one", "max_tokens": 4,
                  "temperature": 0.0, "ignore_eos": True}},
        {"custom_id": "b", "method": "POST", "url": "/v1/chat/completions",
         "body": {"model": "tiny-llama",
                  "messages": [{"role": "user", "content": "two"}],
                  "max_tokens": 4, "temperature": 0.0, "ignore_eos": True}},
        {"custom_id": "c", "method": "GET", "url": "/nope", "body": {}},
    ]
    inp.write_text("\n".join(json.dumps(x) for x in lines))
    r = subprocess.run(
        [sys.executable, "-m", "vllm_amd", "run-batch", "-i", str(inp),
         "-o", str(outp), "--model", "tiny-llama", "--dtype", "fp32",
         "--device", "cpu", "--block-size", "16", "--num-gpu-blocks", "64",
         "--max-model-len", "256"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    results = {json.loads(ln)["custom_id"]: json.loads(ln)
               for ln in outp.read_text().splitlines()}
    assert results["a"]["error"] is None
    assert results["a"]["response"]["body"]["usage"][
        "completion_tokens"] == 4
    assert results["b"]["response"]["body"]["choices"][0][
        "message"] is not None
    assert results["c"]["error"] is not None


def test_embeddings_endpoint(client):
    r = client.post("/v1/embeddings", json={
        "model": "tiny-llama",
        "input": ["hello world", "second input text"],
    })
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["object"] == "list"
    assert len(data["data"]) == 2
    emb = data["data"][0]["embedding"]
    assert isinstance(emb, list) and len(emb) > 0
    assert data["usage"]["prompt_tokens"] > 0
    # base64 format round-trips
    r2 = client.post("/v1/embeddings", json={
        "model": "tiny-llama", "input": "hello world",
        "encoding_format": "base64"})
    import base64
    import struct
    raw = base64.b64decode(r2.json()["data"][0]["embedding"])
    vec = struct.unpack(f"<{len(raw)//4}f", raw)
    assert len(vec) == len(emb)
    # determinism: same input -> same embedding
    assert list(vec) == pytest.approx(emb, abs=1e-6)


def test_embeddings_mean_pooling(client):
    r = client.post("/v1/embeddings", json={
        "model": "tiny-llama", "input": "mean pool me",
        "pooling": "mean"})
    assert r.status_code == 200, r.text
    vec = r.json()["data"][0]["embedding"]
    assert len(vec) > 0


def test_parallel_sampling_api(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "n test", "max_tokens": 5,
        "n": 3, "temperature": 1.0, "seed": 11, "ignore_eos": True})
    assert r.status_code == 200, r.text
    data = r.json()
    assert [c["index"] for c in data["choices"]] == [0, 1, 2]
    assert data["usage"]["completion_tokens"] == 15
    assert all(c["finish_reason"] == "length" for c in data["choices"])
    # (seeded branch divergence is asserted on token ids in
    # test_engine_cpu.test_parallel_sampling_n — the mock tokenizer
    # decodes arbitrary sampled ids to empty text here)
    # chat non-stream n=2
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 4, "n": 2, "temperature": 0.0, "ignore_eos": True})
    assert r.status_code == 200, r.text
    assert len(r.json()["choices"]) == 2
    # streaming completions with n=2: every chunk labels its branch
    with client.stream("POST", "/v1/completions", json={
            "model": "tiny-llama", "prompt": "s", "max_tokens": 4,
            "n": 2, "temperature": 0.0, "ignore_eos": True,
            "stream": True}) as r:
        idxs = set()
        for line in r.iter_lines():
            if line.startswith("data: ") and line != "data: [DONE]":
                idxs.add(json.loads(line[6:])["choices"][0]["index"])
    assert idxs == {0, 1}


def test_score_and_rerank(client):
    r = client.post("/v1/score", json={
        "model": "tiny-llama", "text_1": "query text",
        "text_2": ["doc one", "doc two", "query text"]})
    assert r.status_code == 200, r.text
    data = r.json()["data"]
    assert len(data) == 3
    # identical texts -> maximal self-similarity
    assert data[2]["score"] == pytest.approx(1.0, abs=1e-5)
    assert all(-1.0 <= d["score"] <= 1.0 + 1e-6 for d in data)

    r = client.post("/v1/rerank", json={
        "model": "tiny-llama", "query": "the query",
        "documents": ["aaa", "the query", "bbb"], "top_n": 2})
    assert r.status_code == 200, r.text
    results = r.json()["results"]
    assert len(results) == 2
    # exact-match document ranks first
    assert results[0]["index"] == 1
    assert results[0]["relevance_score"] >= results[1]["relevance_score"]

    r = client.post("/pooling", json={
        "model": "tiny-llama", "input": "pool me"})
    assert r.status_code == 200 and len(r.json()["data"]) == 1


def test_stream_include_usage(client):
    for url, body in [
        ("/v1/completions", {"model": "tiny-llama", "prompt": "u",
                             "max_tokens": 3, "temperature": 0.0,
                             "ignore_eos": True}),
        ("/v1/chat/completions", {"model": "tiny-llama",
                                  "messages": [{"role": "user",
                                                "content": "u"}],
                                  "max_tokens": 3, "temperature": 0.0,
                                  "ignore_eos": True}),
    ]:
        body.update(stream=True, stream_options={"include_usage": True})
        with client.stream("POST", url, json=body) as r:
            chunks = [json.loads(ln[6:]) for ln in r.iter_lines()
                      if ln.startswith("data: ") and ln != "data: [DONE]"]
        assert chunks[-1].get("usage", {}).get("completion_tokens") == 3, url
        assert chunks[-1]["choices"] == []


def test_completions_prompt_logprobs(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": [5, 9, 13, 17, 21],
        "max_tokens": 1, "temperature": 0.0, "ignore_eos": True,
        "prompt_logprobs": 2})
    assert r.status_code == 200, r.text
    plp = r.json()["choices"][0]["prompt_logprobs"]
    assert len(plp) == 4  # prompt_len - 1
    assert all(len(d) >= 2 for d in plp)


def test_responses_api(client):
    r = client.post("/v1/responses", json={
        "model": "tiny-llama", "input": "say hi",
        "instructions": "be brief", "max_output_tokens": 5,
        "temperature": 0.0})
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["object"] == "response"
    assert data["status"] == "completed"
    (item,) = data["output"]
    assert item["type"] == "message" and item["role"] == "assistant"
    assert item["content"][0]["type"] == "output_text"
    assert data["usage"]["output_tokens"] == 5


def test_responses_api_stream(client):
    with client.stream("POST", "/v1/responses", json={
            "model": "tiny-llama",
            "input": [{"role": "user", "content": [
                {"type": "input_text", "text": "stream"}]}],
            "max_output_tokens": 4, "temperature": 0.0,
            "stream": True}) as r:
        events = [ln.split(": ", 1)[1] for ln in r.iter_lines()
                  if ln.startswith("event: ")]
    assert events[0] == "response.created"
    assert "response.output_item.added" in events
    assert "response.completed" in events


def test_openai_shaped_logprobs(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": [5, 9, 13], "max_tokens": 3,
        "temperature": 0.0, "ignore_eos": True, "logprobs": 2})
    assert r.status_code == 200, r.text
    lp = r.json()["choices"][0]["logprobs"]
    assert len(lp["tokens"]) == 3
    assert len(lp["token_logprobs"]) == 3
    assert all(v is None or v <= 0 for v in lp["token_logprobs"])
    assert all(len(t) >= 1 for t in lp["top_logprobs"])

    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 3, "temperature": 0.0, "ignore_eos": True,
        "logprobs": True, "top_logprobs": 2})
    assert r.status_code == 200, r.text
    content = r.json()["choices"][0]["logprobs"]["content"]
    assert len(content) == 3
    assert {"token", "logprob", "bytes", "top_logprobs"} <= set(content[0])


def test_api_key_auth_and_request_id():
    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-llama", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=128,
                      max_num_batched_tokens=64, max_num_seqs=2)
    app, state = make_server(args, api_key="sk-test")
    with TestClient(app) as c:
        # /v1 without key -> 401; /health stays open
        r = c.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "x", "max_tokens": 2})
        assert r.status_code == 401
        assert c.get("/health").status_code == 200
        r = c.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "x", "max_tokens": 2,
            "temperature": 0.0, "ignore_eos": True},
            headers={"Authorization": "Bearer sk-test"})
        assert r.status_code == 200
        assert r.headers.get("X-Request-Id")
        # client-supplied id is echoed
        r = c.get("/health", headers={"X-Request-Id": "rid-42"})
        assert r.headers["X-Request-Id"] == "rid-42"
    state.engine.shutdown()


def test_profile_endpoints(tmp_path):
    from vllm_amd.config import ObservabilityConfig
    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-llama", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=128,
                      max_num_batched_tokens=64, max_num_seqs=2)
    app, state = make_server(args)
    state.engine.config.observability_config = ObservabilityConfig(
        profile_dir=str(tmp_path))
    with TestClient(app) as c:
        assert c.post("/stop_profile").status_code == 400  # not running
        assert c.post("/start_profile").status_code == 200
        assert c.post("/start_profile").status_code == 400  # double start
        c.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "p", "max_tokens": 3,
            "temperature": 0.0, "ignore_eos": True})
        r = c.post("/stop_profile")
        assert r.status_code == 200
        trace = r.json()["trace"]
        assert trace.startswith(str(tmp_path))
        import os
        assert os.path.getsize(trace) > 0
    state.engine.shutdown()


def test_anthropic_messages_tools(client):
    """Anthropic tool specs render into the prompt and hermes-format
    model output maps to tool_use content blocks (no tool call emitted
    by the tiny model -> plain text content, request still valid)."""
    r = client.post("/v1/messages", json={
        "model": "tiny-llama", "max_tokens": 6,
        "messages": [{"role": "user", "content": "weather?"}],
        "tools": [{"name": "get_weather",
                   "description": "weather lookup",
                   "input_schema": {"type": "object", "properties": {
                       "city": {"type": "string"}}}}],
        "temperature": 0.0,
    })
    assert r.status_code == 200, r.text
    data = r.json()
    assert data["type"] == "message"
    assert data["content"][0]["type"] in ("text", "tool_use")
    # tool_result round-trip turn is accepted
    r = client.post("/v1/messages", json={
        "model": "tiny-llama", "max_tokens": 4,
        "messages": [
            {"role": "user", "content": "weather?"},
            {"role": "assistant", "content": "checking"},
            {"role": "user", "content": [
                {"type": "tool_result", "tool_use_id": "toolu_1",
                 "content": "sunny"}]},
        ],
        "temperature": 0.0,
    })
    assert r.status_code == 200, r.text


def test_anthropic_tool_use_mapping_unit():
    """Unit-map a hermes call into a tool_use block via the handler's
    parser (deterministic path check)."""
    from vllm_amd.entrypoints.tool_parser import parse_hermes_tool_calls

    text = ('<tool_call>{"name": "get_weather", "arguments": '
            '{"city": "Oslo"}}</tool_call>')
    content, calls = parse_hermes_tool_calls(text)
    block = {"type": "tool_use",
             "id": calls[0].id.replace("call_", "toolu_", 1),
             "name": calls[0].name,
             "input": json.loads(calls[0].arguments)}
    assert block["name"] == "get_weather"
    assert block["input"] == {"city": "Oslo"}
    assert block["id"].startswith("toolu_")


def test_chat_stream_n2(client):
    with client.stream("POST", "/v1/chat/completions", json={
            "model": "tiny-llama",
            "messages": [{"role": "user", "content": "hi"}],
            "max_tokens": 4, "n": 2, "temperature": 0.0,
            "ignore_eos": True, "stream": True}) as r:
        finishes = {}
        idxs = set()
        for ln in r.iter_lines():
            if ln.startswith("data: ") and ln != "data: [DONE]":
                ch = json.loads(ln[6:])["choices"]
                if ch:
                    idxs.add(ch[0]["index"])
                    if ch[0].get("finish_reason"):
                        finishes[ch[0]["index"]] = ch[0]["finish_reason"]
    assert idxs == {0, 1}
    assert finishes == {0: "length", 1: "length"}


def test_multi_turn_prefix_cache_hit(client):
    """Second chat turn extends the first conversation: its prompt must
    hit the prefix cache (prefix_cache_hits grows)."""
    conv = [{"role": "user", "content": "tell me about caching " * 8}]
    r1 = client.post("/v1/chat/completions", json={
        "model": "tiny-llama", "messages": conv, "max_tokens": 4,
        "temperature": 0.0, "ignore_eos": True})
    assert r1.status_code == 200
    before = {ln.split()[0]: float(ln.split()[-1])
              for ln in client.get("/metrics").text.splitlines()
              if ln.startswith("vllm_amd:prefix_cache")}
    conv = conv + [
        {"role": "assistant",
         "content": r1.json()["choices"][0]["message"]["content"] or "ok"},
        {"role": "user", "content": "more"},
    ]
    r2 = client.post("/v1/chat/completions", json={
        "model": "tiny-llama", "messages": conv, "max_tokens": 4,
        "temperature": 0.0, "ignore_eos": True})
    assert r2.status_code == 200
    after = {ln.split()[0]: float(ln.split()[-1])
             for ln in client.get("/metrics").text.splitlines()
             if ln.startswith("vllm_amd:prefix_cache")}
    assert after["vllm_amd:prefix_cache_hits_total"] > \
        before["vllm_amd:prefix_cache_hits_total"]


def test_priority_param_reaches_scheduler():
    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(model="tiny-llama", dtype="fp32", device="cpu",
                      block_size=16, num_gpu_blocks=64, max_model_len=128,
                      max_num_batched_tokens=64, max_num_seqs=4,
                      scheduling_policy="priority")
    app, state = make_server(args)
    with TestClient(app) as c:
        r = c.post("/v1/completions", json={
            "model": "tiny-llama", "prompt": "p", "max_tokens": 2,
            "temperature": 0.0, "ignore_eos": True, "priority": -5})
        assert r.status_code == 200, r.text
    state.engine.shutdown()


def test_usage_reports_cached_tokens(client):
    body = {"model": "tiny-llama",
            "prompt": "repeatable cached prompt text " * 4,
            "max_tokens": 2, "temperature": 0.0, "ignore_eos": True}
    client.post("/v1/completions", json=body)
    r = client.post("/v1/completions", json=body)
    details = r.json()["usage"].get("prompt_tokens_details")
    assert details and details["cached_tokens"] > 0


def test_completions_batched_prompts(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": ["one two", "three four five"],
        "max_tokens": 4, "temperature": 0.0, "ignore_eos": True})
    assert r.status_code == 200
    body = r.json()
    assert [c["index"] for c in body["choices"]] == [0, 1]
    assert all(c["finish_reason"] == "length" for c in body["choices"])
    assert body["usage"]["completion_tokens"] == 8


def test_completions_batched_prompts_with_n(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": ["a b", "c d"], "n": 2, "seed": 5,
        "max_tokens": 3, "temperature": 1.0, "ignore_eos": True})
    assert r.status_code == 200
    body = r.json()
    # prompt-major indexing: p*n + branch
    assert [c["index"] for c in body["choices"]] == [0, 1, 2, 3]


def test_completions_best_of(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "pick the best", "n": 1, "best_of": 4,
        "seed": 11, "max_tokens": 4, "temperature": 1.0,
        "ignore_eos": True})
    assert r.status_code == 200
    body = r.json()
    assert len(body["choices"]) == 1
    assert body["choices"][0]["index"] == 0
    # logprobs were forced internally for selection but not returned
    assert body["choices"][0].get("logprobs") is None
    # invalid combos rejected
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "x", "n": 3, "best_of": 2,
        "max_tokens": 2})
    assert r.status_code != 200 or "error" in r.json()


def test_anthropic_count_tokens(client):
    r = client.post("/v1/messages/count_tokens", json={
        "model": "tiny-llama", "max_tokens": 8,
        "messages": [{"role": "user", "content": "count these tokens"}]})
    assert r.status_code == 200
    n = r.json()["input_tokens"]
    assert isinstance(n, int) and n > 0


def test_truncate_prompt_tokens_and_allowed_ids(client):
    # truncate_prompt_tokens keeps only the last N prompt tokens.
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": [5, 6, 7, 8, 9, 10, 11, 12],
        "truncate_prompt_tokens": 3,
        "max_tokens": 2,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    assert r.json()["usage"]["prompt_tokens"] == 3
    # allowed_token_ids restricts sampling to the given vocabulary.
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": [5, 6, 7],
        "allowed_token_ids": [42, 43],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    # Token ids come back detokenized; check via logprobs-free route:
    # re-request with logprobs to read the chosen ids.
    r = client.post("/v1/completions", json={
        "model": "tiny-llama",
        "prompt": [5, 6, 7],
        "allowed_token_ids": [42, 43],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
        "logprobs": 1,
    })
    toks = r.json()["choices"][0]["logprobs"]["tokens"]
    assert len(toks) == 4
    # Mock tokenizer is byte-level: id 42 -> '*', 43 -> '+'. Every
    # sampled token must come from the allowed set.
    assert set(toks) <= {"*", "+"}, toks


def test_suffix_rejected(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "abc", "suffix": "def",
        "max_tokens": 2})
    assert r.status_code == 400
    assert "suffix" in r.json()["message"]


def test_responses_store_background_and_chaining(client):
    import time as _time

    # store=true (default): result retrievable by id.
    r = client.post("/v1/responses", json={
        "model": "tiny-llama", "input": "hello", "max_output_tokens": 6,
        "temperature": 0.0})
    assert r.status_code == 200, r.text
    rid = r.json()["id"]
    got = client.get(f"/v1/responses/{rid}")
    assert got.status_code == 200
    assert got.json()["output_text"] == r.json()["output_text"]
    # previous_response_id chains the stored output into the context.
    r2 = client.post("/v1/responses", json={
        "model": "tiny-llama", "input": "continue",
        "previous_response_id": rid, "max_output_tokens": 4,
        "temperature": 0.0})
    assert r2.status_code == 200, r2.text
    # background: immediate queued envelope, poll to completion.
    rb = client.post("/v1/responses", json={
        "model": "tiny-llama", "input": "bg", "background": True,
        "max_output_tokens": 4, "temperature": 0.0})
    assert rb.status_code == 200, rb.text
    bid = rb.json()["id"]
    assert rb.json()["status"] in ("queued", "in_progress")
    for _ in range(100):
        body = client.get(f"/v1/responses/{bid}").json()
        if body["status"] == "completed":
            break
        _time.sleep(0.05)
    assert body["status"] == "completed", body
    assert isinstance(body["output_text"], str)
    # unknown id -> 404; background without store -> 400.
    assert client.get("/v1/responses/resp-nope").status_code == 404
    bad = client.post("/v1/responses", json={
        "model": "tiny-llama", "input": "x", "background": True,
        "store": False})
    assert bad.status_code == 400


def test_model_retrieval_and_ping(client):
    r = client.get("/v1/models/tiny-llama")
    assert r.status_code == 200 and r.json()["id"] == "tiny-llama"
    assert client.get("/v1/models/nope").status_code == 404
    assert client.get("/ping").status_code == 200
    assert client.post("/ping").status_code == 200


def test_chat_stream_logprobs(client):
    toks = []
    with client.stream("POST", "/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "max_tokens": 4, "temperature": 0.0, "ignore_eos": True,
        "stream": True, "logprobs": True, "top_logprobs": 1,
    }) as r:
        assert r.status_code == 200
        for ln in r.iter_lines():
            if ln.startswith("data: ") and ln != "data: [DONE]":
                ch = json.loads(ln[6:])["choices"][0]
                lp = ch.get("logprobs")
                if lp:
                    toks += lp["content"]
    assert len(toks) == 4
    for t in toks:
        assert isinstance(t["logprob"], float)
        assert t["top_logprobs"]


def test_completions_stream_logprobs(client):
    got = 0
    with client.stream("POST", "/v1/completions", json={
        "model": "tiny-llama", "prompt": "hi", "max_tokens": 4,
        "temperature": 0.0, "ignore_eos": True, "stream": True,
        "logprobs": 1,
    }) as r:
        assert r.status_code == 200
        for ln in r.iter_lines():
            if ln.startswith("data: ") and ln != "data: [DONE]":
                ch = json.loads(ln[6:])["choices"]
                if ch and ch[0].get("logprobs"):
                    got += len(ch[0]["logprobs"]["token_logprobs"])
    assert got == 4


def test_echo_with_logprobs_covers_prompt(client):
    r = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": [5, 6, 7], "max_tokens": 3,
        "temperature": 0.0, "ignore_eos": True,
        "echo": True, "logprobs": 1})
    assert r.status_code == 200, r.text
    lp = r.json()["choices"][0]["logprobs"]
    # 3 prompt tokens (first with null logprob) + 3 generated.
    assert len(lp["tokens"]) == 6
    assert lp["token_logprobs"][0] is None
    assert all(isinstance(v, float) for v in lp["token_logprobs"][1:])


def test_include_stop_str_in_output(client):
    # Find a character the greedy continuation actually produces.
    base = client.post("/v1/completions", json={
        "model": "tiny-llama", "prompt": "hello there friend",
        "max_tokens": 12, "temperature": 0.0, "ignore_eos": True,
    }).json()["choices"][0]["text"]
    printable = [ch for ch in base if ch.isprintable() and ch != " "]
    if not printable:
        import pytest
        pytest.skip("tiny model produced no printable text")
    stop = printable[len(printable) // 2]
    common = {"model": "tiny-llama", "prompt": "hello there friend",
              "max_tokens": 12, "temperature": 0.0, "ignore_eos": True,
              "stop": [stop]}
    r1 = client.post("/v1/completions", json=common).json()["choices"][0]
    r2 = client.post("/v1/completions", json={
        **common, "include_stop_str_in_output": True,
    }).json()["choices"][0]
    assert r1["finish_reason"] == "stop"
    assert stop not in r1["text"]
    assert r2["text"].endswith(stop)
