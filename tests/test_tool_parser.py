"""Tool-call + reasoning parser tests (reference surface:
vllm/entrypoints/openai/tool_parsers hermes parser and vllm/reasoning
deepseek_r1 parser; test pattern: tests/tool_use, tests/reasoning)."""

import json

import pytest
from fastapi.testclient import TestClient

from vllm_amd.entrypoints.tool_parser import (
    ParsedToolCall,
    StreamingReasoningParser,
    StreamingToolParser,
    named_tool_schema,
    parse_hermes_tool_calls,
    render_tools_block,
    split_reasoning,
)

WEATHER_TOOL = {
    "type": "function",
    "function": {
        "name": "get_weather",
        "description": "Get the weather for a city",
        "parameters": {
            "type": "object",
            "properties": {"city": {"enum": ["Paris", "Oslo"]}},
            "required": ["city"],
        },
    },
}


def test_parse_single_tool_call():
    text = ('Sure, let me check.\n<tool_call>{"name": "get_weather", '
            '"arguments": {"city": "Paris"}}</tool_call>')
    content, calls = parse_hermes_tool_calls(text)
    assert content == "Sure, let me check."
    assert len(calls) == 1
    assert calls[0].name == "get_weather"
    assert json.loads(calls[0].arguments) == {"city": "Paris"}
    assert calls[0].id.startswith("call_")


def test_parse_parallel_tool_calls():
    text = ('<tool_call>{"name": "a", "arguments": {}}</tool_call>'
            '<tool_call>{"name": "b", "arguments": {"x": 1}}</tool_call>')
    content, calls = parse_hermes_tool_calls(text)
    assert content == ""
    assert [c.name for c in calls] == ["a", "b"]


def test_parse_unterminated_block():
    text = '<tool_call>{"name": "a", "arguments": {"x": 2}}'
    content, calls = parse_hermes_tool_calls(text)
    assert len(calls) == 1 and json.loads(calls[0].arguments) == {"x": 2}


def test_parse_malformed_degrades_to_content():
    text = "<tool_call>not json</tool_call> after"
    content, calls = parse_hermes_tool_calls(text)
    assert calls == []
    assert "not json" in content and "after" in content


def test_split_reasoning():
    r, c = split_reasoning("<think>step by step</think>the answer")
    assert r == "step by step" and c == "the answer"
    r, c = split_reasoning("no tags here")
    assert r is None and c == "no tags here"
    # bare closing tag (opening tag in the prompt)
    r, c = split_reasoning("thinking...</think>done")
    assert r == "thinking..." and c == "done"
    # unterminated think
    r, c = split_reasoning("<think>never stops")
    assert r == "never stops" and c == ""


@pytest.mark.parametrize("chunk", [1, 3, 7, 100])
def test_streaming_reasoning_parser_chunked(chunk):
    text = "<think>abc def</think>result here"
    p = StreamingReasoningParser()
    reasoning, content = "", ""
    for i in range(0, len(text), chunk):
        r, c = p.feed(text[i:i + chunk])
        reasoning += r
        content += c
    r, c = p.flush()
    reasoning += r
    content += c
    assert reasoning == "abc def"
    assert content == "result here"


@pytest.mark.parametrize("chunk", [1, 4, 9, 1000])
def test_streaming_tool_parser_chunked(chunk):
    text = ('before <tool_call>{"name": "get_weather", "arguments": '
            '{"city": "Oslo"}}</tool_call> after')
    p = StreamingToolParser()
    content, calls = "", []
    for i in range(0, len(text), chunk):
        c, k = p.feed(text[i:i + chunk])
        content += c
        calls += k
    c, k = p.flush()
    content += c
    calls += k
    assert content == "before  after"
    assert len(calls) == 1
    assert calls[0]["function"]["name"] == "get_weather"
    assert json.loads(calls[0]["function"]["arguments"]) == {"city": "Oslo"}
    assert p.saw_tool_call


def test_streaming_tool_parser_plain_text_passthrough():
    p = StreamingToolParser()
    c1, k1 = p.feed("hello < world")
    c2, k2 = p.flush()
    assert c1 + c2 == "hello < world"
    assert k1 == k2 == []
    assert not p.saw_tool_call


def test_named_tool_schema_and_render():
    s = named_tool_schema([WEATHER_TOOL], "get_weather")
    assert s["required"] == ["city"]
    assert s["properties"]["city"]["enum"] == ["Paris", "Oslo"]
    assert named_tool_schema([WEATHER_TOOL], "nope") is None
    block = render_tools_block([WEATHER_TOOL])
    assert "get_weather" in block and "<tools>" in block


def test_parsed_tool_call_openai_shape():
    c = ParsedToolCall(id="call_x", name="f", arguments="{}")
    d = c.as_openai(2)
    assert d == {"index": 2, "id": "call_x", "type": "function",
                 "function": {"name": "f", "arguments": "{}"}}


# ---------------------------------------------------------------- API level

@pytest.fixture(scope="module")
def client():
    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    args = EngineArgs(
        model="tiny-llama", dtype="fp32", device="cpu", block_size=16,
        num_gpu_blocks=256, max_model_len=512,
        max_num_batched_tokens=512, max_num_seqs=8,
    )
    app, state = make_server(args, served_model_name="tiny-llama",
                             reasoning_parser="deepseek_r1")
    with TestClient(app) as c:
        yield c
    state.engine.shutdown()


def test_chat_named_tool_choice_guided(client):
    """tool_choice naming a function guides decoding with its parameter
    schema; the response must be a tool_calls message whose arguments
    validate against that schema."""
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "weather in Paris?"}],
        "tools": [WEATHER_TOOL],
        "tool_choice": {"type": "function",
                        "function": {"name": "get_weather"}},
        "max_tokens": 48,
        "temperature": 0.0,
    })
    assert r.status_code == 200, r.text
    msg = r.json()["choices"][0]["message"]
    assert r.json()["choices"][0]["finish_reason"] == "tool_calls"
    assert msg["content"] is None
    (call,) = msg["tool_calls"]
    assert call["function"]["name"] == "get_weather"
    args = json.loads(call["function"]["arguments"])
    assert args["city"] in ("Paris", "Oslo")


def test_chat_tools_auto_no_call(client):
    """tools present + auto choice: a model that emits no <tool_call>
    returns plain content and a normal finish_reason."""
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "tools": [WEATHER_TOOL],
        "max_tokens": 8,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text
    choice = r.json()["choices"][0]
    assert choice["message"]["tool_calls"] is None
    assert choice["finish_reason"] == "length"


def test_chat_stream_with_tools_enabled(client):
    """Streaming with tools enabled still delivers content deltas and a
    terminal finish_reason."""
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [{"role": "user", "content": "hi"}],
        "tools": [WEATHER_TOOL],
        "stream": True,
        "max_tokens": 6,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200
    finishes = []
    for line in r.text.splitlines():
        if line.startswith("data: ") and line != "data: [DONE]":
            chunk = json.loads(line[6:])
            finishes.append(chunk["choices"][0].get("finish_reason"))
    assert "length" in finishes


def test_chat_tool_role_message_accepted(client):
    """Tool-result turns (role=tool with tool_call_id) round-trip through
    the chat template."""
    r = client.post("/v1/chat/completions", json={
        "model": "tiny-llama",
        "messages": [
            {"role": "user", "content": "weather?"},
            {"role": "assistant", "tool_calls": [
                {"id": "call_1", "type": "function",
                 "function": {"name": "get_weather",
                              "arguments": "{\"city\": \"Paris\"}"}}]},
            {"role": "tool", "tool_call_id": "call_1", "content": "sunny"},
        ],
        "max_tokens": 4,
        "temperature": 0.0,
        "ignore_eos": True,
    })
    assert r.status_code == 200, r.text


def test_parse_mistral_format():
    from vllm_amd.entrypoints.tool_parser import parse_mistral_tool_calls

    text = ('Let me check. [TOOL_CALLS][{"name": "get_weather", '
            '"arguments": {"city": "Paris"}}, {"name": "b", '
            '"arguments": {}}]')
    content, calls = parse_mistral_tool_calls(text)
    assert content == "Let me check."
    assert [c.name for c in calls] == ["get_weather", "b"]
    assert json.loads(calls[0].arguments) == {"city": "Paris"}
    # no marker -> plain content
    c2, k2 = parse_mistral_tool_calls("no tools here")
    assert c2 == "no tools here" and k2 == []


def test_parse_llama3_json_format():
    from vllm_amd.entrypoints.tool_parser import (
        parse_llama3_json_tool_calls)

    content, calls = parse_llama3_json_tool_calls(
        '{"name": "f", "parameters": {"x": 1}}')
    assert content == ""
    assert calls[0].name == "f"
    assert json.loads(calls[0].arguments) == {"x": 1}
    c2, k2 = parse_llama3_json_tool_calls("plain answer")
    assert c2 == "plain answer" and k2 == []


@pytest.mark.parametrize("chunk", [1, 7, 1000])
def test_streaming_mistral_parser(chunk):
    from vllm_amd.entrypoints.tool_parser import (
        make_streaming_tool_parser)

    text = ('before [TOOL_CALLS][{"name": "a", "arguments": {"k": 2}}]')
    p = make_streaming_tool_parser("mistral")
    content, calls = "", []
    for i in range(0, len(text), chunk):
        c, k = p.feed(text[i:i + chunk])
        content += c
        calls += k
    c, k = p.flush()
    content += c
    calls += k
    assert content.strip() == "before"
    assert len(calls) == 1 and calls[0]["function"]["name"] == "a"
    assert p.saw_tool_call


def test_streaming_llama3_json_parser():
    from vllm_amd.entrypoints.tool_parser import (
        make_streaming_tool_parser)

    p = make_streaming_tool_parser("llama3_json")
    c1, k1 = p.feed('{"name": "f", "param')
    c2, k2 = p.feed('eters": {}}')
    c3, k3 = p.flush()
    assert c1 == c2 == "" and not k1 and not k2
    assert len(k3) == 1 and k3[0]["function"]["name"] == "f"
    # non-JSON output streams through
    p2 = make_streaming_tool_parser("llama3_json")
    c, _ = p2.feed("hello world")
    c2, k = p2.flush()
    assert c + c2 == "hello world" and not k


def test_pythonic_parser_complete():
    from vllm_amd.entrypoints.tool_parser import parse_tool_calls

    content, calls = parse_tool_calls(
        "pythonic", '[get_weather(city="SF", unit="c"), t2(x=3, ok=True)]')
    assert content == ""
    assert [c.name for c in calls] == ["get_weather", "t2"]
    import json
    assert json.loads(calls[0].arguments) == {"city": "SF", "unit": "c"}
    assert json.loads(calls[1].arguments) == {"x": 3, "ok": True}
    # Non-call text passes through.
    content, calls = parse_tool_calls("pythonic", "just text [1, 2]")
    assert calls == [] and "just text" in content


def test_granite_parser_complete_and_streaming():
    import json

    from vllm_amd.entrypoints.tool_parser import (
        make_streaming_tool_parser, parse_tool_calls)

    msg = ('I will call a tool. <|tool_call|>'
           '[{"name": "lookup", "arguments": {"q": "x"}}]')
    content, calls = parse_tool_calls("granite", msg)
    assert content == "I will call a tool."
    assert calls[0].name == "lookup"
    assert json.loads(calls[0].arguments) == {"q": "x"}

    sp = make_streaming_tool_parser("granite")
    streamed = ""
    for i in range(0, len(msg), 7):
        c, _ = sp.feed(msg[i:i + 7])
        streamed += c
    c, out = sp.flush()
    streamed += c
    assert streamed.strip() == "I will call a tool."
    assert out[0]["function"]["name"] == "lookup"


def test_internlm2_parser():
    import json

    from vllm_amd.entrypoints.tool_parser import parse_tool_calls

    msg = ('Let me check.<|action_start|><|plugin|>'
           '{"name": "calc", "parameters": {"a": 1}}<|action_end|>')
    content, calls = parse_tool_calls("internlm2", msg)
    assert content == "Let me check."
    assert calls[0].name == "calc"
    assert json.loads(calls[0].arguments) == {"a": 1}
