"""Tool-call and reasoning parsers for chat completions.

Role of the reference's vllm/entrypoints/openai/tool_parsers/ (hermes
parser: hermes_tool_parser.py) and vllm/reasoning/ (deepseek_r1 parser):
turn raw model text into OpenAI-shaped `tool_calls` entries and split
`<think>…</think>` reasoning from the visible answer, both for complete
outputs and incrementally for SSE streaming.

Design notes (ours, not the reference's): the streaming parsers are
plain pushdown scanners over the decoded-text delta stream — no token
ids needed — and they withhold only the minimal suffix that could be a
tag prefix, so streamed content latency stays one chunk behind at most.
Tool-call bodies are emitted as one delta per completed call (name +
full arguments), which every OpenAI client accepts.
"""

from __future__ import annotations

import json
import uuid
from dataclasses import dataclass
from typing import Any, Optional

TOOL_OPEN = "<tool_call>"
TOOL_CLOSE = "</tool_call>"
THINK_OPEN = "<think>"
THINK_CLOSE = "</think>"


def _call_id() -> str:
    return "call_" + uuid.uuid4().hex[:24]


def _longest_tag_prefix(text: str, tag: str) -> int:
    """Length of the longest suffix of `text` that is a proper prefix of
    `tag` (text we must withhold because the tag may continue next chunk)."""
    for n in range(min(len(tag) - 1, len(text)), 0, -1):
        if text.endswith(tag[:n]):
            return n
    return 0


@dataclass
class ParsedToolCall:
    id: str
    name: str
    arguments: str  # raw JSON string, as OpenAI sends it

    def as_openai(self, index: int) -> dict[str, Any]:
        return {
            "index": index,
            "id": self.id,
            "type": "function",
            "function": {"name": self.name, "arguments": self.arguments},
        }


def parse_hermes_tool_calls(text: str) -> tuple[str, list[ParsedToolCall]]:
    """Split complete model output into (content, tool_calls).

    Hermes format: zero or more `<tool_call>{"name": .., "arguments": ..}
    </tool_call>` blocks interleaved with plain content. Malformed JSON
    inside a block degrades to plain content (never drop model output).
    """
    calls: list[ParsedToolCall] = []
    content_parts: list[str] = []
    pos = 0
    while True:
        start = text.find(TOOL_OPEN, pos)
        if start < 0:
            content_parts.append(text[pos:])
            break
        content_parts.append(text[pos:start])
        end = text.find(TOOL_CLOSE, start + len(TOOL_OPEN))
        if end < 0:
            # Unterminated block: try to parse the remainder as JSON
            # (models often stop at the eos before closing the tag).
            body = text[start + len(TOOL_OPEN):]
            call = _parse_call_body(body)
            if call is not None:
                calls.append(call)
            else:
                content_parts.append(text[start:])
            break
        body = text[start + len(TOOL_OPEN):end]
        call = _parse_call_body(body)
        if call is not None:
            calls.append(call)
        else:
            content_parts.append(text[start:end + len(TOOL_CLOSE)])
        pos = end + len(TOOL_CLOSE)
    content = "".join(content_parts).strip()
    return content, calls


def _parse_call_body(body: str) -> Optional[ParsedToolCall]:
    try:
        obj = json.loads(body.strip())
    except (json.JSONDecodeError, ValueError):
        return None
    if not isinstance(obj, dict) or "name" not in obj:
        return None
    args = obj.get("arguments", obj.get("parameters", {}))
    if not isinstance(args, str):
        args = json.dumps(args)
    return ParsedToolCall(id=_call_id(), name=str(obj["name"]),
                          arguments=args)


def split_reasoning(text: str) -> tuple[Optional[str], str]:
    """Complete-output split: returns (reasoning_content, content).

    DeepSeek-R1 style: reasoning lives in a leading `<think>…</think>`
    block (the opening tag is often part of the generation prompt, so a
    bare leading `</think>` also counts)."""
    stripped = text.lstrip()
    if stripped.startswith(THINK_OPEN):
        rest = stripped[len(THINK_OPEN):]
        end = rest.find(THINK_CLOSE)
        if end < 0:
            return rest.strip(), ""
        return rest[:end].strip(), rest[end + len(THINK_CLOSE):].lstrip()
    end = text.find(THINK_CLOSE)
    if end >= 0:
        return text[:end].strip(), text[end + len(THINK_CLOSE):].lstrip()
    return None, text


@dataclass
class StreamingReasoningParser:
    """Incremental `<think>` splitter. feed(delta) -> (reasoning_delta,
    content_delta); call flush() after the last delta."""

    _buf: str = ""
    _state: str = "start"  # start | think | content

    def feed(self, delta: str) -> tuple[str, str]:
        self._buf += delta
        reasoning, content = "", ""
        while self._buf:
            if self._state == "start":
                s = self._buf.lstrip()
                if not s:
                    return reasoning, content
                if THINK_OPEN.startswith(s[:len(THINK_OPEN)]) and \
                        len(s) < len(THINK_OPEN):
                    return reasoning, content  # could still be the tag
                if s.startswith(THINK_OPEN):
                    self._buf = s[len(THINK_OPEN):]
                    self._state = "think"
                else:
                    self._state = "content"
            elif self._state == "think":
                end = self._buf.find(THINK_CLOSE)
                if end >= 0:
                    reasoning += self._buf[:end]
                    self._buf = self._buf[end + len(THINK_CLOSE):].lstrip()
                    self._state = "content"
                else:
                    hold = _longest_tag_prefix(self._buf, THINK_CLOSE)
                    emit = len(self._buf) - hold
                    reasoning += self._buf[:emit]
                    self._buf = self._buf[emit:]
                    return reasoning, content
            else:  # content
                content += self._buf
                self._buf = ""
        return reasoning, content

    def flush(self) -> tuple[str, str]:
        out = (self._buf, "") if self._state == "think" else ("", self._buf)
        self._buf = ""
        return out


@dataclass
class StreamingToolParser:
    """Incremental hermes scanner. feed(delta) -> (content_delta,
    [openai tool_call deltas]); flush() after the stream ends."""

    _buf: str = ""
    _in_call: bool = False
    _n_calls: int = 0
    _saw_call: bool = False

    @property
    def saw_tool_call(self) -> bool:
        return self._saw_call

    def feed(self, delta: str) -> tuple[str, list[dict[str, Any]]]:
        self._buf += delta
        content, calls = "", []
        while True:
            if self._in_call:
                end = self._buf.find(TOOL_CLOSE)
                if end < 0:
                    return content, calls
                call = _parse_call_body(self._buf[:end])
                if call is not None:
                    calls.append(call.as_openai(self._n_calls))
                    self._n_calls += 1
                    self._saw_call = True
                else:
                    content += TOOL_OPEN + self._buf[:end] + TOOL_CLOSE
                self._buf = self._buf[end + len(TOOL_CLOSE):]
                self._in_call = False
            else:
                start = self._buf.find(TOOL_OPEN)
                if start >= 0:
                    content += self._buf[:start]
                    self._buf = self._buf[start + len(TOOL_OPEN):]
                    self._in_call = True
                    continue
                hold = _longest_tag_prefix(self._buf, TOOL_OPEN)
                emit = len(self._buf) - hold
                content += self._buf[:emit]
                self._buf = self._buf[emit:]
                return content, calls

    def flush(self) -> tuple[str, list[dict[str, Any]]]:
        content, calls = "", []
        if self._in_call:
            call = _parse_call_body(self._buf)
            if call is not None:
                calls.append(call.as_openai(self._n_calls))
                self._n_calls += 1
                self._saw_call = True
            else:
                content = TOOL_OPEN + self._buf
        else:
            content = self._buf
        self._buf = ""
        self._in_call = False
        return content, calls


def render_tools_block(tools: list[dict[str, Any]]) -> str:
    """Fallback tool prompt for tokenizers without a tools-aware chat
    template: hermes-style system preamble listing the function schemas."""
    lines = [
        "You may call functions. The available functions are listed as "
        "JSON schemas inside <tools></tools>:",
        "<tools>",
    ]
    for t in tools:
        fn = t.get("function", t)
        lines.append(json.dumps(fn, separators=(",", ":")))
    lines += [
        "</tools>",
        'To call a function, emit <tool_call>{"name": <name>, '
        '"arguments": <args-json>}</tool_call>.',
    ]
    return "\n".join(lines)


def named_tool_schema(tools: list[dict[str, Any]],
                      name: str) -> Optional[dict[str, Any]]:
    """JSON schema of the named function's parameters (for guided
    decoding when tool_choice pins one function)."""
    for t in tools:
        fn = t.get("function", t)
        if fn.get("name") == name:
            return fn.get("parameters") or {"type": "object"}
    return None


# --------------------------------------------------------------------------
# Mistral format: optional content, then "[TOOL_CALLS]" followed by a JSON
# array of {"name": ..., "arguments": {...}} (reference
# tool_parsers/mistral_tool_parser.py).
MISTRAL_MARKER = "[TOOL_CALLS]"


def parse_mistral_tool_calls(text: str) -> tuple[str, list[ParsedToolCall]]:
    idx = text.find(MISTRAL_MARKER)
    if idx < 0:
        return text.strip(), []
    content = text[:idx].strip()
    body = text[idx + len(MISTRAL_MARKER):].strip()
    try:
        arr = json.loads(body)
    except (json.JSONDecodeError, ValueError):
        return text.strip(), []
    calls = []
    for obj in arr if isinstance(arr, list) else [arr]:
        if isinstance(obj, dict) and "name" in obj:
            args = obj.get("arguments", obj.get("parameters", {}))
            if not isinstance(args, str):
                args = json.dumps(args)
            calls.append(ParsedToolCall(id=_call_id(),
                                        name=str(obj["name"]),
                                        arguments=args))
    return content, calls


# Llama-3 JSON format: the WHOLE assistant message is one JSON object
# {"name": ..., "parameters": {...}} (reference llama_tool_parser.py).
def parse_llama3_json_tool_calls(
        text: str) -> tuple[str, list[ParsedToolCall]]:
    stripped = text.strip()
    if not stripped.startswith("{"):
        return stripped, []
    call = _parse_call_body(stripped)
    if call is None:
        return stripped, []
    return "", [call]


@dataclass
class StreamingBufferedToolParser:
    """Streaming wrapper for formats that cannot be parsed incrementally
    (mistral marker / whole-message JSON): content before the trigger
    streams through; once triggered, the rest buffers and parses at
    flush. `parse` is the complete-output parser; `trigger` returns the
    index where buffering must start, or -1."""

    parse: Any = None
    trigger: Any = None
    _buf: str = ""
    _buffering: bool = False
    _saw_call: bool = False

    @property
    def saw_tool_call(self) -> bool:
        return self._saw_call

    def feed(self, delta: str) -> tuple[str, list[dict[str, Any]]]:
        self._buf += delta
        if self._buffering:
            return "", []
        idx = self.trigger(self._buf)
        if idx >= 0:
            content = self._buf[:idx]
            self._buf = self._buf[idx:]
            self._buffering = True
            return content, []
        # Withhold a potential trigger prefix at the tail.
        hold = self._hold_len(self._buf)
        emit = len(self._buf) - hold
        out = self._buf[:emit]
        self._buf = self._buf[emit:]
        return out, []

    def _hold_len(self, text: str) -> int:
        return _longest_tag_prefix(text, MISTRAL_MARKER)

    def flush(self) -> tuple[str, list[dict[str, Any]]]:
        content, calls = self.parse(self._buf)
        self._buf = ""
        out = [c.as_openai(i) for i, c in enumerate(calls)]
        if out:
            self._saw_call = True
        return content, out




# Pythonic format (Llama-4 / reference pythonic_tool_parser.py): the
# message is a Python-literal list of calls: [get_weather(city="SF"),
# f(x=3)]. Parsed with ast — no eval.
def parse_pythonic_tool_calls(text: str) -> tuple[str, list[ParsedToolCall]]:
    import ast

    stripped = text.strip()
    if not (stripped.startswith("[") and stripped.endswith("]")):
        return stripped, []
    try:
        tree = ast.parse(stripped, mode="eval")
    except SyntaxError:
        return stripped, []
    if not isinstance(tree.body, ast.List):
        return stripped, []
    calls = []
    for node in tree.body.elts:
        if not (isinstance(node, ast.Call)
                and isinstance(node.func, ast.Name)):
            return stripped, []
        args = {}
        for kw in node.keywords:
            try:
                args[kw.arg] = ast.literal_eval(kw.value)
            except (ValueError, SyntaxError):
                return stripped, []
        calls.append(ParsedToolCall(id=_call_id(), name=node.func.id,
                                    arguments=json.dumps(args)))
    return "", calls


# Granite-3 format (reference granite_tool_parser.py): the message
# starts with "<|tool_call|>" followed by a JSON array of
# {"name": ..., "arguments": {...}}.
GRANITE_MARKER = "<|tool_call|>"


def parse_granite_tool_calls(text: str) -> tuple[str, list[ParsedToolCall]]:
    idx = text.find(GRANITE_MARKER)
    if idx < 0:
        return text.strip(), []
    content = text[:idx].strip()
    body = text[idx + len(GRANITE_MARKER):].strip()
    try:
        arr = json.loads(body)
    except (json.JSONDecodeError, ValueError):
        return text.strip(), []
    calls = []
    for obj in arr if isinstance(arr, list) else [arr]:
        if isinstance(obj, dict) and "name" in obj:
            args = obj.get("arguments", obj.get("parameters", {}))
            if not isinstance(args, str):
                args = json.dumps(args)
            calls.append(ParsedToolCall(id=_call_id(),
                                        name=str(obj["name"]),
                                        arguments=args))
    return content, calls


# InternLM2 format (reference internlm2_tool_parser.py):
# content<|action_start|><|plugin|>{json}<|action_end|>
IL2_START = "<|action_start|><|plugin|>"
IL2_END = "<|action_end|>"


def parse_internlm2_tool_calls(
        text: str) -> tuple[str, list[ParsedToolCall]]:
    idx = text.find(IL2_START)
    if idx < 0:
        return text.strip(), []
    content = text[:idx].strip()
    body = text[idx + len(IL2_START):]
    end = body.find(IL2_END)
    if end >= 0:
        body = body[:end]
    call = _parse_call_body(body.strip())
    if call is None:
        return text.strip(), []
    return content, [call]


def make_streaming_tool_parser(fmt: str):
    if fmt == "mistral":
        return StreamingBufferedToolParser(
            parse=parse_mistral_tool_calls,
            trigger=lambda s: s.find(MISTRAL_MARKER))
    if fmt == "llama3_json":
        def trig(s):
            st = s.lstrip()
            return (len(s) - len(st)) if st.startswith("{") else -1

        p = StreamingBufferedToolParser(
            parse=parse_llama3_json_tool_calls, trigger=trig)
        p._hold_len = lambda text: 0
        return p
    if fmt == "pythonic":
        def trig(s):
            st = s.lstrip()
            return (len(s) - len(st)) if st.startswith("[") else -1

        p = StreamingBufferedToolParser(
            parse=parse_pythonic_tool_calls, trigger=trig)
        p._hold_len = lambda text: 0
        return p
    if fmt == "granite":
        p = StreamingBufferedToolParser(
            parse=parse_granite_tool_calls,
            trigger=lambda s: s.find(GRANITE_MARKER))
        p._hold_len = lambda text: _longest_tag_prefix(text, GRANITE_MARKER)
        return p
    if fmt == "internlm2":
        p = StreamingBufferedToolParser(
            parse=parse_internlm2_tool_calls,
            trigger=lambda s: s.find(IL2_START))
        p._hold_len = lambda text: _longest_tag_prefix(text, IL2_START)
        return p
    return StreamingToolParser()


def parse_tool_calls(fmt: str, text: str):
    if fmt == "mistral":
        return parse_mistral_tool_calls(text)
    if fmt == "llama3_json":
        return parse_llama3_json_tool_calls(text)
    if fmt == "pythonic":
        return parse_pythonic_tool_calls(text)
    if fmt == "granite":
        return parse_granite_tool_calls(text)
    if fmt == "internlm2":
        return parse_internlm2_tool_calls(text)
    # "hermes" and aliases that use the same <tool_call> JSON format
    # (qwen, qwen3, glm4 — reference hermes_tool_parser.py family).
    return parse_hermes_tool_calls(text)
