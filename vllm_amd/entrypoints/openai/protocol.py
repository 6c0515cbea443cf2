"""OpenAI-compatible request/response schemas (role of the reference's
vllm/entrypoints/openai/protocol.py, trimmed to the supported surface)."""

from __future__ import annotations

import time
import uuid
from typing import Any, Literal, Optional, Union

from pydantic import BaseModel, Field

from vllm_amd.sampling_params import RequestOutputKind, SamplingParams


def random_id(prefix: str) -> str:
    return f"{prefix}-{uuid.uuid4().hex[:24]}"


class CompletionRequest(BaseModel):
    model: str
    prompt: Union[str, list[str], list[int], list[list[int]]]
    best_of: Optional[int] = None
    echo: bool = False
    frequency_penalty: float = 0.0
    logit_bias: Optional[dict[str, float]] = None
    logprobs: Optional[int] = None
    max_tokens: Optional[int] = 16
    n: int = 1
    presence_penalty: float = 0.0
    seed: Optional[int] = None
    stop: Optional[Union[str, list[str]]] = None
    stream: bool = False
    stream_options: Optional[dict[str, Any]] = None
    suffix: Optional[str] = None
    temperature: float = 1.0
    top_p: float = 1.0
    user: Optional[str] = None
    # Extensions (same names as the reference).
    top_k: int = 0
    min_p: float = 0.0
    repetition_penalty: float = 1.0
    min_tokens: int = 0
    stop_token_ids: Optional[list[int]] = None
    ignore_eos: bool = False
    skip_special_tokens: bool = True
    guided_choice: Optional[list[str]] = None
    guided_regex: Optional[str] = None
    guided_json: Optional[Union[dict, str]] = None
    guided_grammar: Optional[str] = None
    bad_words: Optional[list[str]] = None
    allowed_token_ids: Optional[list[int]] = None
    truncate_prompt_tokens: Optional[int] = None
    include_stop_str_in_output: bool = False
    spaces_between_special_tokens: bool = True
    priority: int = 0  # lower = sooner (priority policy)
    # Extension (same name as the reference): per-prompt-token logprobs.
    prompt_logprobs: Optional[int] = None

    def to_sampling_params(self, stream: bool) -> SamplingParams:
        logit_bias = (
            {int(k): v for k, v in self.logit_bias.items()}
            if self.logit_bias else None
        )
        return SamplingParams(
            n=1,
            presence_penalty=self.presence_penalty,
            frequency_penalty=self.frequency_penalty,
            repetition_penalty=self.repetition_penalty,
            temperature=self.temperature,
            top_p=self.top_p,
            top_k=self.top_k,
            min_p=self.min_p,
            seed=self.seed,
            stop=self.stop,
            stop_token_ids=self.stop_token_ids,
            ignore_eos=self.ignore_eos,
            max_tokens=self.max_tokens,
            min_tokens=self.min_tokens,
            logprobs=self.logprobs,
            prompt_logprobs=self.prompt_logprobs,
            logit_bias=logit_bias,
            skip_special_tokens=self.skip_special_tokens,
            guided_choice=self.guided_choice,
            guided_regex=self.guided_regex,
            guided_json=self.guided_json,
            guided_grammar=self.guided_grammar,
            bad_words=self.bad_words,
            allowed_token_ids=self.allowed_token_ids,
            truncate_prompt_tokens=self.truncate_prompt_tokens,
            include_stop_str_in_output=self.include_stop_str_in_output,
            spaces_between_special_tokens=self.spaces_between_special_tokens,
            priority=self.priority,
            output_kind=(RequestOutputKind.DELTA if stream
                         else RequestOutputKind.FINAL_ONLY),
        )


class ChatMessage(BaseModel):
    role: str
    content: Optional[Union[str, list[dict[str, Any]]]] = None
    name: Optional[str] = None
    tool_calls: Optional[list[dict[str, Any]]] = None
    tool_call_id: Optional[str] = None

    def text(self, image_sentinel: Optional[str] = None) -> str:
        if isinstance(self.content, str):
            return self.content
        if self.content is None:
            return ""
        parts = []
        for seg in self.content:
            if seg.get("type") == "text":
                parts.append(seg.get("text", ""))
            elif seg.get("type") == "image_url" and image_sentinel:
                # Placeholder spliced back into token ids by the server
                # (reference: multimodal chat content parts become
                # per-model placeholder tokens via the mm processor).
                parts.append(image_sentinel)
        return "".join(parts)

    def audio_inputs(self) -> list[dict]:
        """input_audio content parts ({"data": b64, "format": ...})."""
        if not isinstance(self.content, list):
            return []
        return [seg.get("input_audio") or {} for seg in self.content
                if seg.get("type") == "input_audio"]

    def image_urls(self) -> list[str]:
        """URLs of image_url content parts, in order of appearance."""
        if not isinstance(self.content, list):
            return []
        out = []
        for seg in self.content:
            if seg.get("type") == "image_url":
                u = seg.get("image_url")
                out.append(u.get("url", "") if isinstance(u, dict) else
                           str(u or ""))
        return out


class ChatCompletionRequest(BaseModel):
    model: str
    messages: list[ChatMessage]
    frequency_penalty: float = 0.0
    logit_bias: Optional[dict[str, float]] = None
    logprobs: bool = False
    top_logprobs: Optional[int] = None
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    n: int = 1
    presence_penalty: float = 0.0
    seed: Optional[int] = None
    stop: Optional[Union[str, list[str]]] = None
    stream: bool = False
    stream_options: Optional[dict[str, Any]] = None
    temperature: float = 1.0
    top_p: float = 1.0
    user: Optional[str] = None
    top_k: int = 0
    min_p: float = 0.0
    repetition_penalty: float = 1.0
    min_tokens: int = 0
    stop_token_ids: Optional[list[int]] = None
    ignore_eos: bool = False
    skip_special_tokens: bool = True
    guided_choice: Optional[list[str]] = None
    guided_regex: Optional[str] = None
    guided_json: Optional[Union[dict, str]] = None
    guided_grammar: Optional[str] = None
    bad_words: Optional[list[str]] = None
    allowed_token_ids: Optional[list[int]] = None
    truncate_prompt_tokens: Optional[int] = None
    include_stop_str_in_output: bool = False
    spaces_between_special_tokens: bool = True
    priority: int = 0  # lower = sooner (priority policy)
    response_format: Optional[dict[str, Any]] = None
    add_generation_prompt: bool = True
    # Tool calling (OpenAI function-calling surface).
    tools: Optional[list[dict[str, Any]]] = None
    tool_choice: Optional[Union[str, dict[str, Any]]] = None

    def named_tool(self) -> Optional[str]:
        """Function name when tool_choice pins a single function."""
        if isinstance(self.tool_choice, dict):
            fn = self.tool_choice.get("function") or {}
            return fn.get("name")
        return None

    def to_sampling_params(self, stream: bool,
                           default_max_tokens: int) -> SamplingParams:
        max_tokens = (self.max_completion_tokens or self.max_tokens
                      or default_max_tokens)
        logit_bias = (
            {int(k): v for k, v in self.logit_bias.items()}
            if self.logit_bias else None
        )
        n_logprobs = (self.top_logprobs or 1) if self.logprobs else None
        # response_format: json_object -> any-JSON grammar; json_schema ->
        # schema grammar (OpenAI structured outputs shape).
        guided_json = self.guided_json
        json_object = False
        # tool_choice naming one function: guide decoding with that
        # function's parameter schema; the server wraps the raw JSON
        # output as the tool call's arguments.
        named = self.named_tool()
        if named and self.tools:
            from vllm_amd.entrypoints.tool_parser import named_tool_schema
            schema = named_tool_schema(self.tools, named)
            if schema is not None:
                guided_json = schema
        if self.response_format:
            kind = self.response_format.get("type")
            if kind == "json_object":
                json_object = True
            elif kind == "json_schema":
                js = self.response_format.get("json_schema") or {}
                guided_json = js.get("schema") or js
        return SamplingParams(
            n=1,
            presence_penalty=self.presence_penalty,
            frequency_penalty=self.frequency_penalty,
            repetition_penalty=self.repetition_penalty,
            temperature=self.temperature,
            top_p=self.top_p,
            top_k=self.top_k,
            min_p=self.min_p,
            seed=self.seed,
            stop=self.stop,
            stop_token_ids=self.stop_token_ids,
            ignore_eos=self.ignore_eos,
            max_tokens=max_tokens,
            min_tokens=self.min_tokens,
            logprobs=n_logprobs,
            logit_bias=logit_bias,
            skip_special_tokens=self.skip_special_tokens,
            guided_choice=self.guided_choice,
            guided_regex=self.guided_regex,
            guided_json=guided_json,
            guided_grammar=self.guided_grammar,
            bad_words=self.bad_words,
            allowed_token_ids=self.allowed_token_ids,
            truncate_prompt_tokens=self.truncate_prompt_tokens,
            include_stop_str_in_output=self.include_stop_str_in_output,
            spaces_between_special_tokens=self.spaces_between_special_tokens,
            priority=self.priority,
            guided_json_object=json_object,
            output_kind=(RequestOutputKind.DELTA if stream
                         else RequestOutputKind.FINAL_ONLY),
        )


class UsageInfo(BaseModel):
    prompt_tokens: int = 0
    completion_tokens: int = 0
    total_tokens: int = 0
    # OpenAI prompt_tokens_details parity: prefix-cache hits.
    prompt_tokens_details: Optional[dict[str, int]] = None


class CompletionChoice(BaseModel):
    index: int
    text: str
    logprobs: Optional[dict] = None
    prompt_logprobs: Optional[list[dict[int, float]]] = None
    finish_reason: Optional[str] = None
    stop_reason: Optional[Union[int, str]] = None


class CompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("cmpl"))
    object: Literal["text_completion"] = "text_completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str
    choices: list[CompletionChoice]
    usage: UsageInfo = Field(default_factory=UsageInfo)


class ChatCompletionMessage(BaseModel):
    role: str = "assistant"
    content: Optional[str] = None
    reasoning_content: Optional[str] = None
    tool_calls: Optional[list[dict[str, Any]]] = None


class ChatChoice(BaseModel):
    index: int
    message: ChatCompletionMessage
    logprobs: Optional[dict] = None
    finish_reason: Optional[str] = None


class ChatCompletionResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("chatcmpl"))
    object: Literal["chat.completion"] = "chat.completion"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str
    choices: list[ChatChoice]
    usage: UsageInfo = Field(default_factory=UsageInfo)


class DeltaMessage(BaseModel):
    role: Optional[str] = None
    content: Optional[str] = None
    reasoning_content: Optional[str] = None
    tool_calls: Optional[list[dict[str, Any]]] = None


class ChatStreamChoice(BaseModel):
    index: int
    delta: DeltaMessage
    finish_reason: Optional[str] = None
    logprobs: Optional[dict[str, Any]] = None


class ChatCompletionStreamResponse(BaseModel):
    id: str
    object: Literal["chat.completion.chunk"] = "chat.completion.chunk"
    created: int = Field(default_factory=lambda: int(time.time()))
    model: str
    choices: list[ChatStreamChoice]
    usage: Optional[UsageInfo] = None


class EmbeddingRequest(BaseModel):
    model: str
    input: Union[str, list[str], list[int], list[list[int]]]
    encoding_format: Literal["float", "base64"] = "float"
    # Matryoshka-style truncation: keep the first N dimensions and
    # re-normalize (OpenAI `dimensions` semantics).
    dimensions: Optional[int] = None
    user: Optional[str] = None
    # Extension: pooling strategy (reference pools per model config).
    pooling: Literal["last", "mean"] = "last"


class EmbeddingData(BaseModel):
    object: Literal["embedding"] = "embedding"
    index: int
    embedding: Union[list[float], str]


class EmbeddingResponse(BaseModel):
    object: Literal["list"] = "list"
    data: list[EmbeddingData]
    model: str
    usage: UsageInfo = Field(default_factory=UsageInfo)


class ResponsesRequest(BaseModel):
    """OpenAI Responses API (role of the reference's /v1/responses)."""

    model: str
    input: Union[str, list[dict[str, Any]]]
    instructions: Optional[str] = None
    max_output_tokens: Optional[int] = None
    temperature: float = 1.0
    top_p: float = 1.0
    stream: bool = False
    metadata: Optional[dict[str, Any]] = None
    # Stored-response surface (reference responses API): store lets
    # GET /v1/responses/{id} retrieve the result; background returns
    # immediately and generates server-side; previous_response_id
    # chains onto a stored response's output.
    store: bool = True
    background: bool = False
    previous_response_id: Optional[str] = None


class ScoreRequest(BaseModel):
    model: str
    text_1: Union[str, list[str]]
    text_2: Union[str, list[str]]
    pooling: Literal["last", "mean"] = "mean"


class ScoreData(BaseModel):
    object: Literal["score"] = "score"
    index: int
    score: float


class ScoreResponse(BaseModel):
    object: Literal["list"] = "list"
    data: list[ScoreData]
    model: str
    usage: UsageInfo = Field(default_factory=UsageInfo)


class RerankRequest(BaseModel):
    model: str
    query: str
    documents: list[str]
    top_n: Optional[int] = None
    pooling: Literal["last", "mean"] = "mean"


class RerankResult(BaseModel):
    index: int
    document: dict
    relevance_score: float


class RerankResponse(BaseModel):
    id: str = Field(default_factory=lambda: random_id("rerank"))
    model: str
    results: list[RerankResult]
    usage: UsageInfo = Field(default_factory=UsageInfo)


class ModelCard(BaseModel):
    id: str
    object: Literal["model"] = "model"
    created: int = Field(default_factory=lambda: int(time.time()))
    owned_by: str = "vllm_amd"
    max_model_len: Optional[int] = None


class ModelList(BaseModel):
    object: Literal["list"] = "list"
    data: list[ModelCard] = Field(default_factory=list)


class TokenizeRequest(BaseModel):
    model: Optional[str] = None
    prompt: str
    add_special_tokens: bool = True


class TokenizeResponse(BaseModel):
    tokens: list[int]
    count: int
    max_model_len: int


class DetokenizeRequest(BaseModel):
    model: Optional[str] = None
    tokens: list[int]


class DetokenizeResponse(BaseModel):
    prompt: str


class ErrorResponse(BaseModel):
    object: Literal["error"] = "error"
    message: str
    type: str = "invalid_request_error"
    code: int = 400
