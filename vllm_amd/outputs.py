"""User-facing request outputs (role of vllm/outputs.py in the reference)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class CompletionOutput:
    index: int
    text: str
    token_ids: list[int]
    cumulative_logprob: Optional[float] = None
    logprobs: Optional[list[dict[int, float]]] = None
    finish_reason: Optional[str] = None  # "stop" | "length" | "abort"
    stop_reason: Optional[object] = None

    @property
    def finished(self) -> bool:
        return self.finish_reason is not None


@dataclass
class RequestOutput:
    request_id: str
    prompt: Optional[str]
    prompt_token_ids: list[int]
    outputs: list[CompletionOutput]
    finished: bool
    metrics: Optional[dict] = None
    # Embedding/pooling requests: the pooled hidden-state vector.
    pooled: Optional[list[float]] = None
    # SamplingParams.prompt_logprobs: one {token: logprob} dict per
    # prompt token starting at index 1 (token 0 is unconditioned).
    prompt_logprobs: Optional[list[dict[int, float]]] = None
    # Prompt tokens served from the prefix cache (usage reporting).
    num_cached_tokens: int = 0

    def __repr__(self) -> str:
        return (
            f"RequestOutput(request_id={self.request_id!r}, "
            f"finished={self.finished}, "
            f"outputs={self.outputs!r})"
        )
