"""Scheduler -> model-runner contract.

Mirrors the *shape* of the reference's SchedulerOutput
(vllm/v1/core/sched/output.py:193): new requests carry full state, cached
(already-running) requests carry columnar diffs only.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Optional

from vllm_amd.sampling_params import SamplingParams


@dataclass
class NewRequestData:
    req_id: str
    prompt_token_ids: list[int]
    block_ids: list[int]
    num_computed_tokens: int
    sampling_params: SamplingParams
    grammar: object = None  # CompiledGrammar for structured output
    lora_id: int = 0
    mm_data: object = None  # {"image": pixels} for vision models
    # Window-group block ids (mixed sliding+global models); None for
    # single-group models.
    block_ids_w: object = None


@dataclass(slots=True)
class CachedRequestData:
    """Columnar diffs for requests the runner has already seen."""

    req_ids: list[str] = field(default_factory=list)
    # True if the request resumed from preemption: new_block_ids then holds
    # the FULL block list, and new_token_ids the full output-token history.
    resumed: list[bool] = field(default_factory=list)
    new_block_ids: list[list[int]] = field(default_factory=list)
    num_computed_tokens: list[int] = field(default_factory=list)
    new_token_ids: list[list[int]] = field(default_factory=list)
    # Parallel to new_block_ids for the window group (mixed models);
    # entries are None for single-group models.
    new_block_ids_w: list = field(default_factory=list)

    @property
    def num_reqs(self) -> int:
        return len(self.req_ids)


@dataclass
class SchedulerOutput:
    scheduled_new_reqs: list[NewRequestData]
    scheduled_cached_reqs: CachedRequestData
    # req_id -> number of tokens to run through the model this step.
    num_scheduled_tokens: dict[str, int]
    total_num_scheduled_tokens: int
    finished_req_ids: set[str]
    # req_id -> draft tokens scheduled for verification this step
    # (spec decode; role of scheduled_spec_decode_tokens in the reference).
    scheduled_spec_decode_tokens: dict[str, list[int]] = field(
        default_factory=dict)
    # Ordered KV offload copies for the runner, executed BEFORE the
    # forward: ("out", gpu_block_id, host_slot) = D2H save of an evicted
    # prefix block; ("in", gpu_block_id, host_slot) = H2D restore of a
    # host-tier prefix hit.
    kv_swap_ops: list[tuple[str, int, int]] = field(default_factory=list)

    @property
    def num_reqs(self) -> int:
        return len(self.num_scheduled_tokens)


@dataclass
class ModelRunnerOutput:
    """Runner -> scheduler result (role of vllm/v1/outputs.py:261)."""

    req_ids: list[str]
    # One (possibly empty) list per req: empty when the request's prompt is
    # not fully computed yet (mid chunked-prefill) so no token was sampled.
    sampled_token_ids: list[list[int]]
    # Optional per-request logprobs of sampled tokens:
    # req_id -> list of {token_id: logprob} dicts, one per sampled token.
    logprobs: Optional[dict[str, list[dict[int, float]]]] = None
    # Model-based draft proposals (medusa heads): req_id -> draft tokens
    # for the NEXT step, conditioned on the last accepted position.
    draft_token_ids: Optional[dict[str, list[int]]] = None
    # Pooling requests that completed prefill this step: req_id -> pooled
    # hidden vector (embedding models).
    pooled: Optional[dict[str, list[float]]] = None
    # Prompt logprobs computed this step (chunked-prefill order):
    # req_id -> one {token_id: logprob} dict per prompt position covered.
    prompt_logprobs: Optional[dict[str, list[dict[int, float]]]] = None


EMPTY_MODEL_RUNNER_OUTPUT = ModelRunnerOutput(req_ids=[], sampled_token_ids=[])


@dataclass(slots=True)
class EngineCoreOutput:
    req_id: str
    new_token_ids: list[int]
    finish_reason: Optional[str] = None
    stop_reason: Optional[object] = None
    num_cached_tokens: int = 0
    new_logprobs: Optional[list[dict[int, float]]] = None
    new_prompt_logprobs: Optional[list[dict[int, float]]] = None
    pooled: Optional[list[float]] = None

    @property
    def finished(self) -> bool:
        return self.finish_reason is not None


@dataclass
class SchedulerStats:
    num_running_reqs: int = 0
    num_waiting_reqs: int = 0
    kv_cache_usage: float = 0.0
    prefix_cache_queries: int = 0
    prefix_cache_hits: int = 0
