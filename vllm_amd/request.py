"""Scheduler-side request state (role of vllm/v1/request.py:59).

A ``Request`` is the single source of truth for one sequence's lifecycle:
prompt/output tokens, how many tokens have been computed into the KV
cache, and status transitions WAITING -> RUNNING -> (PREEMPTED ->) ...
-> FINISHED_*.
"""

from __future__ import annotations

import enum
import time
from typing import TYPE_CHECKING, Optional

from vllm_amd.sampling_params import SamplingParams

if TYPE_CHECKING:
    from vllm_amd.core.kv_cache_utils import BlockHash


class RequestStatus(enum.IntEnum):
    WAITING = 0
    RUNNING = 1
    PREEMPTED = 2
    FINISHED_STOPPED = 3
    FINISHED_LENGTH_CAPPED = 4
    FINISHED_ABORTED = 5

    @staticmethod
    def is_finished(status: "RequestStatus") -> bool:
        return status >= RequestStatus.FINISHED_STOPPED


FINISH_REASON_STRINGS = {
    RequestStatus.FINISHED_STOPPED: "stop",
    RequestStatus.FINISHED_LENGTH_CAPPED: "length",
    RequestStatus.FINISHED_ABORTED: "abort",
}


class Request:

    def __init__(
        self,
        request_id: str,
        prompt_token_ids: list[int],
        sampling_params: SamplingParams,
        eos_token_id: Optional[int] = None,
        arrival_time: Optional[float] = None,
        priority: int = 0,
        prompt: Optional[str] = None,
        lora_id: int = 0,
        mm_data: Optional[dict] = None,
        mm_hash: int = 0,
        encoder_tokens: int = 0,
    ) -> None:
        self.request_id = request_id
        self.prompt_token_ids = prompt_token_ids
        self.prompt = prompt
        self.sampling_params = sampling_params
        self.eos_token_id = eos_token_id
        self.arrival_time = arrival_time if arrival_time is not None else time.time()
        self.priority = priority
        self.lora_id = lora_id
        # Multimodal payload ({"image": pixels}) + content hash used to
        # salt prefix-cache block hashes (vllm mm_hash extra_keys role).
        self.mm_data = mm_data
        self.mm_hash = mm_hash
        # Encoder tokens this request's first scheduled chunk triggers
        # (vision patches / audio frames / encoder prompt length);
        # metered by the scheduler's encoder admission budget.
        self.encoder_tokens = encoder_tokens

        self.status = RequestStatus.WAITING
        self.stop_reason: Optional[object] = None

        self.output_token_ids: list[int] = []
        # Async scheduling: tokens scheduled whose sampled values are not
        # yet known (role of AsyncScheduler's num_output_placeholders,
        # vllm/v1/core/sched/async_scheduler.py:12).
        self.num_output_placeholders = 0
        # Draft tokens proposed for the next step (spec decode).
        self.spec_token_ids: list[int] = []
        # Structured output (compiled by the engine frontend).
        self.grammar = None
        # All token ids: prompt + generated. Kept as one list so attention
        # metadata / block hashing index into a single sequence.
        self._all_token_ids: list[int] = list(prompt_token_ids)
        self.num_tokens: int = len(self._all_token_ids)
        self.num_output_tokens: int = 0

        # Number of tokens whose KV is already computed (incl. prefix-cache
        # hits). Catches up to num_tokens under the scheduler token budget;
        # there is no prefill/decode dichotomy (reference scheduler.py:442).
        self.num_computed_tokens = 0

        # Prefix-cache block hashes for this request's token stream
        # (computed lazily by the KV cache manager).
        self.block_hashes: list["BlockHash"] = []

        # Number of times this request was preempted.
        self.num_preemptions = 0
        # Stats
        self.first_token_time: Optional[float] = None
        self.num_cached_tokens = 0  # prefix-cache hit length

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    # num_tokens / num_output_tokens are PLAIN INTS maintained by
    # append_output_token_ids: they sit on the scheduler's per-request
    # per-step hot path, and Python property+len() cost is measurable at
    # batch 1024 (profiles/r02_summary.md control-plane section).

    @property
    def all_token_ids(self) -> list[int]:
        return self._all_token_ids

    @property
    def max_tokens(self) -> int:
        mt = self.sampling_params.max_tokens
        return mt if mt is not None else 2**31

    def append_output_token_ids(self, token_ids: list[int]) -> None:
        self.output_token_ids.extend(token_ids)
        self._all_token_ids.extend(token_ids)
        n = len(token_ids)
        self.num_tokens += n
        self.num_output_tokens += n

    def append_output_token(self, token_id: int) -> None:
        self.output_token_ids.append(token_id)
        self._all_token_ids.append(token_id)
        self.num_tokens += 1
        self.num_output_tokens += 1

    def is_finished(self) -> bool:
        return RequestStatus.is_finished(self.status)

    def get_finished_reason(self) -> Optional[str]:
        return FINISH_REASON_STRINGS.get(self.status)

    def __lt__(self, other: "Request") -> bool:
        # Priority queue ordering: lower priority value first, FCFS tiebreak.
        return (self.priority, self.arrival_time) < (
            other.priority,
            other.arrival_time,
        )

    def __repr__(self) -> str:
        return (
            f"Request(id={self.request_id}, status={self.status.name}, "
            f"prompt={self.num_prompt_tokens}, out={self.num_output_tokens}, "
            f"computed={self.num_computed_tokens})"
        )
