"""User-facing sampling parameters.

Parameter surface mirrors the reference engine's SamplingParams
(vllm/sampling_params.py:199): penalties, temperature, top-p/top-k/min-p,
seed, stop conditions, token limits, logprobs.
"""

from __future__ import annotations

import dataclasses
from dataclasses import dataclass
from enum import Enum
from typing import Optional, Union


class SamplingType(Enum):
    GREEDY = 0
    RANDOM = 1
    RANDOM_SEED = 2


class RequestOutputKind(Enum):
    # Return the full accumulated output every step.
    CUMULATIVE = 0
    # Return only the newly generated delta each step.
    DELTA = 1
    # Return only the final output when the request finishes.
    FINAL_ONLY = 2


_SAMPLING_EPS = 1e-5


@dataclass
class SamplingParams:
    n: int = 1
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    repetition_penalty: float = 1.0
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0  # 0 or -1 -> disabled
    min_p: float = 0.0
    seed: Optional[int] = None
    stop: Union[None, str, list[str]] = None
    # Keep the matched stop string in the output text (reference param).
    include_stop_str_in_output: bool = False
    stop_token_ids: Optional[list[int]] = None
    ignore_eos: bool = False
    max_tokens: Optional[int] = 16
    # Pooling request ("last" | "mean"): no tokens are sampled — the
    # request finishes after prefill with the pooled hidden state
    # (embedding models; role of the reference's PoolingParams).
    pooling: Optional[str] = None
    min_tokens: int = 0
    logprobs: Optional[int] = None
    prompt_logprobs: Optional[int] = None
    detokenize: bool = True
    skip_special_tokens: bool = True
    spaces_between_special_tokens: bool = True
    logit_bias: Optional[dict[int, float]] = None
    allowed_token_ids: Optional[list[int]] = None
    # Keep only the LAST N prompt tokens (reference extension of the
    # same name; applied at admission in llm_engine.add_request).
    truncate_prompt_tokens: Optional[int] = None
    bad_words: Optional[list[str]] = None
    # Structured output: generation constrained to one of these strings
    # (compiled to a token trie by the engine; see structured_output.py).
    guided_choice: Optional[list[str]] = None
    # Structured output: regex / JSON-schema constraints (compiled to a
    # DFA with per-state token masks; see guided_regex.py / guided_json.py).
    guided_regex: Optional[str] = None
    guided_json: Optional[object] = None  # dict schema, JSON string, or True
    # GBNF-style EBNF grammar (non-recursive subset; guided_grammar.py).
    guided_grammar: Optional[str] = None
    # Words that must never appear in the output (tokenized at request
    # admission into _bad_words_token_ids; a token is banned when it
    # would complete one of the sequences).
    bad_words: Optional[list[str]] = None
    _bad_words_token_ids: Optional[list[list[int]]] = None
    # Offline-API pluggable processors: fn(output_token_ids, logits) ->
    # logits, applied in request order before temperature (role of the
    # reference's per-request logits_processors).
    logits_processors: Optional[list] = None
    # Scheduling priority (lower = sooner under --scheduling-policy
    # priority; ties broken FCFS).
    priority: int = 0
    # response_format={"type": "json_object"}: any JSON object.
    guided_json_object: bool = False
    output_kind: RequestOutputKind = RequestOutputKind.CUMULATIVE

    def __post_init__(self) -> None:
        if isinstance(self.stop, str):
            self.stop = [self.stop]
        elif self.stop is None:
            self.stop = []
        if self.stop_token_ids is None:
            self.stop_token_ids = []
        self._verify()

    def _verify(self) -> None:
        if self.n < 1:
            raise ValueError(f"n must be >= 1, got {self.n}")
        if not -2.0 <= self.presence_penalty <= 2.0:
            raise ValueError("presence_penalty must be in [-2, 2]")
        if not -2.0 <= self.frequency_penalty <= 2.0:
            raise ValueError("frequency_penalty must be in [-2, 2]")
        if self.repetition_penalty <= 0.0:
            raise ValueError("repetition_penalty must be > 0")
        if self.temperature < 0.0:
            raise ValueError("temperature must be >= 0")
        if not 0.0 < self.top_p <= 1.0:
            raise ValueError("top_p must be in (0, 1]")
        if self.top_k < -1:
            raise ValueError(f"top_k must be -1, 0, or positive, got {self.top_k}")
        if self.top_k == -1:
            self.top_k = 0
        if not 0.0 <= self.min_p <= 1.0:
            raise ValueError("min_p must be in [0, 1]")
        if self.max_tokens is not None and self.max_tokens < 1:
            raise ValueError("max_tokens must be >= 1")
        if self.pooling not in (None, "last", "mean"):
            raise ValueError("pooling must be 'last' or 'mean'")
        if self.allowed_token_ids is not None \
                and not self.allowed_token_ids:
            raise ValueError(
                "allowed_token_ids must be non-empty when given "
                "(an empty set would mask every token)")
        if self.min_tokens < 0:
            raise ValueError("min_tokens must be >= 0")
        if self.temperature < _SAMPLING_EPS:
            # Greedy: normalize so downstream code can branch on temperature==0.
            self.temperature = 0.0
            self.top_p = 1.0
            self.top_k = 0
            self.min_p = 0.0

    @property
    def sampling_type(self) -> SamplingType:
        if self.temperature == 0.0:
            return SamplingType.GREEDY
        if self.seed is not None:
            return SamplingType.RANDOM_SEED
        return SamplingType.RANDOM

    @property
    def all_stop_token_ids(self) -> set[int]:
        return set(self.stop_token_ids or [])

    def clone(self) -> "SamplingParams":
        return dataclasses.replace(self)
