"""Synchronous LLMEngine + output processing (detokenization, stop strings).

Role of vllm/v1/engine/llm_engine.py and output_processor.py in one
place: add_request -> step() -> RequestOutputs, with incremental
detokenization and stop-string scanning on the client side of the engine.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Optional, Union

from vllm_amd.config import EngineConfig
from vllm_amd.core.sched_output import EngineCoreOutput
from vllm_amd.engine.core import EngineCore
from vllm_amd.outputs import CompletionOutput, RequestOutput
from vllm_amd.request import Request
from vllm_amd.sampling_params import RequestOutputKind, SamplingParams
from vllm_amd.tokenizer import IncrementalDetokenizer, TokenizerWrapper


@dataclass
class RequestState:
    request_id: str
    prompt: Optional[str]
    prompt_token_ids: list[int]
    params: SamplingParams
    detokenizer: Optional[IncrementalDetokenizer]
    output_token_ids: list[int] = field(default_factory=list)
    logprobs: list = field(default_factory=list)
    finished: bool = False
    finish_reason: Optional[str] = None
    stop_reason: Optional[object] = None
    arrival_time: float = field(default_factory=time.time)
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None
    pooled: Optional[list[float]] = None
    prompt_logprobs: Optional[list] = None
    num_cached_tokens: int = 0


class OutputProcessor:
    """EngineCoreOutput -> RequestOutput with stop-string handling."""

    def __init__(self, tokenizer: TokenizerWrapper):
        self.tokenizer = tokenizer
        self.states: dict[str, RequestState] = {}

    def add_request(
        self,
        request_id: str,
        prompt: Optional[str],
        prompt_token_ids: list[int],
        params: SamplingParams,
    ) -> None:
        detok = (
            IncrementalDetokenizer(
                self.tokenizer, len(prompt_token_ids),
                params.skip_special_tokens,
            )
            if params.detokenize
            else None
        )
        self.states[request_id] = RequestState(
            request_id=request_id,
            prompt=prompt,
            prompt_token_ids=prompt_token_ids,
            params=params,
            detokenizer=detok,
        )

    def process_outputs(
        self, core_outputs: list[EngineCoreOutput]
    ) -> tuple[list[RequestOutput], list[str]]:
        """Returns (request_outputs, request_ids_to_abort)."""
        outputs: list[RequestOutput] = []
        to_abort: list[str] = []
        now = time.time()
        for co in core_outputs:
            state = self.states.get(co.req_id)
            if state is None or state.finished:
                continue
            if state.first_token_time is None:
                state.first_token_time = now
            state.output_token_ids.extend(co.new_token_ids)
            if co.new_logprobs:
                state.logprobs.extend(co.new_logprobs)
            delta = ""
            if state.detokenizer is not None:
                delta = state.detokenizer.update(co.new_token_ids)
            finish_reason = co.finish_reason
            stop_reason = co.stop_reason

            # Stop-string scan on the decoded text.
            if state.params.stop and state.detokenizer is not None:
                text = state.detokenizer.output_text
                for s in state.params.stop:
                    idx = text.find(s)
                    if idx != -1:
                        end = (idx + len(s)
                               if state.params.include_stop_str_in_output
                               else idx)
                        state.detokenizer.output_text = text[:end]
                        finish_reason = "stop"
                        stop_reason = s
                        if not co.finish_reason:
                            to_abort.append(co.req_id)
                        break

            if co.pooled is not None:
                state.pooled = co.pooled
            if co.num_cached_tokens:
                state.num_cached_tokens = co.num_cached_tokens
            if co.new_prompt_logprobs:
                state.prompt_logprobs = (
                    (state.prompt_logprobs or []) + co.new_prompt_logprobs)
            if finish_reason:
                state.finished = True
                state.finish_reason = finish_reason
                state.stop_reason = stop_reason
                state.finish_time = now
            outputs.append(self._make_output(state, delta))
        return outputs, to_abort

    def _make_output(self, state: RequestState, delta: str) -> RequestOutput:
        kind = state.params.output_kind
        text = (
            state.detokenizer.output_text
            if state.detokenizer is not None
            else ""
        )
        if kind == RequestOutputKind.DELTA:
            text = delta
        comp = CompletionOutput(
            index=0,
            text=text,
            token_ids=list(state.output_token_ids),
            logprobs=state.logprobs or None,
            finish_reason=state.finish_reason,
            stop_reason=state.stop_reason,
        )
        metrics = None
        if state.finished:
            metrics = {
                "arrival_time": state.arrival_time,
                "first_token_time": state.first_token_time,
                "finish_time": state.finish_time,
            }
        return RequestOutput(
            request_id=state.request_id,
            prompt=state.prompt,
            prompt_token_ids=state.prompt_token_ids,
            outputs=[comp],
            finished=state.finished,
            metrics=metrics,
            pooled=getattr(state, "pooled", None),
            prompt_logprobs=getattr(state, "prompt_logprobs", None),
            num_cached_tokens=getattr(state, "num_cached_tokens", 0),
        )

    def release(self, request_id: str) -> None:
        self.states.pop(request_id, None)


class LLMEngine:

    def __init__(self, config: EngineConfig):
        from vllm_amd.plugins import load_plugins

        load_plugins()
        self.config = config
        if config.parallel_config.multiprocess_engine:
            from vllm_amd.engine.core_client import EngineCoreClient

            self.engine_core = EngineCoreClient(config)
        else:
            self.engine_core = EngineCore(config)
        self.tokenizer = TokenizerWrapper(config.model_config.tokenizer)
        self.output_processor = OutputProcessor(self.tokenizer)
        self._request_counter = 0
        self._trace_fh = None
        trace_file = config.observability_config.trace_file
        if trace_file:
            self._trace_fh = open(trace_file, "a", buffering=1)
        self._otel = None
        otlp = config.observability_config.otlp_traces_endpoint
        if otlp:
            from vllm_amd.tracing import OtelSpanExporter

            self._otel = OtelSpanExporter(otlp)

    @property
    def is_driver(self) -> bool:
        return self.engine_core.is_driver

    def add_request(
        self,
        request_id: Optional[str],
        prompt: Union[str, list[int]],
        params: Optional[SamplingParams] = None,
        lora: Optional[str] = None,
    ) -> str:
        if request_id is None:
            request_id = f"req-{self._request_counter}"
            self._request_counter += 1
        params = params or SamplingParams()
        mm_data = None
        mm_hash = 0
        encoder_tokens = 0
        if isinstance(prompt, dict):
            # {"prompt" | "prompt_token_ids", "multi_modal_data":
            #  {"image": pixels [3,S,S]}} (reference TextPrompt /
            # TokensPrompt with multi_modal_data).
            mm_data = prompt.get("multi_modal_data")
            # Encoder-decoder text prompts (reference ExplicitEncoder
            # DecoderPrompt shape): the encoder prompt rides the same
            # per-request cached-states path as audio.
            enc = prompt.get("encoder_prompt_token_ids")
            if enc is None and prompt.get("encoder_prompt") is not None:
                enc = self.tokenizer.encode(prompt["encoder_prompt"])
            if enc is not None:
                mm_data = dict(mm_data or {})
                mm_data["encoder_tokens"] = list(enc)
            if "prompt_token_ids" in prompt:
                prompt = list(prompt["prompt_token_ids"])
            else:
                prompt = prompt["prompt"]
            if mm_data and mm_data.get("encoder_tokens") is not None:
                spec = self.config.model_config.spec
                if spec.encoder_layers == 0:
                    raise ValueError(
                        f"model {spec.name} has no text encoder "
                        "(encoder_prompt requires an encoder-decoder "
                        "model)")
                import hashlib as _hashlib

                mm_hash = int.from_bytes(_hashlib.sha256(
                    bytes(str(mm_data["encoder_tokens"]),
                          "utf-8")).digest()[:8], "little")
                encoder_tokens = len(mm_data["encoder_tokens"])
            if mm_data and mm_data.get("audio") is not None:
                from vllm_amd.audio import audio_content_hash

                spec = self.config.model_config.spec
                if spec.audio_encoder_layers == 0:
                    raise ValueError(
                        f"model {spec.name} has no audio encoder")
                mm_hash = audio_content_hash(mm_data)
                import torch as _torch

                n_samples = _torch.as_tensor(
                    mm_data["audio"]).numel()
                # conv2 downsamples 2x: encoder rows ~= frames / 2.
                encoder_tokens = max(1, (1 + n_samples // 160) // 2)
            if mm_data and mm_data.get("image") is not None:
                from vllm_amd.multimodal import (expand_image_placeholders,
                                                 mm_content_hash)

                spec = self.config.model_config.spec
                if spec.vision_layers == 0:
                    raise ValueError(
                        f"model {spec.name} has no vision encoder")
                if isinstance(prompt, str):
                    prompt = self.tokenizer.encode(prompt)
                npatch = (spec.image_size // spec.vision_patch) ** 2
                import torch as _torch

                nimg = (1 if _torch.as_tensor(
                    mm_data["image"]).dim() == 3 else
                    _torch.as_tensor(mm_data["image"]).shape[0])
                prompt = expand_image_placeholders(
                    prompt, spec.image_token_id, npatch, nimg)
                mm_hash = mm_content_hash(mm_data)
                encoder_tokens = npatch * nimg
        if isinstance(prompt, str):
            prompt_text = prompt
            prompt_token_ids = self.tokenizer.encode(prompt)
        else:
            prompt_text = None
            prompt_token_ids = list(prompt)
        if params.truncate_prompt_tokens is not None:
            if params.truncate_prompt_tokens < 1:
                raise ValueError("truncate_prompt_tokens must be >= 1")
            if mm_data and mm_data.get("image") is not None:
                # Truncation could cut through the expanded per-patch
                # placeholder spans and desync them from the features.
                raise ValueError(
                    "truncate_prompt_tokens is not supported with "
                    "image prompts")
            prompt_token_ids = prompt_token_ids[
                -params.truncate_prompt_tokens:]
        if not prompt_token_ids:
            raise ValueError("empty prompt")
        max_len = self.config.model_config.max_model_len
        if len(prompt_token_ids) >= max_len:
            raise ValueError(
                f"prompt length {len(prompt_token_ids)} >= max_model_len "
                f"{max_len}"
            )
        spec_cfg = self.config.model_config.spec
        if spec_cfg.pooling_only:
            if not params.pooling:
                raise ValueError(
                    f"model {spec_cfg.name} is an embedding encoder: "
                    "only pooling requests are accepted")
            budget = self.config.scheduler_config.max_num_batched_tokens
            if len(prompt_token_ids) > budget:
                raise ValueError(
                    f"prompt length {len(prompt_token_ids)} exceeds the "
                    f"encoder's single-pass budget {budget} "
                    "(bidirectional attention cannot be chunked)")
        if params.logit_bias:
            vocab = self.config.model_config.spec.vocab_size
            bad = [t for t in params.logit_bias if not 0 <= t < vocab]
            if bad:
                # An out-of-range index would crash the sampler's
                # index_add_ mid-step and poison the engine loop.
                raise ValueError(
                    f"logit_bias token ids out of vocab range: {bad[:5]}")
        eos = self.tokenizer.eos_token_id
        if eos is None:
            eos = self.config.model_config.spec.eos_token_id
        request = Request(
            request_id=request_id,
            prompt_token_ids=prompt_token_ids,
            sampling_params=params,
            eos_token_id=eos,
            prompt=prompt_text,
            priority=params.priority,
            lora_id=self.config.model_config.lora_id_of(lora),
            mm_data=mm_data,
            mm_hash=mm_hash,
            encoder_tokens=encoder_tokens,
        )
        if params.bad_words and params._bad_words_token_ids is None:
            # Tokenize both bare and space-prefixed spellings (the
            # reference's bad_words semantics).
            seqs = []
            for w in params.bad_words:
                for variant in (w, " " + w):
                    ids = self.tokenizer.encode(variant)
                    if ids:
                        seqs.append(ids)
            params._bad_words_token_ids = seqs
        pp = self.config.parallel_config.pipeline_parallel_size
        if pp > 1 and (params.pooling or params.prompt_logprobs):
            # Hidden states / logits live on the LAST pp stage while the
            # driver is stage 0; routing them back is a round-2 item.
            raise ValueError(
                "pooling and prompt_logprobs require pp=1")
        if params.guided_choice:
            from vllm_amd.structured_output import compile_choice_grammar

            request.grammar = compile_choice_grammar(
                params.guided_choice, self.tokenizer, eos
            )
        elif (params.guided_regex or params.guided_grammar
                or params.guided_json is not None
                or params.guided_json_object):
            from vllm_amd.guided_json import any_json_regex, schema_to_regex
            from vllm_amd.guided_regex import RegexGrammar

            if params.guided_regex:
                pattern = params.guided_regex
            elif params.guided_grammar:
                from vllm_amd.guided_grammar import (
                    GrammarError,
                    grammar_to_regex,
                )

                try:
                    pattern = grammar_to_regex(params.guided_grammar)
                except GrammarError as e:
                    if "recursive rule" not in str(e):
                        raise
                    # Rule cycles (balanced parens, nested JSON, ...)
                    # need the pushdown matcher; acyclic grammars stay
                    # on the cheaper regex DFA.
                    from vllm_amd.guided_ebnf import EbnfGrammar

                    pattern = None
                    request.grammar = EbnfGrammar(
                        params.guided_grammar, self.tokenizer, eos)
            elif params.guided_json is not None \
                    and params.guided_json is not True:
                try:
                    pattern = schema_to_regex(params.guided_json)
                except ValueError:
                    raise
                except Exception as e:  # malformed schema ($ref, types)
                    raise ValueError(
                        f"invalid guided_json schema: {e!r}") from e
            else:
                pattern = any_json_regex()
            if request.grammar is None:
                request.grammar = RegexGrammar(
                    pattern, self.tokenizer, eos)
        self.engine_core.add_request(request)
        self.output_processor.add_request(
            request_id, prompt_text, prompt_token_ids, params
        )
        return request_id

    def abort_request(self, request_ids: list[str]) -> None:
        self.engine_core.abort_requests(request_ids)
        for rid in request_ids:
            self.output_processor.release(rid)

    def step(self) -> list[RequestOutput]:
        core_outputs = self.engine_core.step()
        outputs, to_abort = self.output_processor.process_outputs(
            core_outputs
        )
        if to_abort:
            self.engine_core.abort_requests(to_abort)
        for out in outputs:
            if out.finished:
                if self._trace_fh is not None:
                    self._write_trace(out)
                if self._otel is not None:
                    comp = out.outputs[0]
                    self._otel.export_request_span(
                        out.request_id,
                        self.config.model_config.model,
                        out.metrics or {},
                        len(out.prompt_token_ids),
                        len(comp.token_ids),
                        comp.finish_reason)
                self.output_processor.release(out.request_id)
                if self.engine_core.scheduler is not None:
                    self.engine_core.scheduler.release_request(
                        out.request_id)
        return outputs

    def _write_trace(self, out: RequestOutput) -> None:
        import json as _json

        m = out.metrics or {}
        comp = out.outputs[0]
        arrival = m.get("arrival_time")
        first = m.get("first_token_time")
        finish = m.get("finish_time")
        self._trace_fh.write(_json.dumps({
            "request_id": out.request_id,
            "arrival_time": arrival,
            "first_token_time": first,
            "finish_time": finish,
            "ttft_s": (first - arrival
                       if first is not None and arrival else None),
            "e2e_s": (finish - arrival
                      if finish is not None and arrival else None),
            "prompt_tokens": len(out.prompt_token_ids),
            "output_tokens": len(comp.token_ids),
            "finish_reason": comp.finish_reason,
        }) + "\n")

    def has_unfinished_requests(self) -> bool:
        return self.engine_core.has_unfinished_requests()

    def sleep(self, level: int = 1) -> None:
        self.engine_core.sleep(level)

    def wake_up(self) -> None:
        self.engine_core.wake_up()

    def is_sleeping(self) -> bool:
        return self.engine_core.is_sleeping()

    def check_health(self) -> None:
        fn = getattr(self.engine_core, "check_health", None)
        if fn is not None:
            fn()

    def shutdown(self) -> None:
        if self._trace_fh is not None:
            self._trace_fh.close()
            self._trace_fh = None
        if self._otel is not None:
            self._otel.shutdown()
            self._otel = None
        self.engine_core.shutdown()
