"""Offline LLM API (role of vllm/entrypoints/llm.py:67)."""

from __future__ import annotations

from typing import Optional, Sequence, Union

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.engine.llm_engine import LLMEngine
from vllm_amd.outputs import RequestOutput
from vllm_amd.sampling_params import SamplingParams


class LLM:

    def __init__(self, model: str = "llama-3-8b", **kwargs):
        engine_args = EngineArgs(model=model, **kwargs)
        self.engine = LLMEngine(engine_args.create_engine_config())

    def embed(self, prompts, pooling: str = "last"):
        """Embedding API: returns one pooled hidden-state vector per
        prompt (role of the reference's LLM.embed / encode)."""
        from vllm_amd.sampling_params import SamplingParams

        outs = self.generate(
            prompts, SamplingParams(pooling=pooling, max_tokens=1))
        return [o.pooled for o in outs]

    def generate(
        self,
        prompts: Union[str, list[int], Sequence[Union[str, list[int]]]],
        sampling_params: Optional[
            Union[SamplingParams, list[SamplingParams]]
        ] = None,
        lora: Optional[str] = None,
    ) -> list[RequestOutput]:
        if isinstance(prompts, str) or (
            prompts and isinstance(prompts, list)
            and isinstance(prompts[0], int)
        ):
            prompts = [prompts]
        n = len(prompts)
        if sampling_params is None:
            sampling_params = [SamplingParams()] * n
        elif isinstance(sampling_params, SamplingParams):
            sampling_params = [sampling_params] * n
        req_ids = []
        for prompt, params in zip(prompts, sampling_params):
            req_ids.append(
                self.engine.add_request(None, prompt, params, lora=lora))
        order = {rid: i for i, rid in enumerate(req_ids)}
        finals: dict[str, RequestOutput] = {}
        while self.engine.has_unfinished_requests():
            for out in self.engine.step():
                if out.finished:
                    finals[out.request_id] = out
        return [finals[rid] for rid in sorted(finals, key=order.get)]

    def shutdown(self) -> None:
        self.engine.shutdown()
