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
        # Parallel sampling (params.n > 1): fan out n engine requests per
        # prompt — the prefix cache dedups the shared prompt KV; seeded
        # runs offset the seed per branch so branches differ.
        import dataclasses as _dc

        req_ids = []          # one per engine request
        req_of = {}           # engine req id -> (prompt_idx, branch_idx)
        for p_idx, (prompt, params) in enumerate(
                zip(prompts, sampling_params)):
            for b in range(max(params.n, 1)):
                branch = params
                if params.n > 1:
                    branch = _dc.replace(
                        params, n=1,
                        seed=(params.seed + b
                              if params.seed is not None else None))
                rid = self.engine.add_request(None, prompt, branch,
                                              lora=lora)
                req_ids.append(rid)
                req_of[rid] = (p_idx, b)
        finals: dict[str, RequestOutput] = {}
        while self.engine.has_unfinished_requests():
            for out in self.engine.step():
                if out.finished:
                    finals[out.request_id] = out
        results: list[RequestOutput] = []
        for p_idx in range(len(prompts)):
            branches = sorted(
                (rid for rid in req_ids if req_of[rid][0] == p_idx),
                key=lambda rid: req_of[rid][1])
            base = finals[branches[0]]
            if len(branches) > 1:
                outs = []
                for b, rid in enumerate(branches):
                    comp = finals[rid].outputs[0]
                    comp.index = b
                    outs.append(comp)
                base = _dc.replace(base, outputs=outs)
            results.append(base)
        return results

    def shutdown(self) -> None:
        self.engine.shutdown()
