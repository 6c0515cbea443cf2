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

    def chat(
        self,
        messages,
        sampling_params: Optional[SamplingParams] = None,
        add_generation_prompt: bool = True,
    ) -> list[RequestOutput]:
        """Offline chat API (role of the reference's LLM.chat): messages
        go through the same chat templating as the server, then
        generate(). Accepts one conversation (list of dicts) or a list
        of conversations."""
        from vllm_amd.entrypoints.openai.api_server import (
            apply_chat_template)
        from vllm_amd.entrypoints.openai.protocol import ChatMessage

        if messages and isinstance(messages[0], dict):
            messages = [messages]
        prompts = []
        for conv in messages:
            msgs = [ChatMessage(**m) for m in conv]
            image_urls = [u for m in msgs for u in m.image_urls()]
            if image_urls:
                # Same path as the server's multimodal chat: data: URLs
                # decoded, placeholder tokens spliced, pixels on the
                # engine dict prompt.
                from vllm_amd.entrypoints.openai.api_server import (
                    build_mm_chat_prompt)

                prompts.append(build_mm_chat_prompt(
                    self.engine, msgs, add_generation_prompt,
                    image_urls))
            else:
                prompts.append(apply_chat_template(
                    self.engine.tokenizer, msgs, add_generation_prompt))
        return self.generate(prompts, sampling_params)

    def beam_search(
        self,
        prompts,
        beam_width: int = 4,
        max_tokens: int = 16,
        length_penalty: float = 1.0,
    ) -> list[list[tuple[list[int], float]]]:
        """Beam search emulated over the engine (role of the reference's
        LLM.beam_search, vllm/entrypoints/llm.py beam_search): each round
        extends every live beam by one token using top-`beam_width`
        logprobs, then keeps the best `beam_width` beams by cumulative
        logprob. The prefix cache makes each one-token extension a
        cache-hit prefill, so a round costs ~one decode step per beam.

        Returns, per prompt, beams as (token_ids, cumulative_logprob)
        sorted best-first (token_ids exclude the prompt)."""
        from vllm_amd.sampling_params import SamplingParams

        if isinstance(prompts, (str, dict)) or (
            prompts and isinstance(prompts, list)
            and isinstance(prompts[0], int)
        ):
            prompts = [prompts]
        results = []
        for prompt in prompts:
            if isinstance(prompt, str):
                prompt = self.engine.tokenizer.encode(prompt)
            # beams: (tokens_so_far_incl_prompt, cum_logprob, finished)
            beams = [(list(prompt), 0.0, False)]
            for _ in range(max_tokens):
                live = [b for b in beams if not b[2]]
                if not live:
                    break
                params = SamplingParams(
                    temperature=0.0, max_tokens=1, logprobs=beam_width,
                    ignore_eos=False, detokenize=False)
                outs = self.generate([b[0] for b in live], params)
                candidates = [b for b in beams if b[2]]
                for (toks, cum, _), out in zip(live, outs):
                    comp = out.outputs[0]
                    eos_id = self.engine.config.model_config.spec.                         eos_token_id
                    lps = (comp.logprobs or [{}])[0]
                    if not lps:
                        tok = comp.token_ids[0]
                        candidates.append((toks + [tok], cum, True))
                        continue
                    for tok, lp in lps.items():
                        done = tok == eos_id
                        candidates.append(
                            (toks + [int(tok)], cum + lp, done))
                # keep the best beam_width by length-penalized logprob
                def key(b):
                    gen_len = max(len(b[0]) - len(prompt), 1)
                    return b[1] / (gen_len ** length_penalty)

                beams = sorted(candidates, key=key, reverse=True)
                beams = beams[:beam_width]
            results.append(sorted(
                [(b[0][len(prompt):], b[1]) for b in beams],
                key=lambda x: -x[1]))
        return results

    def generate(
        self,
        prompts: Union[str, list[int], Sequence[Union[str, list[int]]]],
        sampling_params: Optional[
            Union[SamplingParams, list[SamplingParams]]
        ] = None,
        lora: Optional[str] = None,
    ) -> list[RequestOutput]:
        if isinstance(prompts, (str, dict)) or (
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
