"""Anthropic Messages API (/v1/messages) compatibility router (role of
the reference's vllm/entrypoints/anthropic/). Mounted by the OpenAI API
server onto the same app/engine."""

from __future__ import annotations

import json
import time
from typing import Any, AsyncGenerator, Optional, Union

from fastapi import APIRouter
from fastapi.responses import JSONResponse, StreamingResponse
from pydantic import BaseModel

from vllm_amd.sampling_params import RequestOutputKind, SamplingParams


class AnthropicMessage(BaseModel):
    role: str
    content: Union[str, list[dict[str, Any]]]

    def text(self, image_sentinel: str = "") -> str:
        if isinstance(self.content, str):
            return self.content
        parts = []
        for seg in self.content:
            if seg.get("type") == "text":
                parts.append(seg.get("text", ""))
            elif seg.get("type") == "image" and image_sentinel:
                parts.append(image_sentinel)
        return "".join(parts)

    def image_payloads(self) -> list[str]:
        """base64 payloads of image content blocks, in order."""
        if not isinstance(self.content, list):
            return []
        return [(seg.get("source") or {}).get("data", "")
                for seg in self.content
                if seg.get("type") == "image"
                and (seg.get("source") or {}).get("type") == "base64"]


class MessagesRequest(BaseModel):
    model: str
    messages: list[AnthropicMessage]
    max_tokens: int
    system: Optional[Union[str, list[dict[str, Any]]]] = None
    stop_sequences: Optional[list[str]] = None
    stream: bool = False
    temperature: float = 1.0
    top_k: int = 0
    top_p: float = 1.0
    metadata: Optional[dict] = None
    # Anthropic tool use: [{name, description, input_schema}].
    tools: Optional[list[dict[str, Any]]] = None
    tool_choice: Optional[dict[str, Any]] = None


def build_anthropic_router(state) -> APIRouter:
    router = APIRouter()
    engine = state.engine

    def to_prompt(req: MessagesRequest, image_sentinel: str = "") -> str:
        parts = []
        if req.system:
            sys_text = req.system if isinstance(req.system, str) else \
                "".join(s.get("text", "") for s in req.system)
            parts.append(f"<|system|>\n{sys_text}")
        if req.tools:
            # Map Anthropic tool specs onto the hermes preamble the
            # open models are trained for (input_schema -> parameters).
            from vllm_amd.entrypoints.tool_parser import render_tools_block

            hermes = [{"name": t.get("name"),
                       "description": t.get("description", ""),
                       "parameters": t.get("input_schema", {})}
                      for t in req.tools]
            parts.append(f"<|system|>\n{render_tools_block(hermes)}")
        for m in req.messages:
            if isinstance(m.content, list):
                # tool_result blocks round-trip as tool turns.
                for seg in m.content:
                    if seg.get("type") == "tool_result":
                        parts.append(
                            "<|tool|>\n"
                            + json.dumps(seg.get("content", "")))
            parts.append(f"<|{m.role}|>\n{m.text(image_sentinel)}")
        parts.append("<|assistant|>\n")
        return "\n".join(parts)

    def build_prompt(req: MessagesRequest):
        """Engine prompt: plain text, or the multimodal dict prompt
        when messages carry Anthropic base64 image blocks (decoded and
        spliced as placeholder tokens, same route as the OpenAI chat mm
        path)."""
        payloads = [p for m in req.messages for p in m.image_payloads()]
        if not payloads:
            return to_prompt(req)
        import base64

        from vllm_amd.entrypoints.openai.api_server import (
            _IMG_SENTINEL, decode_image_bytes)

        spec = engine.config.model_config.spec
        if spec.vision_layers == 0:
            raise ValueError(
                f"model {spec.name} does not support image input")
        text = to_prompt(req, image_sentinel=_IMG_SENTINEL)
        pieces = text.split(_IMG_SENTINEL)
        ids = []
        for i, piece in enumerate(pieces):
            if piece:
                ids.extend(engine.tokenizer.encode(
                    piece, add_special_tokens=(i == 0)))
            if i < len(pieces) - 1:
                ids.append(spec.image_token_id)
        import torch

        try:
            imgs = [decode_image_bytes(base64.b64decode(p),
                                       spec.image_size)
                    for p in payloads]
        except Exception as e:  # noqa: BLE001
            raise ValueError(f"could not decode image: {e}") from e
        image = torch.stack(imgs) if len(imgs) > 1 else imgs[0]
        return {"prompt_token_ids": ids,
                "multi_modal_data": {"image": image}}

    @router.post("/v1/messages/count_tokens")
    async def count_tokens(req: MessagesRequest):
        """Anthropic token-counting endpoint: tokenize the rendered
        prompt without generating."""
        prompt = to_prompt(req)
        ids = state.engine.tokenizer.encode(prompt)
        return {"input_tokens": len(ids)}

    @router.post("/v1/messages")
    async def messages(req: MessagesRequest):
        params = SamplingParams(
            temperature=req.temperature,
            top_p=req.top_p,
            top_k=req.top_k,
            max_tokens=req.max_tokens,
            stop=req.stop_sequences,
            output_kind=(RequestOutputKind.DELTA if req.stream
                         else RequestOutputKind.FINAL_ONLY),
        )
        try:
            prompt = build_prompt(req)
        except ValueError as e:
            from fastapi.responses import JSONResponse

            return JSONResponse(
                {"type": "error",
                 "error": {"type": "invalid_request_error",
                           "message": str(e)}}, status_code=400)
        lora = req.model if req.model in state.lora_names else None
        rid = f"msg_{int(time.time() * 1e6):x}"
        state.num_requests += 1

        if req.stream:
            async def gen() -> AsyncGenerator[str, None]:
                start = {
                    "type": "message_start",
                    "message": {"id": rid, "type": "message",
                                "role": "assistant", "content": [],
                                "model": req.model,
                                "usage": {"input_tokens": 0,
                                          "output_tokens": 0}},
                }
                yield ("event: message_start\n"
                       f"data: {json.dumps(start)}\n\n")
                yield ("event: content_block_start\n"
                       'data: {"type": "content_block_start", "index": 0, '
                       '"content_block": {"type": "text", "text": ""}}\n\n')
                n_out = 0
                stop_reason = "end_turn"
                async for out in engine.generate(prompt, params, rid,
                                                 lora=lora):
                    comp = out.outputs[0]
                    n_out = len(comp.token_ids)
                    if comp.text:
                        delta = {"type": "content_block_delta", "index": 0,
                                 "delta": {"type": "text_delta",
                                           "text": comp.text}}
                        yield ("event: content_block_delta\n"
                               f"data: {json.dumps(delta)}\n\n")
                    if comp.finish_reason == "length":
                        stop_reason = "max_tokens"
                    elif comp.finish_reason == "stop":
                        stop_reason = ("stop_sequence"
                                       if comp.stop_reason is not None
                                       and not isinstance(comp.stop_reason,
                                                          int)
                                       else "end_turn")
                yield ("event: content_block_stop\n"
                       'data: {"type": "content_block_stop", "index": 0}'
                       "\n\n")
                md = {"type": "message_delta",
                      "delta": {"stop_reason": stop_reason},
                      "usage": {"output_tokens": n_out}}
                yield f"event: message_delta\ndata: {json.dumps(md)}\n\n"
                yield ('event: message_stop\n'
                       'data: {"type": "message_stop"}\n\n')
            return StreamingResponse(gen(), media_type="text/event-stream")

        final = None
        try:
            async for out in engine.generate(prompt, params, rid, lora=lora):
                final = out
        except ValueError as e:
            return JSONResponse(
                {"type": "error",
                 "error": {"type": "invalid_request_error",
                           "message": str(e)}},
                status_code=400,
            )
        comp = final.outputs[0]
        state.num_prompt_tokens += len(final.prompt_token_ids)
        state.num_generation_tokens += len(comp.token_ids)
        stop_reason = ("max_tokens" if comp.finish_reason == "length"
                       else "end_turn")
        content: list[dict[str, Any]] = []
        text = comp.text
        if req.tools:
            # Hermes-format calls map to Anthropic tool_use blocks.
            from vllm_amd.entrypoints.tool_parser import (
                parse_hermes_tool_calls)

            text, calls = parse_hermes_tool_calls(text)
            for c in calls:
                content.append({
                    "type": "tool_use",
                    "id": c.id.replace("call_", "toolu_", 1),
                    "name": c.name,
                    "input": json.loads(c.arguments or "{}"),
                })
            if calls:
                stop_reason = "tool_use"
        if text:
            content.insert(0, {"type": "text", "text": text})
        if not content:
            content = [{"type": "text", "text": ""}]
        return {
            "id": rid,
            "type": "message",
            "role": "assistant",
            "model": req.model,
            "content": content,
            "stop_reason": stop_reason,
            "stop_sequence": (comp.stop_reason
                              if isinstance(comp.stop_reason, str) else None),
            "usage": {
                "input_tokens": len(final.prompt_token_ids),
                "output_tokens": len(comp.token_ids),
            },
        }

    return router
