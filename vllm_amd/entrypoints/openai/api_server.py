"""OpenAI-compatible API server (role of the reference's
vllm/entrypoints/openai/api_server.py:189 build_app / :751 run_server).

FastAPI app over AsyncLLM: /v1/completions, /v1/chat/completions (both
streaming and non-streaming), /v1/models, /tokenize, /detokenize,
/health, /version, /metrics (Prometheus text format).

Run:  python -m vllm_amd.entrypoints.openai.api_server --model llama-3-8b
(or `python -m vllm_amd serve ...`).
"""

from __future__ import annotations

import argparse
import json
import time
from typing import AsyncGenerator, Optional

from fastapi import FastAPI, Request
from fastapi.responses import JSONResponse, Response, StreamingResponse

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.engine.async_llm import AsyncLLM
from vllm_amd.entrypoints.openai.protocol import (
    ChatChoice, ChatCompletionMessage, ChatCompletionRequest,
    ChatCompletionResponse, ChatCompletionStreamResponse, ChatStreamChoice,
    CompletionChoice, CompletionRequest, CompletionResponse, DeltaMessage,
    DetokenizeRequest, DetokenizeResponse, ErrorResponse, ModelCard,
    ModelList, TokenizeRequest, TokenizeResponse, UsageInfo, random_id,
)

VERSION = "0.1.0"


def _error(msg: str, code: int = 400) -> JSONResponse:
    return JSONResponse(
        ErrorResponse(message=msg, code=code).model_dump(), status_code=code
    )


def apply_chat_template(tokenizer, messages, add_generation_prompt=True):
    """HF chat template when a real tokenizer is loaded; otherwise a
    simple role-tagged fallback (mock tokenizer / no template)."""
    hf = getattr(tokenizer, "tokenizer", None)
    if hf is not None and getattr(hf, "chat_template", None):
        return hf.apply_chat_template(
            [{"role": m.role, "content": m.text()} for m in messages],
            tokenize=False,
            add_generation_prompt=add_generation_prompt,
        )
    parts = [f"<|{m.role}|>\n{m.text()}" for m in messages]
    if add_generation_prompt:
        parts.append("<|assistant|>\n")
    return "\n".join(parts)


class ServerState:
    def __init__(self, engine: AsyncLLM, model_name: str):
        self.engine = engine
        self.model_name = model_name
        self.lora_names = list(
            engine.config.model_config.lora_modules or {})
        self.max_model_len = engine.config.model_config.max_model_len
        # Prometheus counters.
        self.num_requests = 0
        self.num_prompt_tokens = 0
        self.num_generation_tokens = 0
        self.start_time = time.time()


def build_app(state: ServerState) -> FastAPI:
    app = FastAPI(title="vllm_amd OpenAI-compatible server")
    engine = state.engine

    from vllm_amd.entrypoints.anthropic_api import build_anthropic_router

    app.include_router(build_anthropic_router(state))

    @app.get("/health")
    async def health() -> Response:
        return Response(status_code=200)

    @app.get("/version")
    async def version():
        return {"version": VERSION}

    @app.get("/v1/models")
    async def list_models() -> ModelList:
        cards = [ModelCard(id=state.model_name,
                           max_model_len=state.max_model_len)]
        cards += [ModelCard(id=n, max_model_len=state.max_model_len)
                  for n in state.lora_names]
        return ModelList(data=cards)

    @app.post("/tokenize")
    async def tokenize(req: TokenizeRequest) -> TokenizeResponse:
        ids = engine.tokenizer.encode(req.prompt)
        return TokenizeResponse(tokens=ids, count=len(ids),
                                max_model_len=state.max_model_len)

    @app.post("/detokenize")
    async def detokenize(req: DetokenizeRequest) -> DetokenizeResponse:
        return DetokenizeResponse(prompt=engine.tokenizer.decode(req.tokens))

    @app.get("/metrics")
    async def metrics() -> Response:
        s = engine.stats()
        lines = [
            "# TYPE vllm_amd:num_requests_total counter",
            f"vllm_amd:num_requests_total {state.num_requests}",
            "# TYPE vllm_amd:prompt_tokens_total counter",
            f"vllm_amd:prompt_tokens_total {state.num_prompt_tokens}",
            "# TYPE vllm_amd:generation_tokens_total counter",
            f"vllm_amd:generation_tokens_total {state.num_generation_tokens}",
            "# TYPE vllm_amd:num_requests_running gauge",
            f"vllm_amd:num_requests_running {s.get('num_running', 0)}",
            "# TYPE vllm_amd:num_requests_waiting gauge",
            f"vllm_amd:num_requests_waiting {s.get('num_waiting', 0)}",
            "# TYPE vllm_amd:kv_blocks_free gauge",
            f"vllm_amd:kv_blocks_free {s.get('kv_blocks_free', 0)}",
        ]
        return Response("\n".join(lines) + "\n",
                        media_type="text/plain; version=0.0.4")

    # ------------------------------------------------------------------
    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        if req.n != 1 or (req.best_of or 1) != 1:
            return _error("only n=1 is supported")
        prompts = req.prompt
        if isinstance(prompts, str):
            prompts = [prompts]
        elif prompts and isinstance(prompts[0], int):
            prompts = [prompts]
        if len(prompts) != 1:
            return _error("batched prompts: send one prompt per request")
        prompt = prompts[0]
        params = req.to_sampling_params(req.stream)
        lora = req.model if req.model in state.lora_names else None
        state.num_requests += 1
        rid = random_id("cmpl")

        if req.stream:
            async def gen() -> AsyncGenerator[str, None]:
                try:
                    async for out in engine.generate(prompt, params, rid,
                                                     lora=lora):
                        comp = out.outputs[0]
                        state.num_generation_tokens += len(comp.token_ids)
                        chunk = {
                            "id": rid,
                            "object": "text_completion",
                            "created": int(time.time()),
                            "model": req.model,
                            "choices": [{
                                "index": 0,
                                "text": comp.text,
                                "logprobs": None,
                                "finish_reason": comp.finish_reason,
                            }],
                        }
                        yield f"data: {json.dumps(chunk)}\n\n"
                    yield "data: [DONE]\n\n"
                except Exception as e:  # noqa: BLE001
                    err = {"error": {"message": str(e)}}
                    yield f"data: {json.dumps(err)}\n\n"
            return StreamingResponse(gen(), media_type="text/event-stream")

        final = None
        try:
            async for out in engine.generate(prompt, params, rid, lora=lora):
                final = out
        except ValueError as e:
            return _error(str(e))
        comp = final.outputs[0]
        state.num_prompt_tokens += len(final.prompt_token_ids)
        state.num_generation_tokens += len(comp.token_ids)
        usage = UsageInfo(
            prompt_tokens=len(final.prompt_token_ids),
            completion_tokens=len(comp.token_ids),
            total_tokens=len(final.prompt_token_ids) + len(comp.token_ids),
        )
        return CompletionResponse(
            id=rid,
            model=req.model,
            choices=[CompletionChoice(
                index=0,
                text=(final.prompt or "") + comp.text if req.echo
                else comp.text,
                finish_reason=comp.finish_reason,
                stop_reason=comp.stop_reason
                if isinstance(comp.stop_reason, (int, str)) else None,
            )],
            usage=usage,
        )

    # ------------------------------------------------------------------
    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest, raw: Request):
        if req.n != 1:
            return _error("only n=1 is supported")
        prompt = apply_chat_template(
            engine.tokenizer, req.messages, req.add_generation_prompt
        )
        default_max = state.max_model_len
        params = req.to_sampling_params(req.stream, default_max)
        lora = req.model if req.model in state.lora_names else None
        state.num_requests += 1
        rid = random_id("chatcmpl")

        if req.stream:
            async def gen() -> AsyncGenerator[str, None]:
                first = ChatCompletionStreamResponse(
                    id=rid, model=req.model,
                    choices=[ChatStreamChoice(
                        index=0, delta=DeltaMessage(role="assistant",
                                                    content=""))],
                )
                yield f"data: {first.model_dump_json()}\n\n"
                try:
                    async for out in engine.generate(prompt, params, rid,
                                                     lora=lora):
                        comp = out.outputs[0]
                        state.num_generation_tokens += len(comp.token_ids)
                        chunk = ChatCompletionStreamResponse(
                            id=rid, model=req.model,
                            choices=[ChatStreamChoice(
                                index=0,
                                delta=DeltaMessage(content=comp.text),
                                finish_reason=comp.finish_reason,
                            )],
                        )
                        yield f"data: {chunk.model_dump_json()}\n\n"
                    yield "data: [DONE]\n\n"
                except Exception as e:  # noqa: BLE001
                    err = {"error": {"message": str(e)}}
                    yield f"data: {json.dumps(err)}\n\n"
            return StreamingResponse(gen(), media_type="text/event-stream")

        final = None
        try:
            async for out in engine.generate(prompt, params, rid, lora=lora):
                final = out
        except ValueError as e:
            return _error(str(e))
        comp = final.outputs[0]
        state.num_prompt_tokens += len(final.prompt_token_ids)
        state.num_generation_tokens += len(comp.token_ids)
        usage = UsageInfo(
            prompt_tokens=len(final.prompt_token_ids),
            completion_tokens=len(comp.token_ids),
            total_tokens=len(final.prompt_token_ids) + len(comp.token_ids),
        )
        return ChatCompletionResponse(
            id=rid,
            model=req.model,
            choices=[ChatChoice(
                index=0,
                message=ChatCompletionMessage(content=comp.text),
                finish_reason=comp.finish_reason or "stop",
            )],
            usage=usage,
        )

    return app


def make_server(engine_args: EngineArgs,
                served_model_name: Optional[str] = None):
    engine = AsyncLLM(engine_args.create_engine_config())
    state = ServerState(engine, served_model_name or engine_args.model)
    return build_app(state), state


def main() -> None:
    import uvicorn

    parser = argparse.ArgumentParser(
        description="vllm_amd OpenAI-compatible server")
    parser.add_argument("--host", type=str, default="0.0.0.0")
    parser.add_argument("--port", type=int, default=8000)
    parser.add_argument("--served-model-name", type=str, default=None)
    EngineArgs.add_cli_args(parser)
    args = parser.parse_args()
    engine_args = EngineArgs.from_cli_args(args)
    app, _ = make_server(engine_args, args.served_model_name)
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
