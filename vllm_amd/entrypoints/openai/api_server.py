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
    ChatChoice, ChatCompletionMessage, ChatCompletionRequest, ChatMessage,
    ChatCompletionResponse, ChatCompletionStreamResponse, ChatStreamChoice,
    CompletionChoice, CompletionRequest, CompletionResponse, DeltaMessage,
    DetokenizeRequest, DetokenizeResponse, EmbeddingData, EmbeddingRequest,
    EmbeddingResponse, ErrorResponse, ModelCard, ModelList,
    ResponsesRequest, RerankRequest,
    RerankResponse, RerankResult, ScoreData, ScoreRequest, ScoreResponse,
    TokenizeRequest, TokenizeResponse, UsageInfo, random_id,
)

VERSION = "0.1.0"


def _error(msg: str, code: int = 400) -> JSONResponse:
    return JSONResponse(
        ErrorResponse(message=msg, code=code).model_dump(), status_code=code
    )


def apply_chat_template(tokenizer, messages, add_generation_prompt=True,
                        tools=None, image_sentinel=None):
    """HF chat template when a real tokenizer is loaded; otherwise a
    simple role-tagged fallback (mock tokenizer / no template). `tools`
    go through the template's `tools=` kwarg when it takes one, else a
    hermes-style system preamble. `image_sentinel` marks image_url
    content parts for post-template token splicing."""
    hf = getattr(tokenizer, "tokenizer", None)
    if hf is not None and getattr(hf, "chat_template", None):
        dicts = []
        for m in messages:
            d = {"role": m.role, "content": m.text(image_sentinel)}
            if getattr(m, "tool_calls", None):
                d["tool_calls"] = m.tool_calls
            if getattr(m, "tool_call_id", None):
                d["tool_call_id"] = m.tool_call_id
            dicts.append(d)
        if tools:
            try:
                return hf.apply_chat_template(
                    dicts, tokenize=False, tools=tools,
                    add_generation_prompt=add_generation_prompt,
                )
            except Exception:  # template has no tools support
                from vllm_amd.entrypoints.tool_parser import (
                    render_tools_block)
                dicts = ([{"role": "system", "content":
                           render_tools_block(tools)}] + dicts)
        return hf.apply_chat_template(
            dicts, tokenize=False,
            add_generation_prompt=add_generation_prompt)
    parts = []
    if tools:
        from vllm_amd.entrypoints.tool_parser import render_tools_block
        parts.append(f"<|system|>\n{render_tools_block(tools)}")
    parts += [f"<|{m.role}|>\n{m.text(image_sentinel)}" for m in messages]
    if add_generation_prompt:
        parts.append("<|assistant|>\n")
    return "\n".join(parts)


# Marks image positions in templated text; NUL bytes cannot appear in
# JSON chat content, so the sentinel never collides with user text.
_IMG_SENTINEL = "\x00<image>\x00"


def _decode_image_data_url(url: str, size: int):
    """`data:image/...;base64,...` -> [3, size, size] float tensor scaled
    to [-1, 1] (the vision tower's expected range). Only data: URLs are
    accepted: this server has no network egress, so remote image_url
    fetching (which the reference does) fails fast with a clear error."""
    if not url.startswith("data:"):
        raise ValueError(
            "only data: image URLs are supported (no network egress)")
    import base64
    import io

    payload = url.split(",", 1)
    if len(payload) != 2:
        raise ValueError("malformed data: URL (missing comma)")
    return decode_image_bytes(base64.b64decode(payload[1]), size)


def decode_image_bytes(raw: bytes, size: int):
    """Encoded image bytes -> [3, size, size] float tensor in [-1, 1]."""
    import io

    from PIL import Image

    img = Image.open(io.BytesIO(raw)).convert("RGB")
    img = img.resize((size, size), Image.BILINEAR)
    import numpy as np
    import torch

    arr = torch.from_numpy(np.asarray(img, dtype=np.float32).copy())
    return arr.permute(2, 0, 1) / 127.5 - 1.0


def build_mm_chat_prompt(engine, messages, add_generation_prompt,
                         image_urls, tools=None):
    """Multimodal chat prompt: template with sentinels, splice one
    image placeholder token per image into the token ids, decode the
    data-URL images to a stacked pixel tensor. Returns the engine dict
    prompt ({"prompt_token_ids", "multi_modal_data"}); the engine
    expands each placeholder to per-patch tokens (llm_engine.py
    add_request) and salts prefix-cache hashes with the pixel content."""
    spec = engine.config.model_config.spec
    if spec.vision_layers == 0:
        raise ValueError(
            f"model {spec.name} does not support image input")
    text = apply_chat_template(
        engine.tokenizer, messages, add_generation_prompt,
        tools=tools, image_sentinel=_IMG_SENTINEL)
    pieces = text.split(_IMG_SENTINEL)
    if len(pieces) - 1 != len(image_urls):
        raise ValueError("chat template dropped image placeholders")
    ids: list[int] = []
    for i, piece in enumerate(pieces):
        if piece:
            ids.extend(engine.tokenizer.encode(
                piece, add_special_tokens=(i == 0)))
        if i < len(pieces) - 1:
            ids.append(spec.image_token_id)
    import torch

    try:
        imgs = [_decode_image_data_url(u, spec.image_size)
                for u in image_urls]
    except ValueError:
        raise
    except Exception as e:  # binascii / PIL decode errors -> 400
        raise ValueError(f"could not decode image: {e}") from e
    image = torch.stack(imgs) if len(imgs) > 1 else imgs[0]
    return {"prompt_token_ids": ids, "multi_modal_data": {"image": image}}


def _parse_multipart(body: bytes, content_type: str) -> dict:
    """Minimal multipart/form-data parser (the environment has no
    python-multipart): field name -> raw bytes."""
    import re as _re

    m = _re.search(r'boundary="?([^";,\s]+)"?', content_type)
    if not m:
        raise ValueError("multipart body without boundary")
    delim = b"--" + m.group(1).encode()
    out: dict[str, bytes] = {}
    for part in body.split(delim):
        part = part.strip(b"\r\n")
        if not part or part == b"--":
            continue
        head, sep, data = part.partition(b"\r\n\r\n")
        if not sep:
            continue
        hm = _re.search(rb'name="([^"]+)"', head)
        if hm:
            out[hm.group(1).decode()] = data
    return out


def _decode_wav(data: bytes):
    """WAV (PCM 8/16-bit) -> mono float32 waveform at 16 kHz. Only WAV:
    compressed formats would need ffmpeg, which this image lacks."""
    import io
    import wave as _wave

    import numpy as _np

    try:
        with _wave.open(io.BytesIO(data)) as w:
            sr = w.getframerate()
            nch = w.getnchannels()
            sw = w.getsampwidth()
            raw = w.readframes(w.getnframes())
    except Exception as e:
        raise ValueError(f"could not decode WAV audio: {e}") from e
    if sw == 2:
        arr = _np.frombuffer(raw, dtype=_np.int16).astype(
            _np.float32) / 32768.0
    elif sw == 1:
        arr = (_np.frombuffer(raw, dtype=_np.uint8).astype(_np.float32)
               - 128.0) / 128.0
    else:
        raise ValueError(f"unsupported WAV sample width {sw}")
    if nch > 1:
        arr = arr.reshape(-1, nch).mean(axis=1)
    if sr != 16_000 and len(arr) > 1:
        n_out = int(len(arr) * 16_000 / sr)
        arr = _np.interp(_np.linspace(0, len(arr) - 1, n_out),
                         _np.arange(len(arr)), arr).astype(_np.float32)
    return arr


class ServerState:
    def __init__(self, engine: AsyncLLM, model_name: str,
                 reasoning_parser: Optional[str] = None,
                 api_key: Optional[str] = None,
                 tool_call_parser: str = "hermes"):
        self.engine = engine
        self.model_name = model_name
        # Bearer token required on /v1/* when set (reference --api-key).
        self.api_key = api_key
        # hermes | mistral | llama3_json (tool_parser.py formats).
        self.tool_call_parser = tool_call_parser
        # Stored /v1/responses bodies (bounded LRU) + background tasks.
        from collections import OrderedDict

        class _LRU(OrderedDict):
            def __setitem__(self, k, v):
                super().__setitem__(k, v)
                self.move_to_end(k)
                while len(self) > 256:
                    self.popitem(last=False)

        self.responses_store: dict = _LRU()
        self.responses_tasks: dict = {}
        # "deepseek_r1" enables <think> splitting into reasoning_content.
        self.reasoning_parser = reasoning_parser
        self.lora_names = list(
            engine.config.model_config.lora_modules or {})
        self.max_model_len = engine.config.model_config.max_model_len
        # Prometheus counters + histograms.
        from vllm_amd.metrics import ServerMetrics

        self.metrics = ServerMetrics()
        self.num_requests = 0
        self.num_prompt_tokens = 0
        self.num_generation_tokens = 0
        self.start_time = time.time()


def build_app(state: ServerState) -> FastAPI:
    app = FastAPI(title="vllm_amd OpenAI-compatible server")
    engine = state.engine

    @app.middleware("http")
    async def request_id_and_auth(request: Request, call_next):
        if (state.api_key is not None
                and request.url.path.startswith("/v1")):
            auth = request.headers.get("Authorization", "")
            if auth != f"Bearer {state.api_key}":
                return JSONResponse(
                    {"error": {"message": "invalid or missing API key",
                               "type": "authentication_error"}},
                    status_code=401)
        response = await call_next(request)
        rid = request.headers.get("X-Request-Id") or random_id("req")
        response.headers["X-Request-Id"] = rid
        return response

    from vllm_amd.entrypoints.anthropic_api import build_anthropic_router

    app.include_router(build_anthropic_router(state))

    @app.get("/health")
    async def health() -> Response:
        try:
            engine.check_health()
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"error": str(e)}, status_code=503)
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

    @app.get("/v1/models/{model_id}")
    async def get_model(model_id: str):
        if model_id == state.model_name or model_id in state.lora_names:
            return ModelCard(id=model_id,
                             max_model_len=state.max_model_len)
        return _error(f"model {model_id!r} not found", 404)

    @app.get("/ping")
    @app.post("/ping")
    async def ping():
        # SageMaker-style liveness alias for /health (reference parity).
        try:
            engine.check_health()
        except Exception as e:  # noqa: BLE001
            return _error(str(e), 503)
        return Response(status_code=200)

    @app.post("/tokenize")
    async def tokenize(req: TokenizeRequest) -> TokenizeResponse:
        ids = engine.tokenizer.encode(req.prompt)
        return TokenizeResponse(tokens=ids, count=len(ids),
                                max_model_len=state.max_model_len)

    @app.post("/detokenize")
    async def detokenize(req: DetokenizeRequest):
        try:
            text = engine.tokenizer.decode(req.tokens)
        except Exception as e:  # out-of-range / negative ids
            return _error(f"could not detokenize: {e}")
        return DetokenizeResponse(prompt=text)

    @app.post("/sleep")
    async def sleep(raw: Request):
        level = int(raw.query_params.get("level", "1"))
        try:
            engine.sleep(level)
        except RuntimeError as e:
            return _error(str(e), 400)
        return Response(status_code=200)

    @app.post("/wake_up")
    async def wake_up():
        engine.wake_up()
        return Response(status_code=200)

    @app.post("/start_profile")
    async def start_profile():
        try:
            engine.start_profile()
        except Exception as e:  # noqa: BLE001
            return _error(str(e), 400)
        return Response(status_code=200)

    @app.post("/stop_profile")
    async def stop_profile():
        try:
            path = engine.stop_profile()
        except Exception as e:  # noqa: BLE001
            return _error(str(e), 400)
        return {"trace": path}

    @app.get("/is_sleeping")
    async def is_sleeping():
        return {"is_sleeping": engine.is_sleeping()}

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
            "# TYPE vllm_amd:prefix_cache_queries_total counter",
            "vllm_amd:prefix_cache_queries_total "
            f"{s.get('prefix_cache_queries', 0)}",
            "# TYPE vllm_amd:prefix_cache_hits_total counter",
            f"vllm_amd:prefix_cache_hits_total {s.get('prefix_cache_hits', 0)}",
            "# TYPE vllm_amd:num_preemptions_total counter",
            f"vllm_amd:num_preemptions_total {s.get('num_preemptions', 0)}",
            "# TYPE vllm_amd:encoder_deferrals_total counter",
            "vllm_amd:encoder_deferrals_total "
            f"{s.get('num_encoder_deferrals', 0)}",
            "# TYPE vllm_amd:spec_decode_num_draft_tokens_total counter",
            "vllm_amd:spec_decode_num_draft_tokens_total "
            f"{s.get('spec_tokens_drafted', 0)}",
            "# TYPE vllm_amd:spec_decode_num_accepted_tokens_total counter",
            "vllm_amd:spec_decode_num_accepted_tokens_total "
            f"{s.get('spec_tokens_accepted', 0)}",
        ]
        lines += state.metrics.render()
        return Response("\n".join(lines) + "\n",
                        media_type="text/plain; version=0.0.4")

    def _completion_logprobs(comp, lo: int = 0,
                             hi: Optional[int] = None):
        """OpenAI completions logprobs object from the engine's raw
        per-token {token_id: logprob} dicts."""
        if not comp.logprobs:
            return None
        tok = engine.tokenizer
        tokens, token_logprobs, top = [], [], []
        for tid, d in zip(comp.token_ids[lo:hi], comp.logprobs[lo:hi]):
            tokens.append(tok.decode([tid]))
            token_logprobs.append(d.get(tid))
            top.append({tok.decode([t]): lp for t, lp in d.items()})
        return {"tokens": tokens, "token_logprobs": token_logprobs,
                "top_logprobs": top, "text_offset": []}

    def _chat_logprobs(comp, lo: int = 0, hi: Optional[int] = None):
        """OpenAI chat logprobs object ({"content": [...]}); lo/hi
        slice the (cumulative) token list for streaming chunks."""
        if not comp.logprobs:
            return None
        tok = engine.tokenizer
        content = []
        for tid, d in zip(comp.token_ids[lo:hi], comp.logprobs[lo:hi]):
            tstr = tok.decode([tid])
            content.append({
                "token": tstr,
                "logprob": d.get(tid),
                "bytes": list(tstr.encode()),
                "top_logprobs": [
                    {"token": tok.decode([t]), "logprob": lp,
                     "bytes": list(tok.decode([t]).encode())}
                    for t, lp in d.items()
                ],
            })
        return {"content": content}

    def _branch_params(params, n):
        """n>1 parallel sampling: one engine request per branch, seed
        offset per branch so seeded branches differ."""
        import dataclasses as _dc

        if n <= 1:
            return [params]
        return [
            _dc.replace(params, seed=(params.seed + b
                                      if params.seed is not None else None))
            for b in range(n)
        ]

    async def _merge_streams(gens):
        """Merge n async generators into (index, item) events."""
        import asyncio

        queue: asyncio.Queue = asyncio.Queue()
        DONE = object()

        async def pump(i, g):
            try:
                async for item in g:
                    await queue.put((i, item, None))
            except Exception as e:  # noqa: BLE001
                await queue.put((i, None, e))
            await queue.put((i, DONE, None))

        tasks = [asyncio.ensure_future(pump(i, g))
                 for i, g in enumerate(gens)]
        done = 0
        try:
            while done < len(gens):
                i, item, err = await queue.get()
                if err is not None:
                    raise err
                if item is DONE:
                    done += 1
                    continue
                yield i, item
        finally:
            for t in tasks:
                t.cancel()

    # ------------------------------------------------------------------
    @app.post("/v1/completions")
    async def completions(req: CompletionRequest, raw: Request):
        best_of = req.best_of or req.n
        if best_of < req.n:
            return _error("best_of must be >= n")
        if best_of > req.n and req.stream:
            return _error("best_of with streaming is not supported")
        if not 1 <= req.n <= 64 or best_of > 64:
            return _error("n/best_of must be in [1, 64]")
        if req.suffix:
            # Same behavior as the reference: insertion mode is not
            # implemented, reject rather than silently ignore.
            return _error("suffix is not supported")
        prompts = req.prompt
        if isinstance(prompts, str):
            prompts = [prompts]
        elif prompts and isinstance(prompts[0], int):
            prompts = [prompts]
        if not prompts:
            return _error("prompt must not be empty")
        try:
            params = req.to_sampling_params(req.stream)
        except ValueError as e:
            return _error(str(e))
        echo_lp = bool(req.echo) and req.logprobs is not None \
            and not req.stream
        if echo_lp and params.prompt_logprobs is None:
            # OpenAI echo+logprobs includes the prompt tokens' logprobs
            # (first token null); request them engine-side.
            import dataclasses as _dc
            params = _dc.replace(params, prompt_logprobs=req.logprobs)
        forced_lp = False
        if best_of > req.n and params.logprobs is None:
            # Branch selection scores by chosen-token logprob; request it
            # engine-side and strip it from the response.
            import dataclasses as _dc
            params = _dc.replace(params, logprobs=1)
            forced_lp = True
        branches = _branch_params(params, best_of)
        lora = req.model if req.model in state.lora_names else None
        state.num_requests += 1
        rid = random_id("cmpl")
        # (prompt index, branch) flat fan-out; OpenAI choice index is
        # prompt-major: index = p * n + branch.
        fan = [(p_i, b, prompt, bp)
               for p_i, prompt in enumerate(prompts)
               for b, bp in enumerate(branches)]

        if req.stream:
            async def gen() -> AsyncGenerator[str, None]:
                from vllm_amd.metrics import RequestTimer
                timer = RequestTimer(state.metrics)
                n_gen = 0
                n_prompt_by: dict[int, int] = {}
                gens = [engine.generate(prompt, bp, f"{rid}-{p_i}-{b}",
                                        lora=lora)
                        for p_i, b, prompt, bp in fan]
                seen: dict[int, int] = {}
                try:
                    async for gi, out in _merge_streams(gens):
                        p_i, b = fan[gi][0], fan[gi][1]
                        comp = out.outputs[0]
                        # comp.token_ids is cumulative; count the delta.
                        new = len(comp.token_ids) - seen.get(gi, 0)
                        seen[gi] = len(comp.token_ids)
                        state.num_generation_tokens += new
                        timer.on_tokens(new)
                        n_gen += new
                        n_prompt_by[p_i] = len(out.prompt_token_ids)
                        n_prompt = sum(n_prompt_by.values())
                        lp_chunk = None
                        if req.logprobs is not None and new:
                            prev = seen[gi] - new
                            lp_chunk = _completion_logprobs(
                                comp, prev, prev + new)
                        chunk = {
                            "id": rid,
                            "object": "text_completion",
                            "created": int(time.time()),
                            "model": req.model,
                            "choices": [{
                                "index": p_i * req.n + b,
                                "text": comp.text,
                                "logprobs": lp_chunk,
                                "finish_reason": comp.finish_reason,
                            }],
                        }
                        yield f"data: {json.dumps(chunk)}\n\n"
                    timer.on_finish(n_prompt, n_gen)
                    if (req.stream_options or {}).get("include_usage"):
                        usage_chunk = {
                            "id": rid, "object": "text_completion",
                            "created": int(time.time()),
                            "model": req.model, "choices": [],
                            "usage": {
                                "prompt_tokens": n_prompt,
                                "completion_tokens": n_gen,
                                "total_tokens": n_prompt + n_gen,
                            },
                        }
                        yield f"data: {json.dumps(usage_chunk)}\n\n"
                    yield "data: [DONE]\n\n"
                except Exception as e:  # noqa: BLE001
                    err = {"error": {"message": str(e)}}
                    yield f"data: {json.dumps(err)}\n\n"
            return StreamingResponse(gen(), media_type="text/event-stream")

        import asyncio

        from vllm_amd.metrics import RequestTimer
        timer = RequestTimer(state.metrics)

        async def run_branch(p_i, b, prompt, bp):
            final = None
            async for out in engine.generate(prompt, bp,
                                             f"{rid}-{p_i}-{b}",
                                             lora=lora):
                final = out
            return final

        try:
            finals = await asyncio.gather(
                *(run_branch(*item) for item in fan))
        except ValueError as e:
            return _error(str(e))
        # best_of > n: keep the n best branches per prompt by mean
        # chosen-token logprob (OpenAI legacy semantics).
        out_index = [p_i * req.n + b for p_i, b, _, _ in fan]
        if best_of > req.n:
            kept = []
            for p_i in range(len(prompts)):
                group = [(gi, finals[gi]) for gi, item in enumerate(fan)
                         if item[0] == p_i]

                def score(f):
                    c = f.outputs[0]
                    lp = getattr(c, "cumulative_logprob", None)
                    if lp is None and c.logprobs and c.token_ids:
                        lp = sum(d.get(t, 0.0) if isinstance(
                            d.get(t, 0.0), float) else
                            getattr(d.get(t), "logprob", 0.0)
                            for d, t in zip(c.logprobs, c.token_ids))
                    if lp is not None and c.token_ids:
                        return lp / len(c.token_ids)
                    return 0.0

                group.sort(key=lambda t: -score(t[1]))
                kept.extend(g for g in group[: req.n])
            fan = [fan[gi] for gi, _ in kept]
            finals = [f for _, f in kept]
            out_index = []
            per_prompt_counter = {}
            for p_i, b, _, _ in fan:
                k = per_prompt_counter.get(p_i, 0)
                per_prompt_counter[p_i] = k + 1
                out_index.append(p_i * req.n + k)
        n_prompt = sum(len(f.prompt_token_ids)
                       for gi, f in enumerate(finals)
                       if fan[gi][1] == 0)
        n_gen = sum(len(f.outputs[0].token_ids) for f in finals)
        timer.on_finish(n_prompt, n_gen)
        state.num_prompt_tokens += n_prompt
        state.num_generation_tokens += n_gen
        cached = max(f.num_cached_tokens for f in finals)
        usage = UsageInfo(
            prompt_tokens=n_prompt,
            completion_tokens=n_gen,
            total_tokens=n_prompt + n_gen,
            prompt_tokens_details=({"cached_tokens": cached}
                                   if cached else None),
        )
        def _echo_logprobs(final, comp):
            tok = engine.tokenizer
            ids = final.prompt_token_ids
            tokens = [tok.decode([t]) for t in ids]
            token_logprobs = [None]
            top = [None]
            plps = final.prompt_logprobs or []
            for i, t in enumerate(ids[1:]):  # plps[i] -> prompt tok i+1
                d = plps[i] if i < len(plps) else {}
                token_logprobs.append(d.get(t))
                top.append({tok.decode([k]): lp for k, lp in d.items()})
            base = _completion_logprobs(comp) or {
                "tokens": [], "token_logprobs": [], "top_logprobs": []}
            return {"tokens": tokens + base["tokens"],
                    "token_logprobs":
                        token_logprobs + base["token_logprobs"],
                    "top_logprobs": top + base["top_logprobs"],
                    "text_offset": []}

        choices = []
        for out_i, final in enumerate(finals):
            comp = final.outputs[0]
            choices.append(CompletionChoice(
                index=out_index[out_i],
                text=(final.prompt or "") + comp.text if req.echo
                else comp.text,
                logprobs=(None if forced_lp
                          else _echo_logprobs(final, comp) if echo_lp
                          else _completion_logprobs(comp)),
                prompt_logprobs=(final.prompt_logprobs
                                 if req.prompt_logprobs is not None
                                 else None),
                finish_reason=comp.finish_reason,
                stop_reason=comp.stop_reason
                if isinstance(comp.stop_reason, (int, str)) else None,
            ))
        return CompletionResponse(
            id=rid, model=req.model, choices=choices, usage=usage)

    # ------------------------------------------------------------------
    @app.post("/v1/responses")
    async def responses(req: "ResponsesRequest", raw: Request):
        """OpenAI Responses API over the chat path: input items become
        chat messages (instructions -> system), output is one assistant
        message item. Streaming emits the typed response.* SSE events."""
        from vllm_amd.entrypoints.openai.protocol import random_id
        from vllm_amd.sampling_params import (
            RequestOutputKind, SamplingParams)

        msgs: list[ChatMessage] = []
        if req.instructions:
            msgs.append(ChatMessage(role="system",
                                    content=req.instructions))
        if req.previous_response_id:
            prev = state.responses_store.get(req.previous_response_id)
            if prev is None:
                return _error(
                    f"response {req.previous_response_id!r} not found",
                    404)
            for item in prev.get("output", []):
                if item.get("type") == "message":
                    msgs.append(ChatMessage(
                        role=item.get("role", "assistant"),
                        content="".join(
                            c.get("text", "")
                            for c in item.get("content", []))))
        if isinstance(req.input, str):
            msgs.append(ChatMessage(role="user", content=req.input))
        else:
            for item in req.input:
                content = item.get("content")
                if isinstance(content, list):
                    content = "".join(
                        seg.get("text", "") for seg in content
                        if seg.get("type") in ("input_text", "text",
                                               "output_text"))
                msgs.append(ChatMessage(role=item.get("role", "user"),
                                        content=content))
        prompt = apply_chat_template(engine.tokenizer, msgs, True)
        params = SamplingParams(
            temperature=req.temperature,
            top_p=req.top_p,
            max_tokens=req.max_output_tokens or state.max_model_len,
            output_kind=(RequestOutputKind.DELTA if req.stream
                         else RequestOutputKind.FINAL_ONLY),
        )
        rid = random_id("resp")
        msg_id = random_id("msg")
        state.num_requests += 1
        created = int(time.time())

        def envelope(status, output, usage=None):
            return {
                "id": rid, "object": "response", "created_at": created,
                "status": status, "model": req.model,
                "output": output, "metadata": req.metadata or {},
                **({"usage": usage} if usage else {}),
            }

        def msg_item(status, text):
            return {"type": "message", "id": msg_id, "role": "assistant",
                    "status": status,
                    "content": [{"type": "output_text", "text": text,
                                 "annotations": []}]}

        if req.background:
            if req.stream:
                return _error(
                    "background and stream are mutually exclusive")
            if not req.store:
                return _error("background requires store=true")
            state.responses_store[rid] = envelope("queued", [])

            async def run_background():
                state.responses_store[rid] = envelope("in_progress", [])
                try:
                    final = None
                    async for out in engine.generate(prompt, params, rid):
                        final = out
                    comp = final.outputs[0]
                    usage = {
                        "input_tokens": len(final.prompt_token_ids),
                        "output_tokens": len(comp.token_ids),
                        "total_tokens": len(final.prompt_token_ids)
                        + len(comp.token_ids)}
                    body = envelope(
                        "completed", [msg_item("completed", comp.text)],
                        usage)
                    body["output_text"] = comp.text
                    state.responses_store[rid] = body
                except Exception as e:  # noqa: BLE001
                    err = envelope("failed", [])
                    err["error"] = {"message": str(e)}
                    state.responses_store[rid] = err

            import asyncio as _asyncio

            task = _asyncio.get_running_loop().create_task(
                run_background())
            state.responses_tasks[rid] = task
            task.add_done_callback(
                lambda _t: state.responses_tasks.pop(rid, None))
            return envelope("queued", [])

        if req.stream:
            async def gen() -> AsyncGenerator[str, None]:
                def ev(etype, data):
                    return (f"event: {etype}\n"
                            f"data: {json.dumps(data)}\n\n")

                yield ev("response.created",
                         {"type": "response.created",
                          "response": envelope("in_progress", [])})
                yield ev("response.output_item.added",
                         {"type": "response.output_item.added",
                          "output_index": 0,
                          "item": msg_item("in_progress", "")})
                text = ""
                n_gen = 0
                n_prompt = 0
                try:
                    async for out in engine.generate(prompt, params, rid):
                        comp = out.outputs[0]
                        n_gen = len(comp.token_ids)
                        n_prompt = len(out.prompt_token_ids)
                        if comp.text:
                            text += comp.text
                            yield ev("response.output_text.delta",
                                     {"type": "response.output_text.delta",
                                      "item_id": msg_id,
                                      "output_index": 0,
                                      "content_index": 0,
                                      "delta": comp.text})
                    yield ev("response.output_text.done",
                             {"type": "response.output_text.done",
                              "item_id": msg_id, "output_index": 0,
                              "content_index": 0, "text": text})
                    yield ev("response.output_item.done",
                             {"type": "response.output_item.done",
                              "output_index": 0,
                              "item": msg_item("completed", text)})
                    usage = {"input_tokens": n_prompt,
                             "output_tokens": n_gen,
                             "total_tokens": n_prompt + n_gen}
                    yield ev("response.completed",
                             {"type": "response.completed",
                              "response": envelope(
                                  "completed",
                                  [msg_item("completed", text)], usage)})
                except Exception as e:  # noqa: BLE001
                    yield ev("error", {"type": "error",
                                       "message": str(e)})
            return StreamingResponse(gen(), media_type="text/event-stream")

        final = None
        try:
            async for out in engine.generate(prompt, params, rid):
                final = out
        except ValueError as e:
            return _error(str(e))
        comp = final.outputs[0]
        state.num_prompt_tokens += len(final.prompt_token_ids)
        state.num_generation_tokens += len(comp.token_ids)
        usage = {"input_tokens": len(final.prompt_token_ids),
                 "output_tokens": len(comp.token_ids),
                 "total_tokens": len(final.prompt_token_ids)
                 + len(comp.token_ids)}
        body = envelope("completed", [msg_item("completed", comp.text)],
                        usage)
        body["output_text"] = comp.text
        if req.store:
            state.responses_store[rid] = body
        return body

    @app.get("/v1/responses/{response_id}")
    async def get_response(response_id: str):
        body = state.responses_store.get(response_id)
        if body is None:
            return _error(f"response {response_id!r} not found", 404)
        return body

    @app.post("/v1/responses/{response_id}/cancel")
    async def cancel_response(response_id: str):
        body = state.responses_store.get(response_id)
        if body is None:
            return _error(f"response {response_id!r} not found", 404)
        if body.get("status") in ("queued", "in_progress"):
            await engine.abort(response_id)
            task = state.responses_tasks.get(response_id)
            if task is not None:
                task.cancel()
            body = dict(body)
            body["status"] = "cancelled"
            state.responses_store[response_id] = body
        return body

    # ------------------------------------------------------------------
    @app.post("/v1/audio/transcriptions")
    @app.post("/v1/audio/translations")
    async def audio_transcriptions(raw: Request):
        """Whisper-style STT (reference /v1/audio endpoints): accepts
        multipart/form-data with a WAV `file` (parsed in-process — no
        python-multipart in this image) or JSON {"file": base64-wav}.
        The decoded waveform rides the engine's multimodal dict-prompt
        path; the decoder generates over cross-attention."""
        spec = engine.config.model_config.spec
        if spec.audio_encoder_layers == 0:
            return _error(
                f"model {state.model_name} has no audio encoder")
        ctype = raw.headers.get("content-type", "")
        try:
            if ctype.startswith("multipart/form-data"):
                fields = _parse_multipart(await raw.body(), ctype)
                wav_bytes = fields.get("file")
                if wav_bytes is None:
                    return _error("missing file field")
                opts = {k: v.decode(errors="replace")
                        for k, v in fields.items() if k != "file"}
            else:
                import base64

                data = await raw.json()
                b64 = data.get("file")
                if not b64:
                    return _error("missing file field")
                wav_bytes = base64.b64decode(b64)
                opts = {k: str(v) for k, v in data.items()
                        if k != "file"}
            waveform = _decode_wav(wav_bytes)
        except ValueError as e:
            return _error(str(e))
        max_secs = ((spec.audio_max_frames * 2 - 1) * 160 + 400) / 16000
        if len(waveform) / 16000 > max_secs:
            return _error(
                f"audio longer than {max_secs:.0f}s is not supported")
        from vllm_amd.sampling_params import SamplingParams

        ids = [spec.bos_token_id]
        if opts.get("prompt"):
            ids += engine.tokenizer.encode(opts["prompt"],
                                           add_special_tokens=False)
        params = SamplingParams(
            temperature=float(opts.get("temperature", 0.0)),
            max_tokens=min(state.max_model_len - len(ids) - 1, 200),
        )
        rid = random_id("transcription")
        state.num_requests += 1
        final = None
        try:
            async for out in engine.generate(
                    {"prompt_token_ids": ids,
                     "multi_modal_data": {"audio": waveform}},
                    params, rid):
                final = out
        except ValueError as e:
            return _error(str(e))
        text = final.outputs[0].text
        if opts.get("response_format") == "text":
            from fastapi.responses import PlainTextResponse

            return PlainTextResponse(text)
        return {"text": text}

    @app.post("/v1/embeddings")
    async def embeddings(req: EmbeddingRequest):
        import asyncio
        import base64
        import struct

        from vllm_amd.sampling_params import SamplingParams

        inputs = req.input
        if isinstance(inputs, str):
            inputs = [inputs]
        elif inputs and isinstance(inputs[0], int):
            inputs = [inputs]
        params = SamplingParams(pooling=req.pooling, max_tokens=1)
        state.num_requests += len(inputs)

        async def one(prompt):
            final = None
            async for out in engine.generate(prompt, params,
                                             random_id("embd")):
                final = out
            return final

        try:
            finals = await asyncio.gather(*(one(p) for p in inputs))
        except ValueError as e:
            return _error(str(e))
        if req.dimensions is not None and req.dimensions < 1:
            return _error("dimensions must be >= 1")
        data = []
        n_prompt = 0
        for i, final in enumerate(finals):
            n_prompt += len(final.prompt_token_ids)
            vec = final.pooled or []
            if req.dimensions is not None and vec:
                vec = vec[:req.dimensions]
                norm = sum(v * v for v in vec) ** 0.5
                if norm > 0:
                    vec = [v / norm for v in vec]
            if req.encoding_format == "base64":
                emb = base64.b64encode(
                    struct.pack(f"<{len(vec)}f", *vec)).decode()
            else:
                emb = vec
            data.append(EmbeddingData(index=i, embedding=emb))
        state.num_prompt_tokens += n_prompt
        return EmbeddingResponse(
            data=data, model=req.model,
            usage=UsageInfo(prompt_tokens=n_prompt, total_tokens=n_prompt))

    async def _embed_many(prompts, pooling):
        import asyncio

        from vllm_amd.sampling_params import SamplingParams

        params = SamplingParams(pooling=pooling, max_tokens=1)

        async def one(prompt):
            final = None
            async for out in engine.generate(prompt, params,
                                             random_id("pool")):
                final = out
            return final

        return await asyncio.gather(*(one(p) for p in prompts))

    def _cosine(a, b):
        import math

        dot = sum(x * y for x, y in zip(a, b))
        na = math.sqrt(sum(x * x for x in a)) or 1e-12
        nb = math.sqrt(sum(x * x for x in b)) or 1e-12
        return dot / (na * nb)

    @app.post("/v1/score")
    async def score(req: ScoreRequest):
        """Similarity scores between text_1 and text_2 (embedding-model
        scoring: cosine of pooled hidden states; the reference's
        cross-encoder path needs a classifier-head model class)."""
        t1 = [req.text_1] if isinstance(req.text_1, str) else req.text_1
        t2 = [req.text_2] if isinstance(req.text_2, str) else req.text_2
        if len(t1) == 1 and len(t2) > 1:
            t1 = t1 * len(t2)
        if len(t1) != len(t2):
            return _error("text_1/text_2 length mismatch")
        finals = await _embed_many(t1 + t2, req.pooling)
        n = len(t1)
        n_prompt = sum(len(f.prompt_token_ids) for f in finals)
        data = [ScoreData(index=i,
                          score=_cosine(finals[i].pooled,
                                        finals[n + i].pooled))
                for i in range(n)]
        state.num_requests += 1
        return ScoreResponse(
            data=data, model=req.model,
            usage=UsageInfo(prompt_tokens=n_prompt,
                            total_tokens=n_prompt))

    @app.post("/rerank")
    @app.post("/v1/rerank")
    async def rerank(req: RerankRequest):
        finals = await _embed_many([req.query] + req.documents,
                                   req.pooling)
        qv = finals[0].pooled
        scored = sorted(
            ((i, _cosine(qv, f.pooled))
             for i, f in enumerate(finals[1:])),
            key=lambda x: -x[1])
        if req.top_n:
            scored = scored[:req.top_n]
        n_prompt = sum(len(f.prompt_token_ids) for f in finals)
        state.num_requests += 1
        return RerankResponse(
            model=req.model,
            results=[RerankResult(index=i,
                                  document={"text": req.documents[i]},
                                  relevance_score=s)
                     for i, s in scored],
            usage=UsageInfo(prompt_tokens=n_prompt,
                            total_tokens=n_prompt))

    @app.post("/pooling")
    async def pooling(req: EmbeddingRequest):
        return await embeddings(req)

    # ------------------------------------------------------------------
    @app.post("/v1/chat/completions")
    async def chat_completions(req: ChatCompletionRequest, raw: Request):
        if not 1 <= req.n <= 64:
            return _error("n must be in [1, 64]")
        from vllm_amd.entrypoints import tool_parser as tp

        tools_on = bool(req.tools) and req.tool_choice != "none"
        named = req.named_tool() if tools_on else None
        image_urls = [u for m in req.messages for u in m.image_urls()]
        audio_ins = [a for m in req.messages for a in m.audio_inputs()]
        if audio_ins:
            # OpenAI input_audio content parts (base64 WAV): the decoded
            # waveform rides the engine's multimodal dict-prompt path;
            # the templated text becomes the decoder prompt (no
            # placeholder tokens — the decoder cross-attends).
            spec = engine.config.model_config.spec
            if spec.audio_encoder_layers == 0:
                return _error(f"model {state.model_name} does not "
                              "support audio input")
            if len(audio_ins) > 1:
                return _error("at most one input_audio part per request")
            import base64

            try:
                waveform = _decode_wav(
                    base64.b64decode(audio_ins[0].get("data", "")))
            except Exception as e:  # noqa: BLE001
                return _error(f"could not decode input_audio: {e}")
            text_prompt = apply_chat_template(
                engine.tokenizer, req.messages, req.add_generation_prompt)
            prompt = {"prompt_token_ids":
                      engine.tokenizer.encode(text_prompt),
                      "multi_modal_data": {"audio": waveform}}
        elif image_urls:
            try:
                prompt = build_mm_chat_prompt(
                    engine, req.messages, req.add_generation_prompt,
                    image_urls,
                    tools=req.tools if tools_on and not named else None,
                )
            except ValueError as e:
                return _error(str(e))
        else:
            prompt = apply_chat_template(
                engine.tokenizer, req.messages, req.add_generation_prompt,
                tools=req.tools if tools_on and not named else None,
            )
        default_max = state.max_model_len
        try:
            params = req.to_sampling_params(req.stream, default_max)
        except ValueError as e:
            return _error(str(e))
        lora = req.model if req.model in state.lora_names else None
        state.num_requests += 1
        rid = random_id("chatcmpl")

        if req.stream:
            class _Branch:
                """Per-branch streaming parse state (reasoning + tool
                parsers, named-call bookkeeping, delta accounting)."""

                def __init__(self):
                    self.rparse = (tp.StreamingReasoningParser()
                                   if state.reasoning_parser else None)
                    self.tparse = (tp.make_streaming_tool_parser(
                                       state.tool_call_parser)
                                   if tools_on and not named else None)
                    self.named_id = tp._call_id() if named else None
                    self.named_first = True
                    self.seen_toks = 0

                def deltas(self, text, finish=None):
                    out = []
                    reasoning = ""
                    if self.rparse is not None:
                        reasoning, text = self.rparse.feed(text)
                        if finish is not None:
                            r2, t2 = self.rparse.flush()
                            reasoning += r2
                            text += t2
                    if named:
                        calls = []
                        if text:
                            fn = {"arguments": text}
                            if self.named_first:
                                fn["name"] = named
                                self.named_first = False
                                calls = [{"index": 0, "id": self.named_id,
                                          "type": "function",
                                          "function": fn}]
                            else:
                                calls = [{"index": 0, "function": fn}]
                        out.append(DeltaMessage(
                            reasoning_content=reasoning or None,
                            tool_calls=calls or None))
                        return out, bool(calls)
                    saw = False
                    if self.tparse is not None:
                        text, calls = self.tparse.feed(text)
                        if finish is not None:
                            t2, c2 = self.tparse.flush()
                            text += t2
                            calls += c2
                        saw = self.tparse.saw_tool_call
                        if calls:
                            out.append(DeltaMessage(tool_calls=calls))
                    if text or reasoning:
                        out.append(DeltaMessage(
                            content=text or None,
                            reasoning_content=reasoning or None))
                    return out, saw

            branches = _branch_params(params, req.n)

            async def gen() -> AsyncGenerator[str, None]:
                from vllm_amd.metrics import RequestTimer
                timer = RequestTimer(state.metrics)
                n_gen = 0
                n_prompt = 0
                states = [_Branch() for _ in branches]
                for b in range(len(branches)):
                    first = ChatCompletionStreamResponse(
                        id=rid, model=req.model,
                        choices=[ChatStreamChoice(
                            index=b, delta=DeltaMessage(role="assistant",
                                                        content=""))],
                    )
                    yield f"data: {first.model_dump_json()}\n\n"
                gens = [engine.generate(prompt, bp, f"{rid}-{b}",
                                        lora=lora)
                        for b, bp in enumerate(branches)]
                try:
                    async for b, out in _merge_streams(gens):
                        st = states[b]
                        comp = out.outputs[0]
                        prev_seen = st.seen_toks
                        new = len(comp.token_ids) - st.seen_toks
                        st.seen_toks = len(comp.token_ids)
                        state.num_generation_tokens += new
                        timer.on_tokens(new)
                        n_gen += new
                        n_prompt = len(out.prompt_token_ids)
                        msgs, saw = st.deltas(comp.text, comp.finish_reason)
                        finish = comp.finish_reason
                        if finish and (saw or named):
                            finish = "tool_calls" if finish == "stop" \
                                else finish
                        # Streamed logprobs for plain content (no tool/
                        # reasoning parsers buffering text — there the
                        # token<->delta alignment is undefined, same
                        # limitation as the reference).
                        lp_chunk = None
                        if (req.logprobs and new and st.rparse is None
                                and st.tparse is None and not named):
                            lp_chunk = _chat_logprobs(
                                comp, prev_seen, prev_seen + new)
                        if lp_chunk is not None and not msgs:
                            # Tokens that detokenize to nothing (byte
                            # fragments) still stream their logprobs.
                            msgs = [DeltaMessage()]
                        for i, d in enumerate(msgs):
                            last = comp.finish_reason and i == len(msgs) - 1
                            chunk = ChatCompletionStreamResponse(
                                id=rid, model=req.model,
                                choices=[ChatStreamChoice(
                                    index=b, delta=d,
                                    finish_reason=finish if last else None,
                                    logprobs=lp_chunk,
                                )],
                            )
                            lp_chunk = None  # attach once
                            yield f"data: {chunk.model_dump_json()}\n\n"
                        if comp.finish_reason and not msgs:
                            chunk = ChatCompletionStreamResponse(
                                id=rid, model=req.model,
                                choices=[ChatStreamChoice(
                                    index=b, delta=DeltaMessage(),
                                    finish_reason=finish)],
                            )
                            yield f"data: {chunk.model_dump_json()}\n\n"
                    timer.on_finish(n_prompt, n_gen)
                    if (req.stream_options or {}).get("include_usage"):
                        final_chunk = ChatCompletionStreamResponse(
                            id=rid, model=req.model, choices=[],
                            usage=UsageInfo(
                                prompt_tokens=n_prompt,
                                completion_tokens=n_gen,
                                total_tokens=n_prompt + n_gen))
                        yield f"data: {final_chunk.model_dump_json()}\n\n"
                    yield "data: [DONE]\n\n"
                except Exception as e:  # noqa: BLE001
                    err = {"error": {"message": str(e)}}
                    yield f"data: {json.dumps(err)}\n\n"
            return StreamingResponse(gen(), media_type="text/event-stream")

        import asyncio

        from vllm_amd.metrics import RequestTimer
        timer = RequestTimer(state.metrics)
        branches = _branch_params(params, req.n)

        async def run_branch(b, bp):
            final = None
            async for out in engine.generate(prompt, bp, f"{rid}-{b}",
                                             lora=lora):
                final = out
            return final

        try:
            finals = await asyncio.gather(
                *(run_branch(b, bp) for b, bp in enumerate(branches)))
        except ValueError as e:
            return _error(str(e))
        n_prompt = len(finals[0].prompt_token_ids)
        n_gen = sum(len(f.outputs[0].token_ids) for f in finals)
        timer.on_finish(n_prompt, n_gen)
        state.num_prompt_tokens += n_prompt
        state.num_generation_tokens += n_gen
        cached = max(f.num_cached_tokens for f in finals)
        usage = UsageInfo(
            prompt_tokens=n_prompt,
            completion_tokens=n_gen,
            total_tokens=n_prompt + n_gen,
            prompt_tokens_details=({"cached_tokens": cached}
                                   if cached else None),
        )
        choices = []
        for b, final in enumerate(finals):
            comp = final.outputs[0]
            text = comp.text
            reasoning = None
            if state.reasoning_parser:
                reasoning, text = tp.split_reasoning(text)
            tool_calls = None
            finish = comp.finish_reason or "stop"
            if named:
                call = tp.ParsedToolCall(id=tp._call_id(), name=named,
                                         arguments=text.strip())
                tool_calls = [call.as_openai(0)]
                text = None
                finish = "tool_calls"
            elif tools_on:
                text, calls = tp.parse_tool_calls(
                    state.tool_call_parser, text)
                if calls:
                    tool_calls = [c.as_openai(i)
                                  for i, c in enumerate(calls)]
                    if finish == "stop":
                        finish = "tool_calls"
            choices.append(ChatChoice(
                index=b,
                message=ChatCompletionMessage(
                    content=text if text else None,
                    reasoning_content=reasoning,
                    tool_calls=tool_calls),
                logprobs=_chat_logprobs(comp),
                finish_reason=finish,
            ))
        return ChatCompletionResponse(
            id=rid, model=req.model, choices=choices, usage=usage)

    return app


def make_server(engine_args: EngineArgs,
                served_model_name: Optional[str] = None,
                reasoning_parser: Optional[str] = None,
                api_key: Optional[str] = None,
                tool_call_parser: str = "hermes"):
    config = engine_args.create_engine_config()
    if config.parallel_config.data_parallel_size > 1:
        from vllm_amd.engine.async_llm import DPAsyncLLM

        engine = DPAsyncLLM(config)
    else:
        engine = AsyncLLM(config)
    state = ServerState(engine, served_model_name or engine_args.model,
                        reasoning_parser=reasoning_parser,
                        api_key=api_key,
                        tool_call_parser=tool_call_parser)
    return build_app(state), state


def main() -> None:
    import uvicorn

    parser = argparse.ArgumentParser(
        description="vllm_amd OpenAI-compatible server")
    parser.add_argument("--host", type=str, default="0.0.0.0")
    parser.add_argument("--port", type=int, default=8000)
    parser.add_argument("--served-model-name", type=str, default=None)
    # All listed reasoning models emit the same <think>...</think> span;
    # the aliases map to one splitter (reference vllm/reasoning/ keeps a
    # class per name over the same format).
    parser.add_argument("--reasoning-parser", type=str, default=None,
                        choices=["deepseek_r1", "qwen3", "glm45",
                                 "nemotron"])
    parser.add_argument("--tool-call-parser", type=str, default="hermes",
                        choices=["hermes", "qwen", "glm4", "mistral",
                                 "llama3_json", "pythonic", "granite",
                                 "internlm2"])
    parser.add_argument("--api-key", type=str, default=None,
                        help="require this bearer token on /v1 routes")
    parser.add_argument("--grpc-port", type=int, default=None,
                        help="also serve the gRPC Inference service "
                             "(entrypoints/grpc/inference.proto)")
    EngineArgs.add_cli_args(parser)
    args = parser.parse_args()
    engine_args = EngineArgs.from_cli_args(args)
    app, state = make_server(engine_args, args.served_model_name,
                             reasoning_parser=args.reasoning_parser,
                             api_key=args.api_key,
                             tool_call_parser=args.tool_call_parser)
    if args.grpc_port is not None:
        from vllm_amd.entrypoints.grpc.server import make_grpc_server

        @app.on_event("startup")
        async def _start_grpc() -> None:
            server, port = make_grpc_server(
                state.engine, state.lora_names,
                f"{args.host}:{args.grpc_port}")
            await server.start()
            app.state.grpc_server = server
    uvicorn.run(app, host=args.host, port=args.port, log_level="info")


if __name__ == "__main__":
    main()
