"""gRPC serving tests (reference surface: vllm grpc_server.py +
rust/proto/inference.proto): real grpc.aio server + channel on
localhost, dynamic-descriptor protobuf messages."""

import asyncio

import grpc
import pytest

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.grpc.server import MSG, make_grpc_server

PKG = "vllm_amd.inference"


@pytest.fixture(scope="module")
def engine():
    from vllm_amd.engine.async_llm import AsyncLLM

    args = EngineArgs(
        model="tiny-llama", dtype="fp32", device="cpu", block_size=16,
        num_gpu_blocks=128, max_model_len=256,
        max_num_batched_tokens=128, max_num_seqs=4,
    )
    eng = AsyncLLM(args.create_engine_config())
    yield eng
    eng.shutdown()


def _run(engine, coro_fn):
    async def main():
        server, port = make_grpc_server(engine)
        await server.start()
        try:
            async with grpc.aio.insecure_channel(
                    f"127.0.0.1:{port}") as ch:
                return await coro_fn(ch)
        finally:
            await server.stop(1.0)

    return asyncio.run(main())


def test_grpc_health_and_generate(engine):
    async def scenario(ch):
        health = ch.unary_unary(
            f"/{PKG}.Inference/Health",
            request_serializer=MSG["HealthRequest"].SerializeToString,
            response_deserializer=MSG["HealthResponse"].FromString)
        hr = await health(MSG["HealthRequest"]())
        assert hr.ok

        gen = ch.unary_stream(
            f"/{PKG}.Inference/Generate",
            request_serializer=MSG["GenerateRequest"].SerializeToString,
            response_deserializer=MSG["GenerateChunk"].FromString)
        req = MSG["GenerateRequest"](
            prompt_token_ids=[5, 9, 13, 17, 21],
            sampling=MSG["SamplingOptions"](
                temperature=0.0, max_tokens=6, ignore_eos=True))
        toks = []
        finish = None
        async for chunk in gen(req):
            toks.extend(chunk.token_ids)
            if chunk.finish_reason:
                finish = chunk.finish_reason
        return toks, finish

    toks, finish = _run(engine, scenario)
    assert len(toks) == 6
    assert finish == "length"


def test_grpc_embed(engine):
    async def scenario(ch):
        embed = ch.unary_unary(
            f"/{PKG}.Inference/Embed",
            request_serializer=MSG["EmbedRequest"].SerializeToString,
            response_deserializer=MSG["EmbedResponse"].FromString)
        return await embed(MSG["EmbedRequest"](
            prompt_token_ids=[7, 8, 9, 10]))

    r = _run(engine, scenario)
    assert len(r.values) == 128  # hidden_size of tiny-llama
    assert r.prompt_tokens == 4


def test_proto_file_matches_descriptor():
    """The committed .proto must declare every message the runtime
    descriptor builds (clients codegen from the file)."""
    from pathlib import Path

    proto = (Path(__file__).parent.parent / "vllm_amd" / "entrypoints"
             / "grpc" / "inference.proto").read_text()
    for name in MSG:
        assert f"message {name}" in proto, name
    assert "service Inference" in proto
