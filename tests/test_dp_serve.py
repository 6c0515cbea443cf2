"""Serve-level data-parallel replicas (DPAsyncLLM) on CPU.

Reference: vllm --data-parallel-size N (vllm/v1/engine/core_client.py
DPLBAsyncMPClient) — N engine replicas behind one API server with
least-loaded routing. Here: two CPU replicas, requests spread across
both, aggregated stats, health fan-out.
"""

import asyncio

import pytest
from fastapi.testclient import TestClient

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.engine.async_llm import DPAsyncLLM
from vllm_amd.entrypoints.openai.api_server import make_server
from vllm_amd.sampling_params import SamplingParams


def _args(dp: int = 2) -> EngineArgs:
    return EngineArgs(
        model="tiny-llama",
        dtype="fp32",
        device="cpu",
        block_size=16,
        num_gpu_blocks=256,
        max_model_len=512,
        max_num_batched_tokens=512,
        max_num_seqs=8,
        data_parallel_size=dp,
    )


def test_dp_router_spreads_and_finishes():
    eng = DPAsyncLLM(_args().create_engine_config())
    try:
        assert len(eng.replicas) == 2
        # Distinct rendezvous ports / device offsets per replica.
        pcs = [r.config.parallel_config for r in eng.replicas]
        assert pcs[0].worker_port != pcs[1].worker_port
        assert pcs[0].device_offset == 0 and pcs[1].device_offset == 1

        async def run():
            picked = []

            async def one(i):
                outs = []
                gen = eng.generate(
                    f"prompt number {i} with a few tokens",
                    SamplingParams(max_tokens=8, temperature=0.0,
                                   ignore_eos=True))
                async for out in gen:
                    outs.append(out)
                return outs[-1]

            # Launch concurrently so the in-flight counters overlap and
            # the least-loaded pick alternates replicas.
            t0 = asyncio.ensure_future(one(0))
            await asyncio.sleep(0.01)
            picked.append(list(eng._in_flight))
            t1 = asyncio.ensure_future(one(1))
            await asyncio.sleep(0.01)
            picked.append(list(eng._in_flight))
            r0, r1 = await asyncio.gather(t0, t1)
            return picked, r0, r1

        picked, r0, r1 = asyncio.new_event_loop().run_until_complete(run())
        assert r0.finished and r1.finished
        assert len(r0.outputs[0].token_ids) == 8
        assert len(r1.outputs[0].token_ids) == 8
        # While both were in flight, each replica held one request.
        assert picked[1] == [1, 1], picked
        # Counters drain back to zero.
        assert eng._in_flight == [0, 0]

        st = eng.stats()
        assert st["dp_in_flight"] == [0, 0]
        assert st["kv_blocks_total"] == 512  # 256 per replica, summed
        eng.check_health()
    finally:
        eng.shutdown()


def test_dp_api_server_end_to_end():
    app, state = make_server(_args(), served_model_name="tiny-llama")
    assert isinstance(state.engine, DPAsyncLLM)
    with TestClient(app) as c:
        for i in range(4):
            r = c.post("/v1/completions", json={
                "model": "tiny-llama",
                "prompt": f"dp request {i}",
                "max_tokens": 4,
                "temperature": 0.0,
                "ignore_eos": True,
            })
            assert r.status_code == 200, r.text
            assert r.json()["usage"]["completion_tokens"] == 4
        assert c.get("/health").status_code == 200
    state.engine.shutdown()


def test_dp_router_skips_dead_replica():
    """A replica whose engine died is routed around — requests keep
    succeeding on the survivors."""
    import asyncio

    dp = DPAsyncLLM(_args(2).create_engine_config())
    try:
        # Kill replica 0's engine loop.
        dp.replicas[0]._engine_error = RuntimeError("injected death")
        p = SamplingParams(max_tokens=4, temperature=0.0,
                           ignore_eos=True)

        async def drive():
            for i in range(4):
                out = None
                async for o in dp.generate([5 + i, 6, 7], p, f"r{i}"):
                    out = o
                assert len(out.outputs[0].token_ids) == 4
        asyncio.run(drive())
    finally:
        dp.shutdown()
