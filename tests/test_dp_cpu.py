"""Data-parallel engine replicas on CPU (gloo, world_size=2, tp=1).

With tensor_parallel_size < world_size, each TP group (here: each rank)
runs an INDEPENDENT engine — its own scheduler, KV pool and batch; the
only cross-replica communication is outside the engine (bench.py's
timing all-reduce). Role of the reference's DP engine replicas
(vllm/v1/engine/coordinator.py + data_parallel_size), re-designed as
SPMD over torchrun: rank == replica.
"""

import multiprocessing as mp
import os

import pytest


def _dp_worker(rank: int, world: int, port: int, q):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        import torch

        torch.set_num_threads(1)
        from vllm_amd.config import (
            CacheConfig, DeviceConfig, EngineConfig, ModelConfig,
            ParallelConfig, SchedulerConfig,
        )
        from vllm_amd.engine.core import EngineCore
        from vllm_amd.request import Request
        from vllm_amd.sampling_params import SamplingParams

        config = EngineConfig(
            model_config=ModelConfig(model="tiny-llama", dtype="fp32",
                                     max_model_len=256),
            cache_config=CacheConfig(block_size=16, num_gpu_blocks=64),
            scheduler_config=SchedulerConfig(max_num_batched_tokens=256,
                                             max_num_seqs=4),
            parallel_config=ParallelConfig(tensor_parallel_size=1,
                                           rank=rank, local_rank=rank,
                                           world_size=world,
                                           distributed_backend="gloo"),
            device_config=DeviceConfig(device="cpu"),
        )
        engine = EngineCore(config)
        # Every rank must be a driver of its own replica.
        assert engine.is_driver
        params = SamplingParams(temperature=0.0, max_tokens=8,
                                ignore_eos=True)
        # Each replica gets DIFFERENT prompts — no lockstep required.
        for i in range(2 + rank):
            engine.add_request(Request(
                request_id=f"r{i}",
                prompt_token_ids=[rank * 100 + i * 7 + j + 3
                                  for j in range(10)],
                sampling_params=params,
            ))
        toks = {}
        while engine.has_unfinished_requests():
            for out in engine.step():
                toks.setdefault(out.req_id, []).extend(out.new_token_ids)
        engine.shutdown()
        # The global group still works for cross-replica aggregation.
        import torch.distributed as dist

        t = torch.tensor([float(len(toks))])
        dist.all_reduce(t)
        q.put(("ok", (rank, len(toks), int(t.item()))))
        dist.barrier()
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def test_dp2_replicas_cpu():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dp_worker, args=(r, 2, 29621, q))
             for r in range(2)]
    for p in procs:
        p.start()
    outs = []
    try:
        for _ in range(2):
            outs.append(q.get(timeout=180))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    for status, payload in outs:
        assert status == "ok", payload
    by_rank = {p[0]: p for _, p in outs}
    assert by_rank[0][1] == 2 and by_rank[1][1] == 3  # own request counts
    assert by_rank[0][2] == by_rank[1][2] == 5        # summed across dp


def test_ep_requires_full_world_tp():
    from vllm_amd.config import ParallelConfig

    with pytest.raises(ValueError):
        ParallelConfig(tensor_parallel_size=1, world_size=2,
                       enable_expert_parallel=True)
