"""EPLB (expert-parallel load balancing) tests: gloo world_size=2.

Checks (role of the reference's vllm/distributed/eplb/eplb_state.py):
- rebalance() packs experts onto ranks by EWMA load (balanced, every
  rank computes the identical plan with no control sync),
- weights MOVE with their experts (forward outputs unchanged after an
  assignment change),
- the engine path triggers rebalancing via --eplb-window counters.
"""

import multiprocessing as mp
import os

import pytest


def _eplb_worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    try:
        import torch

        torch.set_num_threads(1)
        torch.manual_seed(0)
        from vllm_amd.layers.fused_moe import FusedMoE
        from vllm_amd.parallel.state import (destroy_distributed,
                                             init_distributed)

        init_distributed(tensor_parallel_size=world, backend="gloo",
                         rank=rank, world_size=world)
        E, K, H, I = 8, 2, 32, 64
        moe = FusedMoE(num_experts=E, top_k=K, hidden_size=H,
                       intermediate_size=I, dtype=torch.float32,
                       enable_expert_parallel=True, eplb_window=0)
        g = torch.Generator().manual_seed(7)
        w1 = torch.randn(E, I, H, generator=g) * 0.2
        w3 = torch.randn(E, I, H, generator=g) * 0.2
        w2 = torch.randn(E, H, I, generator=g) * 0.2
        moe.load_full_weights(w1, w3, w2)
        moe.gate.weight.data.copy_(
            torch.randn(E, H, generator=g) * 0.5)

        x = torch.randn(16, H, generator=g)
        out0 = moe(x)

        # Heavily skewed synthetic load: experts 0..3 hot. The greedy
        # plan must split hot experts across ranks.
        moe.eplb_load.copy_(torch.tensor(
            [100, 90, 80, 70, 1, 1, 1, 1], dtype=torch.int64))
        old_assignment = list(moe.assignment)
        moe.rebalance()
        assert moe.assignment != old_assignment, "plan should change"
        # Hot experts 0 and 1 must land on different ranks.
        assert moe.assignment[0] != moe.assignment[1]
        # Balanced slot counts.
        from collections import Counter
        counts = Counter(moe.assignment)
        assert counts[0] == counts[1] == E // world

        out1 = moe(x)
        assert torch.allclose(out0, out1, atol=1e-5, rtol=1e-5), (
            (out0 - out1).abs().max())

        # Second rebalance with a different skew: weights keep moving
        # correctly (slots shuffle within and across ranks).
        moe._eplb_ewma = None  # fresh EWMA for the new skew
        moe.eplb_load.copy_(torch.tensor(
            [1, 1, 1, 1, 50, 60, 70, 80], dtype=torch.int64))
        moe.rebalance()
        out2 = moe(x)
        assert torch.allclose(out0, out2, atol=1e-5, rtol=1e-5)

        destroy_distributed()
        q.put(("ok", None))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def test_eplb_rebalance_world2():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_eplb_worker, args=(r, 2, 2961, q))
             for r in range(2)]
    for p in procs:
        p.start()
    for _ in range(2):
        status, payload = q.get(timeout=180)
        assert status == "ok", payload
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.kill()


def test_plan_assignment_deterministic_and_balanced():
    """Plan math alone (no distributed): deterministic, capacity-bound."""
    import torch

    from vllm_amd.layers.fused_moe import FusedMoE

    moe = FusedMoE.__new__(FusedMoE)
    moe.num_experts = 6
    moe.ep_size = 3
    moe.num_local_experts = 2
    load = [10.0, 9.0, 8.0, 1.0, 1.0, 1.0]
    p1 = moe._plan_assignment(load)
    p2 = moe._plan_assignment(load)
    assert p1 == p2
    from collections import Counter
    assert all(c == 2 for c in Counter(p1).values())
    # The three hot experts are spread across all three ranks.
    assert len({p1[0], p1[1], p1[2]}) == 3


def _engine_worker(rank: int, world: int, port: int, q, eplb_window: int):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    try:
        import torch

        torch.set_num_threads(1)
        from vllm_amd.config import (CacheConfig, DeviceConfig, EngineConfig,
                                     ModelConfig, ParallelConfig,
                                     SchedulerConfig)
        from vllm_amd.engine.core import EngineCore
        from vllm_amd.request import Request
        from vllm_amd.sampling_params import SamplingParams

        config = EngineConfig(
            model_config=ModelConfig(model="tiny-mixtral", dtype="fp32",
                                     max_model_len=128),
            cache_config=CacheConfig(block_size=16, num_gpu_blocks=64),
            scheduler_config=SchedulerConfig(max_num_batched_tokens=128,
                                             max_num_seqs=2),
            parallel_config=ParallelConfig(tensor_parallel_size=world,
                                           enable_expert_parallel=True,
                                           eplb_window=eplb_window,
                                           distributed_backend="gloo"),
            device_config=DeviceConfig(device="cpu"),
        )
        engine = EngineCore(config)
        toks = None
        if rank == 0:
            engine.add_request(Request(
                request_id="r1",
                prompt_token_ids=[3 + i for i in range(24)],
                sampling_params=SamplingParams(
                    temperature=0.0, max_tokens=16, ignore_eos=True)))
            toks = []
            while engine.has_unfinished_requests():
                for out in engine.step():
                    toks.extend(out.new_token_ids)
            engine.shutdown()
        else:
            engine.run_spmd_worker_loop()
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
        q.put(("ok", toks))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


@pytest.mark.parametrize("port", [2966])
def test_engine_eplb_identical_outputs(port):
    """EP=2 engine with --eplb-window 2 (rebalances mid-generation) must
    produce exactly the tokens of the eplb-off engine."""
    results = {}
    for window, prt in ((0, port), (2, port + 1)):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        procs = [ctx.Process(target=_engine_worker,
                             args=(r, 2, prt, q, window))
                 for r in range(2)]
        for p in procs:
            p.start()
        toks = None
        for _ in range(2):
            status, payload = q.get(timeout=240)
            assert status == "ok", payload
            if payload is not None:
                toks = payload
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.kill()
        results[window] = toks
    assert results[0] == results[2]
    assert len(results[0]) == 16
