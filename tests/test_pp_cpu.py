"""Pipeline-parallel tests on CPU (gloo, world_size=2 = 2 PP stages).

Stage 0 owns the embedding + first half of the layers, stage 1 the rest
+ final norm + lm_head; one [T, hidden] activation crosses the boundary
per step and the sampled tokens broadcast back over the PP group. The
name-seeded dummy init makes pp=2 weights bit-identical to pp=1, so the
generated tokens must match the single-process run EXACTLY.
"""

import multiprocessing as mp
import os


def _baseline(q):
    try:
        from vllm_amd.entrypoints.llm import LLM
        from vllm_amd.sampling_params import SamplingParams

        llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=64, max_model_len=256,
                  max_num_batched_tokens=256, max_num_seqs=4)
        prompts = [[(i * 7 + j) % 900 + 3 for j in range(20)]
                   for i in range(3)]
        p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
        outs = llm.generate(prompts, p)
        llm.shutdown()
        q.put(("ok", [o.outputs[0].token_ids for o in outs]))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def _pp_worker(rank: int, world: int, port: int, q):
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
                                           pipeline_parallel_size=2,
                                           rank=rank, local_rank=rank,
                                           world_size=world,
                                           distributed_backend="gloo"),
            device_config=DeviceConfig(device="cpu"),
        )
        engine = EngineCore(config)
        if rank == 0:
            assert engine.is_driver
            # Stage 0 has the embedding but no lm_head.
            model = engine.worker.runner.model
            assert model.model.embed_tokens is not None
            assert model.lm_head is None
            prompts = [[(i * 7 + j) % 900 + 3 for j in range(20)]
                       for i in range(3)]
            p = SamplingParams(temperature=0.0, max_tokens=8,
                               ignore_eos=True)
            for i, toks in enumerate(prompts):
                engine.add_request(Request(
                    request_id=f"r{i}", prompt_token_ids=toks,
                    sampling_params=p))
            out_toks = {}
            while engine.has_unfinished_requests():
                for out in engine.step():
                    out_toks.setdefault(out.req_id, []).extend(
                        out.new_token_ids)
            engine.shutdown()
            q.put(("ok", [out_toks[f"r{i}"] for i in range(3)]))
        else:
            model = engine.worker.runner.model
            assert model.model.embed_tokens is None
            assert model.lm_head is not None
            # Stage 1 caches only its layer slice.
            n_local = len(engine.worker.runner.kv_caches)
            assert n_local == model.model.hi - model.model.lo
            engine.run_spmd_worker_loop()
            q.put(("ok", None))
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def test_pp2_matches_single_process():
    ctx = mp.get_context("spawn")
    q0 = ctx.Queue()
    pb = ctx.Process(target=_baseline, args=(q0,))
    pb.start()
    status, baseline = q0.get(timeout=180)
    pb.join(timeout=30)
    assert status == "ok", baseline

    q = ctx.Queue()
    procs = [ctx.Process(target=_pp_worker, args=(r, 2, 29631, q))
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
    pp_tokens = next(p for s, p in outs if p is not None)
    assert pp_tokens == baseline
    assert all(len(t) == 8 for t in pp_tokens)


def _tp_pp_worker(rank: int, world: int, port: int, q):
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
            parallel_config=ParallelConfig(tensor_parallel_size=2,
                                           pipeline_parallel_size=2,
                                           rank=rank, local_rank=rank,
                                           world_size=world,
                                           distributed_backend="gloo"),
            device_config=DeviceConfig(device="cpu"),
        )
        engine = EngineCore(config)
        if rank == 0:
            assert engine.is_driver
            prompts = [[(i * 7 + j) % 900 + 3 for j in range(20)]
                       for i in range(2)]
            p = SamplingParams(temperature=0.0, max_tokens=6,
                               ignore_eos=True)
            for i, toks in enumerate(prompts):
                engine.add_request(Request(
                    request_id=f"r{i}", prompt_token_ids=toks,
                    sampling_params=p))
            out_toks = {}
            while engine.has_unfinished_requests():
                for out in engine.step():
                    out_toks.setdefault(out.req_id, []).extend(
                        out.new_token_ids)
            engine.shutdown()
            q.put(("ok", [out_toks[f"r{i}"] for i in range(2)]))
        else:
            engine.run_spmd_worker_loop()
            q.put(("ok", None))
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def test_tp2_pp2_grid():
    """2x2 TP-by-PP grid (world 4): TP groups contiguous, PP strided;
    generation completes and is deterministic in shape."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tp_pp_worker, args=(r, 4, 29641, q))
             for r in range(4)]
    for p in procs:
        p.start()
    outs = []
    try:
        for _ in range(4):
            outs.append(q.get(timeout=240))
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
    for status, payload in outs:
        assert status == "ok", payload
    toks = next(p for s, p in outs if p is not None)
    assert all(len(t) == 6 for t in toks)


def test_pp_rejects_pooling_requests():
    """pooling/prompt_logprobs need the last stage's hidden/logits on the
    driver — LLMEngine.add_request rejects them under pp>1."""
    import pytest

    from vllm_amd.config import EngineConfig, ModelConfig, ParallelConfig
    from vllm_amd.engine.llm_engine import LLMEngine
    from vllm_amd.sampling_params import SamplingParams
    from vllm_amd.tokenizer import TokenizerWrapper

    eng = LLMEngine.__new__(LLMEngine)  # skip engine-core construction
    eng.config = EngineConfig(
        model_config=ModelConfig(model="tiny-llama", dtype="fp32"),
        parallel_config=ParallelConfig(pipeline_parallel_size=2,
                                       tensor_parallel_size=1,
                                       world_size=2))
    eng.tokenizer = TokenizerWrapper(None)
    eng._request_counter = 0
    for bad in (SamplingParams(pooling="last", max_tokens=1),
                SamplingParams(prompt_logprobs=2, max_tokens=1)):
        with pytest.raises(ValueError, match="pp=1"):
            eng.add_request(None, [3, 4, 5], bad)


def _jamba_baseline(q):
    try:
        from vllm_amd.entrypoints.llm import LLM
        from vllm_amd.sampling_params import SamplingParams

        llm = LLM(model="tiny-jamba", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=64, max_model_len=256,
                  max_num_batched_tokens=256, max_num_seqs=4)
        prompts = [[(i * 11 + j) % 900 + 3 for j in range(20)]
                   for i in range(3)]
        p = SamplingParams(temperature=0.0, max_tokens=8, ignore_eos=True)
        outs = llm.generate(prompts, p)
        llm.shutdown()
        q.put(("ok", [o.outputs[0].token_ids for o in outs]))
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def _jamba_pp_worker(rank: int, world: int, port: int, q):
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
            model_config=ModelConfig(model="tiny-jamba", dtype="fp32",
                                     max_model_len=256),
            cache_config=CacheConfig(block_size=16, num_gpu_blocks=64,
                                     enable_prefix_caching=False),
            scheduler_config=SchedulerConfig(max_num_batched_tokens=256,
                                             max_num_seqs=4),
            parallel_config=ParallelConfig(tensor_parallel_size=1,
                                           pipeline_parallel_size=2,
                                           rank=rank, local_rank=rank,
                                           world_size=world,
                                           distributed_backend="gloo"),
            device_config=DeviceConfig(device="cpu"),
        )
        engine = EngineCore(config)
        runner = engine.worker.runner
        # tiny-jamba: 4 layers, pattern [mamba, attn, mamba, attn] —
        # each 2-layer stage holds exactly ONE paged-KV layer and ONE
        # SSM state layer.
        assert len(runner.kv_caches) == 1
        assert runner.mamba_conv.shape[0] == 1
        if rank == 0:
            prompts = [[(i * 11 + j) % 900 + 3 for j in range(20)]
                       for i in range(3)]
            p = SamplingParams(temperature=0.0, max_tokens=8,
                               ignore_eos=True)
            for i, toks in enumerate(prompts):
                engine.add_request(Request(
                    request_id=f"r{i}", prompt_token_ids=toks,
                    sampling_params=p))
            out_toks = {}
            while engine.has_unfinished_requests():
                for out in engine.step():
                    out_toks.setdefault(out.req_id, []).extend(
                        out.new_token_ids)
            engine.shutdown()
            q.put(("ok", [out_toks[f"r{i}"] for i in range(3)]))
        else:
            engine.run_spmd_worker_loop()
            q.put(("ok", None))
        import torch.distributed as dist

        if dist.is_initialized():
            dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        import traceback

        q.put(("err", f"{e}\n{traceback.format_exc()}"))


def test_pp2_jamba_hybrid_matches_single_process():
    """PP over the attention+SSM hybrid: each stage allocates its own
    slice of BOTH cache kinds; pp2 greedy tokens == single-process."""
    ctx = mp.get_context("spawn")
    q0 = ctx.Queue()
    pb = ctx.Process(target=_jamba_baseline, args=(q0,))
    pb.start()
    status, baseline = q0.get(timeout=180)
    pb.join(timeout=30)
    assert status == "ok", baseline

    q = ctx.Queue()
    procs = [ctx.Process(target=_jamba_pp_worker, args=(r, 2, 29637, q))
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
    pp_tokens = next(p for s, p in outs if p is not None)
    assert pp_tokens == baseline
    assert all(len(t) == 8 for t in pp_tokens)
