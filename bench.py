"""Flagship serving benchmark (driver contract).

Measures whole-node output tokens/sec on the BASELINE.json headline
config: Llama-3-8B bf16, synthetic random prompts of the CI benchmark
shape (random 128-token inputs), random-init (dummy) weights, continuous
decode at a fixed running batch.

    python bench.py --gpus N --steps K --warmup W

For N>1 the driver launches this under torch.distributed.run with one
rank per GPU; ranks form a TP=N group over RCCL/xGMI (the reference's
headline TP configs — BASELINE.md). One "step" is one engine iteration:
after warmup (which absorbs the prefill of the whole batch) every step
is a full-batch decode producing `batch` output tokens.

Timing: barrier + torch.cuda.synchronize on both sides of exactly
--steps engine steps; elapsed is MAX over ranks; rank 0 prints one JSON
line.
"""

from __future__ import annotations

import argparse
import json
import os
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=32)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", type=str, default="llama-3-8b")
    p.add_argument("--batch", type=int, default=1024,
                   help="running batch (= concurrent sequences)")
    p.add_argument("--input-len", type=int, default=128,
                   help="synthetic prompt length (CI 'random 128' shape)")
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--block-size", type=int, default=64)
    p.add_argument("--num-gpu-blocks", type=int, default=None,
                   help="skip the profiling pass with a fixed KV pool size")
    p.add_argument("--kv-cache-dtype", type=str, default="auto",
                   choices=["auto", "fp8"],
                   help="fp8 halves KV bytes (attention-bound large-batch)")
    p.add_argument("--quantization", type=str, default=None,
                   choices=["fp8"],
                   help="fp8 W8A8 dense linears (~2x MFMA rate on gfx950)")
    p.add_argument("--parallelism", type=str, default="tp",
                   choices=["dp", "tp"],
                   help="tp (default): one engine, weights sharded over all "
                        "ranks — strong scaling over xGMI (RCCL + custom "
                        "IPC all-reduce; the 70B layout, and the honest "
                        "multi-GPU measurement). dp: one independent engine "
                        "replica per rank, each with its own --batch (weak "
                        "scaling, xGMI-free — the throughput-optimal "
                        "serving layout for an 8B model)")
    return p.parse_args()


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dp_mode = args.parallelism == "dp" and world > 1
    on_gpu = torch.cuda.is_available()
    if not on_gpu and args.model == "llama-3-8b":
        # CPU plumbing check only (no GPU in the build container).
        args.model = "tiny-llama"
        args.dtype = "fp32"
        args.batch = min(args.batch, 8)
        args.steps = min(args.steps, 8)
        args.block_size = 16

    from vllm_amd.config import (
        CacheConfig, DeviceConfig, EngineConfig, ModelConfig,
        ParallelConfig, SchedulerConfig,
    )
    from vllm_amd.engine.core import EngineCore
    from vllm_amd.request import Request
    from vllm_amd.sampling_params import SamplingParams

    prefill_tokens = args.batch * args.input_len
    config = EngineConfig(
        model_config=ModelConfig(
            model=args.model, dtype=args.dtype,
            max_model_len=args.input_len + args.warmup + args.steps + 16,
            load_format="dummy",
            quantization=args.quantization,
        ),
        cache_config=CacheConfig(
            block_size=args.block_size,
            num_gpu_blocks=args.num_gpu_blocks,
            enable_prefix_caching=False,
            kv_cache_dtype=args.kv_cache_dtype,
        ),
        scheduler_config=SchedulerConfig(
            max_num_batched_tokens=max(prefill_tokens, 8192),
            max_num_seqs=args.batch,
            enable_chunked_prefill=True,
        ),
        parallel_config=ParallelConfig(
            tensor_parallel_size=1 if dp_mode else world,
            rank=rank, local_rank=local_rank, world_size=world),
        device_config=DeviceConfig(device="cuda" if on_gpu else "cpu"),
    )

    engine = EngineCore(config)
    # dp: every rank drives its own engine replica over its own batch.
    is_driver = engine.is_driver

    def sync():
        if world > 1:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    total_steps = args.warmup + args.steps
    if is_driver:
        rng = np.random.default_rng(0)
        params = SamplingParams(
            temperature=0.0,
            max_tokens=total_steps + 8,
            ignore_eos=True,
            detokenize=False,
        )
        for i in range(args.batch):
            toks = rng.integers(
                16, config.model_config.spec.vocab_size - 16,
                size=args.input_len,
            ).tolist()
            engine.add_request(Request(
                request_id=f"r{i}", prompt_token_ids=toks,
                sampling_params=params, arrival_time=time.time(),
            ))

        for _ in range(args.warmup):
            engine.step()
        sync()
        t0 = time.perf_counter()
        out_tokens = 0
        for _ in range(args.steps):
            for out in engine.step():
                out_tokens += len(out.new_token_ids)
        sync()
        t1 = time.perf_counter()
        elapsed = t1 - t0
    else:
        out_tokens = 0
        for _ in range(args.warmup):
            engine.step_worker()
        sync()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            engine.step_worker()
        sync()
        t1 = time.perf_counter()
        elapsed = t1 - t0

    if world > 1:
        # MAX elapsed over ranks; SUM of sampled tokens (dp: every rank
        # generated its own batch; tp: only rank 0 counted).
        dev = "cuda" if on_gpu else "cpu"
        t = torch.tensor([elapsed], dtype=torch.float64, device=dev)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
        n = torch.tensor([out_tokens], dtype=torch.float64, device=dev)
        torch.distributed.all_reduce(n, op=torch.distributed.ReduceOp.SUM)
        out_tokens = int(n.item())

    if rank == 0:
        # Count ACTUAL sampled tokens in the timed region (robust to
        # prefill chunks spilling past warmup).
        result = {
            "metric": "output tokens/sec (whole node)",
            "value": round(out_tokens / elapsed, 2),
            "unit": "tokens/s",
            "n_gpus": world if on_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "weak" if dp_mode else "strong",
            "vs_baseline": None,
            "dtype": (args.dtype if not args.quantization
                      else f"{args.dtype}+w8a8-fp8"),
            "data": "synthetic (random 128-token prompts, random-init "
                    "dummy weights; no network)",
            "config": {
                "model": args.model,
                "global_batch": args.batch * (world if dp_mode else 1),
                "seq_len": args.input_len,
                "parallelism": f"{args.parallelism}{world}"
                if world > 1 else "tp1",
            },
        }
        print(json.dumps(result))
    if is_driver:
        engine.shutdown()  # drains the async-pending step (broadcasts)
    else:
        engine.run_spmd_worker_loop()  # consume drain + stop sentinel
    if world > 1:
        torch.distributed.barrier()
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
