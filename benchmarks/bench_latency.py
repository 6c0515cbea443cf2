"""Offline end-to-end batch latency benchmark (role of the reference's
`vllm bench latency`, vllm/benchmarks/latency.py): time a fixed batch of
synthetic prompts through the engine for N iterations and report
latency percentiles.

Usage: python -m vllm_amd bench latency --model llama-3-8b \
           --input-len 32 --output-len 128 --batch-size 8
"""

from __future__ import annotations

import argparse
import json
import time

import numpy as np


def add_args(parser: argparse.ArgumentParser) -> None:
    parser.add_argument("--model", default="llama-3-8b")
    parser.add_argument("--dtype", default=None)
    parser.add_argument("--device", default=None)
    parser.add_argument("--input-len", type=int, default=32)
    parser.add_argument("--output-len", type=int, default=128)
    parser.add_argument("--batch-size", type=int, default=8)
    parser.add_argument("--num-iters-warmup", type=int, default=3)
    parser.add_argument("--num-iters", type=int, default=10)
    parser.add_argument("--quantization", default=None)
    parser.add_argument("--kv-cache-dtype", default=None)
    parser.add_argument("--max-num-seqs", type=int, default=None)
    parser.add_argument("--num-gpu-blocks", type=int, default=None)
    parser.add_argument("--output-json", default=None)
    parser.add_argument("--seed", type=int, default=0)


def _make_llm(args):
    from vllm_amd.entrypoints.llm import LLM

    kw = {}
    for name in ("dtype", "device", "quantization", "kv_cache_dtype",
                 "max_num_seqs", "num_gpu_blocks"):
        v = getattr(args, name)
        if v is not None:
            kw[name] = v
    need = args.batch_size * (args.input_len + args.output_len)
    return LLM(model=args.model, max_model_len=max(
        2048, args.input_len + args.output_len + 64),
        max_num_batched_tokens=max(2048, need), **kw)


def _prompts(args, vocab_size: int):
    rng = np.random.default_rng(args.seed)
    lo, hi = 10, max(11, vocab_size - 100)
    return [{"prompt_token_ids":
             rng.integers(lo, hi, size=args.input_len).tolist()}
            for _ in range(args.batch_size)]


def run(args) -> dict:
    from vllm_amd.sampling_params import SamplingParams

    llm = _make_llm(args)
    vocab = llm.engine.config.model_config.spec.vocab_size
    params = SamplingParams(max_tokens=args.output_len, temperature=0.0,
                            ignore_eos=True)

    def one_iter() -> float:
        # Fresh token ids per iteration so prefix caching cannot shrink
        # the measured prefill (reference latency bench semantics).
        prompts = _prompts(args, vocab)
        args.seed += 1
        t0 = time.perf_counter()
        outs = llm.generate(prompts, params)
        dt = time.perf_counter() - t0
        assert all(len(o.outputs[0].token_ids) == args.output_len
                   for o in outs)
        return dt

    for _ in range(args.num_iters_warmup):
        one_iter()
    lat = np.array([one_iter() for _ in range(args.num_iters)])
    llm.shutdown()
    result = {
        "avg_latency": float(lat.mean()),
        "latencies": lat.tolist(),
        "percentiles": {str(p): float(np.percentile(lat, p))
                        for p in (10, 25, 50, 75, 90, 99)},
        "batch_size": args.batch_size,
        "input_len": args.input_len,
        "output_len": args.output_len,
        "model": args.model,
    }
    return result


def main(argv=None) -> None:
    parser = argparse.ArgumentParser(description=__doc__)
    add_args(parser)
    args = parser.parse_args(argv)
    result = run(args)
    print(f"Avg latency: {result['avg_latency']:.4f} s "
          f"(batch {args.batch_size}, {args.input_len}+{args.output_len} "
          "tokens)")
    for p, v in result["percentiles"].items():
        print(f"  p{p}: {v:.4f} s")
    print(json.dumps(result))
    if args.output_json:
        with open(args.output_json, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
