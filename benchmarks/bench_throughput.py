"""Offline throughput benchmark (role of the reference's
`vllm bench throughput`, vllm/benchmarks/throughput.py): push a set of
synthetic requests through the engine at once — continuous batching
packs them — and report requests/s and tokens/s.

Usage: python -m vllm_amd bench throughput --model llama-3-8b \
           --num-prompts 256 --input-len 128 --output-len 128
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
    parser.add_argument("--num-prompts", type=int, default=256)
    parser.add_argument("--input-len", type=int, default=128)
    parser.add_argument("--output-len", type=int, default=128)
    # 0 = fixed lengths; r in (0,1] draws each length uniformly from
    # [len*(1-r), len] (the reference's random-dataset range ratio).
    parser.add_argument("--random-range-ratio", type=float, default=0.0)
    parser.add_argument("--quantization", default=None)
    parser.add_argument("--kv-cache-dtype", default=None)
    parser.add_argument("--max-num-seqs", type=int, default=None)
    parser.add_argument("--num-gpu-blocks", type=int, default=None)
    parser.add_argument("--output-json", default=None)
    parser.add_argument("--seed", type=int, default=0)


def _lengths(base: int, n: int, ratio: float, rng) -> list[int]:
    if ratio <= 0:
        return [base] * n
    lo = max(1, int(base * (1.0 - ratio)))
    return rng.integers(lo, base + 1, size=n).tolist()


def run(args) -> dict:
    from vllm_amd.entrypoints.llm import LLM
    from vllm_amd.sampling_params import SamplingParams

    kw = {}
    for name in ("dtype", "device", "quantization", "kv_cache_dtype",
                 "max_num_seqs", "num_gpu_blocks"):
        v = getattr(args, name)
        if v is not None:
            kw[name] = v
    llm = LLM(model=args.model,
              max_model_len=max(2048, args.input_len + args.output_len + 64),
              **kw)
    vocab = llm.engine.config.model_config.spec.vocab_size
    rng = np.random.default_rng(args.seed)
    in_lens = _lengths(args.input_len, args.num_prompts,
                       args.random_range_ratio, rng)
    out_lens = _lengths(args.output_len, args.num_prompts,
                        args.random_range_ratio, rng)
    lo, hi = 10, max(11, vocab - 100)
    prompts = [{"prompt_token_ids":
                rng.integers(lo, hi, size=n).tolist()} for n in in_lens]
    params = [SamplingParams(max_tokens=n, temperature=0.0, ignore_eos=True)
              for n in out_lens]

    t0 = time.perf_counter()
    outs = llm.generate(prompts, params)
    elapsed = time.perf_counter() - t0
    llm.shutdown()

    n_out = sum(len(o.outputs[0].token_ids) for o in outs)
    n_in = sum(in_lens)
    result = {
        "elapsed_time": elapsed,
        "num_requests": args.num_prompts,
        "total_num_tokens": n_in + n_out,
        "total_output_tokens": n_out,
        "requests_per_second": args.num_prompts / elapsed,
        "tokens_per_second": (n_in + n_out) / elapsed,
        "output_tokens_per_second": n_out / elapsed,
        "model": args.model,
        "input_len": args.input_len,
        "output_len": args.output_len,
    }
    return result


def main(argv=None) -> None:
    parser = argparse.ArgumentParser(description=__doc__)
    add_args(parser)
    args = parser.parse_args(argv)
    result = run(args)
    print(f"Throughput: {result['requests_per_second']:.2f} requests/s, "
          f"{result['tokens_per_second']:.1f} total tok/s, "
          f"{result['output_tokens_per_second']:.1f} output tok/s")
    print(json.dumps(result))
    if args.output_json:
        with open(args.output_json, "w") as f:
            json.dump(result, f, indent=2)


if __name__ == "__main__":
    main()
