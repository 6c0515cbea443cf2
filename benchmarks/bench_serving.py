"""Serving benchmark: output tokens/sec + TTFT/ITL percentiles at a fixed
request rate (role of the reference's `vllm bench serve`,
vllm/benchmarks/serve.py — the BASELINE.json metric's latency half).

Drives the full serving path in-process (FastAPI app via ASGI transport:
HTTP parsing, SSE streaming, detokenization) with synthetic random
prompts (no network/datasets needed).

    python benchmarks/bench_serving.py --model llama-3-8b \
        --num-prompts 128 --qps 8 --input-len 128 --output-len 128
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--model", type=str, default="llama-3-8b")
    p.add_argument("--num-prompts", type=int, default=128)
    p.add_argument("--qps", type=float, default=8.0,
                   help="request arrival rate (poisson); inf = burst")
    p.add_argument("--input-len", type=int, default=128)
    p.add_argument("--output-len", type=int, default=128)
    p.add_argument("--dtype", type=str, default="bf16")
    p.add_argument("--max-num-seqs", type=int, default=256)
    p.add_argument("--num-gpu-blocks", type=int, default=None)
    p.add_argument("--block-size", type=int, default=64)
    p.add_argument("--server-mode", choices=("subprocess", "inproc"),
                   default="subprocess",
                   help="subprocess: real uvicorn server + TCP sockets "
                        "(the reference methodology — client cost does "
                        "not steal the server's event loop); inproc: "
                        "ASGI transport in one process")
    return p.parse_args()


async def run(args) -> dict:
    import httpx
    import torch

    from vllm_amd.engine.arg_utils import EngineArgs
    from vllm_amd.entrypoints.openai.api_server import make_server

    on_gpu = torch.cuda.is_available()
    if not on_gpu and args.model == "llama-3-8b":
        args.model = "tiny-llama"
        args.dtype = "fp32"
        args.block_size = 16
        args.num_gpu_blocks = args.num_gpu_blocks or 2048

    engine_args = EngineArgs(
        model=args.model,
        dtype=args.dtype,
        device="cuda" if on_gpu else "cpu",
        block_size=args.block_size,
        num_gpu_blocks=args.num_gpu_blocks,
        max_model_len=args.input_len + args.output_len + 32,
        max_num_batched_tokens=max(32768, args.input_len * 8),
        max_num_seqs=args.max_num_seqs,
    )
    rng = np.random.default_rng(0)
    proc = None
    state = None
    if args.server_mode == "subprocess":
        import socket as _socket
        import subprocess
        import urllib.request

        from vllm_amd.config import MODEL_PRESETS

        vocab = MODEL_PRESETS[args.model].vocab_size
        srv = _socket.socket()
        srv.bind(("127.0.0.1", 0))
        port = srv.getsockname()[1]
        srv.close()
        cmd = [sys.executable, "-m", "vllm_amd", "serve", args.model,
               "--dtype", args.dtype, "--device",
               "cuda" if on_gpu else "cpu",
               "--block-size", str(args.block_size),
               "--max-model-len", str(args.input_len + args.output_len + 32),
               "--max-num-batched-tokens",
               str(max(32768, args.input_len * 8)),
               "--max-num-seqs", str(args.max_num_seqs),
               "--host", "127.0.0.1", "--port", str(port)]
        if args.num_gpu_blocks:
            cmd += ["--num-gpu-blocks", str(args.num_gpu_blocks)]
        proc = subprocess.Popen(cmd, stdout=subprocess.DEVNULL,
                                stderr=subprocess.DEVNULL)
        deadline = time.time() + 420
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health", timeout=2) as r:
                    if r.status == 200:
                        break
            except Exception:
                if proc.poll() is not None:
                    raise RuntimeError("server process died during boot")
                await asyncio.sleep(0.5)
        else:
            proc.kill()
            raise RuntimeError("server did not come up in 420s")
        client = httpx.AsyncClient(base_url=f"http://127.0.0.1:{port}",
                                   timeout=600.0)
    else:
        app, state = make_server(engine_args, served_model_name=args.model)
        vocab = state.engine.config.model_config.spec.vocab_size
        transport = httpx.ASGITransport(app=app)
        client = httpx.AsyncClient(transport=transport,
                                   base_url="http://bench", timeout=600.0)

    results = []

    async def one_request(i: int):
        toks = rng.integers(16, vocab - 16, size=args.input_len).tolist()
        t0 = time.perf_counter()
        ttft = None
        token_times = []
        async with client.stream("POST", "/v1/completions", json={
            "model": args.model,
            "prompt": toks,
            "max_tokens": args.output_len,
            "temperature": 0.0,
            "ignore_eos": True,
            "stream": True,
        }) as r:
            async for line in r.aiter_lines():
                if not line.startswith("data: "):
                    continue
                payload = line[6:]
                if payload == "[DONE]":
                    break
                now = time.perf_counter()
                if ttft is None:
                    ttft = now - t0
                token_times.append(now)
        results.append({
            "ttft": ttft,
            "e2e": time.perf_counter() - t0,
            "n_chunks": len(token_times),
            "itl": np.diff(token_times).tolist() if len(token_times) > 1
            else [],
        })

    t_start = time.perf_counter()
    tasks = []
    for i in range(args.num_prompts):
        tasks.append(asyncio.create_task(one_request(i)))
        if np.isfinite(args.qps):
            await asyncio.sleep(rng.exponential(1.0 / args.qps))
    await asyncio.gather(*tasks)
    elapsed = time.perf_counter() - t_start
    await client.aclose()
    if state is not None:
        state.engine.shutdown()
    if proc is not None:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except Exception:  # noqa: BLE001
            proc.kill()

    ttfts = np.array([r["ttft"] for r in results])
    e2es = np.array([r["e2e"] for r in results])
    itls = np.concatenate([r["itl"] for r in results if r["itl"]])
    total_out = args.num_prompts * args.output_len
    return {
        "metric": "serving throughput + TTFT/ITL at fixed QPS",
        "model": args.model,
        "num_prompts": args.num_prompts,
        "qps": args.qps,
        "input_len": args.input_len,
        "output_len": args.output_len,
        "duration_s": round(elapsed, 2),
        "output_tokens_per_s": round(total_out / elapsed, 1),
        "request_rate_achieved": round(args.num_prompts / elapsed, 2),
        "ttft_ms": {
            "p50": round(float(np.percentile(ttfts, 50)) * 1e3, 1),
            "p90": round(float(np.percentile(ttfts, 90)) * 1e3, 1),
            "p99": round(float(np.percentile(ttfts, 99)) * 1e3, 1),
            "mean": round(float(ttfts.mean()) * 1e3, 1),
        },
        "itl_ms": {
            "p50": round(float(np.percentile(itls, 50)) * 1e3, 2),
            "p90": round(float(np.percentile(itls, 90)) * 1e3, 2),
            "p99": round(float(np.percentile(itls, 99)) * 1e3, 2),
            "mean": round(float(itls.mean()) * 1e3, 2),
        },
        "e2e_ms_p50": round(float(np.percentile(e2es, 50)) * 1e3, 1),
    }


def main():
    args = parse_args()
    print(json.dumps(asyncio.run(run(args))))


if __name__ == "__main__":
    main()
