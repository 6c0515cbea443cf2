"""Offline OpenAI-batch-format runner.

Role of the reference's vllm/entrypoints/openai/run_batch.py: read a
JSONL file of OpenAI batch request lines
  {"custom_id": ..., "method": "POST", "url": "/v1/chat/completions",
   "body": {...}}
run them through the normal serving stack (in-process ASGI — same
handlers, no sockets), and write one OpenAI batch response line per
request. Requests are submitted concurrently so the engine batches them.

Usage: python -m vllm_amd run-batch -i in.jsonl -o out.jsonl --model ...
"""

from __future__ import annotations

import argparse
import asyncio
import json
import sys

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.openai.api_server import make_server
from vllm_amd.entrypoints.openai.protocol import random_id

SUPPORTED_URLS = ("/v1/chat/completions", "/v1/completions",
                  "/v1/embeddings")


async def run_batch(app, lines: list[dict], max_concurrency: int = 128
                    ) -> list[dict]:
    import httpx

    sem = asyncio.Semaphore(max_concurrency)
    transport = httpx.ASGITransport(app=app)
    async with httpx.AsyncClient(transport=transport,
                                 base_url="http://batch",
                                 timeout=None) as client:
        async def one(line: dict) -> dict:
            cid = line.get("custom_id")
            url = line.get("url", "")
            out = {"id": random_id("batch_req"), "custom_id": cid,
                   "response": None, "error": None}
            if line.get("method", "POST") != "POST" or \
                    url not in SUPPORTED_URLS:
                out["error"] = {"message": f"unsupported request "
                                           f"{line.get('method')} {url}"}
                return out
            async with sem:
                r = await client.post(url, json=line.get("body") or {})
            body = (r.json() if "application/json" in
                    r.headers.get("content-type", "") else {"raw": r.text})
            out["response"] = {"status_code": r.status_code,
                               "request_id": random_id("req"),
                               "body": body}
            if r.status_code != 200:
                out["error"] = {"message": str(body)}
            return out

        return list(await asyncio.gather(*(one(ln) for ln in lines)))


def main() -> None:
    parser = argparse.ArgumentParser(
        description="offline OpenAI batch-format runner")
    parser.add_argument("-i", "--input-file", required=True)
    parser.add_argument("-o", "--output-file", required=True)
    parser.add_argument("--max-concurrency", type=int, default=128)
    EngineArgs.add_cli_args(parser)
    args = parser.parse_args()

    lines = []
    with open(args.input_file) as f:
        for raw in f:
            raw = raw.strip()
            if raw:
                lines.append(json.loads(raw))

    app, state = make_server(EngineArgs.from_cli_args(args))
    try:
        results = asyncio.run(run_batch(app, lines, args.max_concurrency))
        with open(args.output_file, "w") as f:
            for r in results:
                f.write(json.dumps(r) + "\n")
        ok = sum(1 for r in results if r["error"] is None)
        print(f"run-batch: {ok}/{len(results)} succeeded -> "
              f"{args.output_file}", file=sys.stderr)
    finally:
        state.engine.shutdown()


if __name__ == "__main__":
    main()
