"""End-to-end `python -m vllm_amd serve` boot: real uvicorn process,
HTTP + gRPC ports, health/completion round-trip, clean SIGTERM."""

import json
import os
import signal
import socket
import subprocess
import sys
import time
import urllib.request


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_serve_cli_boot():
    port, gport = _free_port(), _free_port()
    proc = subprocess.Popen(
        [sys.executable, "-m", "vllm_amd", "serve", "tiny-llama",
         "--device", "cpu", "--dtype", "fp32", "--block-size", "16",
         "--num-gpu-blocks", "64", "--max-model-len", "128",
         "--max-num-batched-tokens", "64", "--max-num-seqs", "2",
         "--host", "127.0.0.1", "--port", str(port),
         "--grpc-port", str(gport)],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    try:
        deadline = time.time() + 60
        up = False
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}/health", timeout=2) as r:
                    if r.status == 200:
                        up = True
                        break
            except Exception:
                time.sleep(0.5)
        assert up, "server did not come up"
        body = json.dumps({
            "model": "tiny-llama", "prompt": "boot check",
            "max_tokens": 3, "temperature": 0.0, "ignore_eos": True,
        }).encode()
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/v1/completions", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=30) as r:
            data = json.loads(r.read())
        assert data["usage"]["completion_tokens"] == 3
        # gRPC health over the real socket
        import grpc

        from vllm_amd.entrypoints.grpc.server import MSG
        with grpc.insecure_channel(f"127.0.0.1:{gport}") as ch:
            health = ch.unary_unary(
                "/vllm_amd.inference.Inference/Health",
                request_serializer=MSG["HealthRequest"].SerializeToString,
                response_deserializer=MSG["HealthResponse"].FromString)
            assert health(MSG["HealthRequest"](), timeout=10).ok
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
            proc.wait(timeout=10)


def _run_cli(args):
    out = subprocess.run(
        [sys.executable, "-m", "vllm_amd"] + args,
        capture_output=True, text=True, timeout=240,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert out.returncode == 0, out.stdout + out.stderr
    # Last stdout line is the machine-readable JSON summary.
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_bench_latency_cli():
    res = _run_cli(["bench", "latency", "--model", "tiny-llama",
                    "--device", "cpu", "--dtype", "fp32",
                    "--input-len", "8", "--output-len", "4",
                    "--batch-size", "2", "--num-iters-warmup", "1",
                    "--num-iters", "2"])
    assert res["avg_latency"] > 0
    assert len(res["latencies"]) == 2
    assert "50" in res["percentiles"]


def test_bench_throughput_cli():
    res = _run_cli(["bench", "throughput", "--model", "tiny-llama",
                    "--device", "cpu", "--dtype", "fp32",
                    "--num-prompts", "6", "--input-len", "16",
                    "--output-len", "8"])
    assert res["num_requests"] == 6
    assert res["total_output_tokens"] == 6 * 8
    assert res["requests_per_second"] > 0
