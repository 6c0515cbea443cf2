"""OTLP/HTTP request-span export (reference --otlp-traces-endpoint,
vllm/tracing.py). A local HTTP collector receives the OTLP JSON batch;
spans carry the gen_ai.* attributes."""

import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

from vllm_amd.engine.arg_utils import EngineArgs
from vllm_amd.entrypoints.llm import LLM
from vllm_amd.sampling_params import SamplingParams


class _Collector(BaseHTTPRequestHandler):
    received: list = []

    def do_POST(self):  # noqa: N802
        n = int(self.headers.get("Content-Length", 0))
        _Collector.received.append(
            (self.path, json.loads(self.rfile.read(n))))
        self.send_response(200)
        self.end_headers()
        self.wfile.write(b"{}")

    def log_message(self, *a):  # silence
        pass


def test_otlp_span_export():
    srv = HTTPServer(("127.0.0.1", 0), _Collector)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        llm = LLM(model="tiny-llama", dtype="fp32", device="cpu",
                  block_size=16, num_gpu_blocks=64, max_model_len=256,
                  max_num_batched_tokens=256, max_num_seqs=4,
                  otlp_traces_endpoint=f"http://127.0.0.1:{port}")
        llm.generate(
            ["hello tracing"],
            SamplingParams(max_tokens=6, temperature=0.0,
                           ignore_eos=True))
        llm.shutdown()  # flushes the exporter
        assert _Collector.received, "no OTLP batch arrived"
        path, body = _Collector.received[0]
        assert path == "/v1/traces"
        spans = body["resourceSpans"][0]["scopeSpans"][0]["spans"]
        assert spans[0]["name"] == "llm_request"
        attrs = {a["key"]: a["value"] for a in spans[0]["attributes"]}
        assert attrs["gen_ai.usage.completion_tokens"]["intValue"] == "6"
        assert "gen_ai.latency.time_to_first_token" in attrs
        assert "gen_ai.latency.e2e" in attrs
        res = {a["key"]: a["value"] for a in
               body["resourceSpans"][0]["resource"]["attributes"]}
        assert res["service.name"]["stringValue"] == "vllm_amd"
    finally:
        srv.shutdown()


def test_otlp_cli_flag():
    p = EngineArgs.add_cli_args(__import__("argparse").ArgumentParser())
    a = p.parse_args(["--model", "tiny-llama",
                      "--otlp-traces-endpoint", "http://c:4318"])
    ea = EngineArgs.from_cli_args(a)
    cfg = ea.create_engine_config()
    assert (cfg.observability_config.otlp_traces_endpoint
            == "http://c:4318")
