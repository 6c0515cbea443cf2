"""OTLP/HTTP trace export (role of the reference's vllm/tracing.py +
--otlp-traces-endpoint, OTEL span per finished request).

No OpenTelemetry SDK dependency: spans are serialized straight to the
OTLP/HTTP JSON wire format (ExportTraceServiceRequest) and POSTed from a
background thread with stdlib urllib, batched, fire-and-forget — an
unreachable collector never stalls the engine loop. Attribute names
follow the reference's gen_ai semantic conventions
(vllm/tracing.py SpanAttributes) so existing dashboards work unchanged.
"""

from __future__ import annotations

import json
import logging
import os
import queue
import threading
import time
import urllib.request

logger = logging.getLogger(__name__)

_SERVICE = "vllm_amd"


def _hex(nbytes: int) -> str:
    return os.urandom(nbytes).hex()


def _attr(key: str, value):
    if isinstance(value, bool):
        return {"key": key, "value": {"boolValue": value}}
    if isinstance(value, int):
        return {"key": key, "value": {"intValue": str(value)}}
    if isinstance(value, float):
        return {"key": key, "value": {"doubleValue": value}}
    return {"key": key, "value": {"stringValue": str(value)}}


class OtelSpanExporter:
    """Batching OTLP/HTTP JSON exporter for request spans."""

    def __init__(self, endpoint: str, flush_interval_s: float = 1.0,
                 max_batch: int = 64):
        self.endpoint = endpoint.rstrip("/")
        if not self.endpoint.endswith("/v1/traces"):
            self.endpoint += "/v1/traces"
        self.flush_interval_s = flush_interval_s
        self.max_batch = max_batch
        self._q: queue.Queue = queue.Queue(maxsize=4096)
        self._shutdown = threading.Event()
        self._thread = threading.Thread(
            target=self._run, daemon=True, name="otlp-export")
        self._thread.start()

    # ------------------------------------------------------------------
    def export_request_span(self, request_id: str, model: str,
                            metrics: dict, prompt_tokens: int,
                            output_tokens: int,
                            finish_reason: str | None) -> None:
        arrival = metrics.get("arrival_time")
        first = metrics.get("first_token_time")
        finish = metrics.get("finish_time") or time.time()
        if arrival is None:
            arrival = finish
        attrs = [
            _attr("gen_ai.response.model", model),
            _attr("gen_ai.request.id", request_id),
            _attr("gen_ai.usage.prompt_tokens", prompt_tokens),
            _attr("gen_ai.usage.completion_tokens", output_tokens),
            _attr("gen_ai.latency.e2e", finish - arrival),
        ]
        if first is not None:
            attrs.append(
                _attr("gen_ai.latency.time_to_first_token",
                      first - arrival))
            if output_tokens > 1:
                attrs.append(_attr(
                    "gen_ai.latency.time_per_output_token",
                    (finish - first) / (output_tokens - 1)))
        if finish_reason:
            attrs.append(_attr("gen_ai.response.finish_reasons",
                               finish_reason))
        span = {
            "traceId": _hex(16),
            "spanId": _hex(8),
            "name": "llm_request",
            "kind": 2,  # SERVER
            "startTimeUnixNano": str(int(arrival * 1e9)),
            "endTimeUnixNano": str(int(finish * 1e9)),
            "attributes": attrs,
        }
        try:
            self._q.put_nowait(span)
        except queue.Full:
            pass  # drop rather than block the engine loop

    # ------------------------------------------------------------------
    def _run(self) -> None:
        while not self._shutdown.is_set() or not self._q.empty():
            batch = []
            try:
                batch.append(self._q.get(timeout=self.flush_interval_s))
            except queue.Empty:
                continue
            while len(batch) < self.max_batch:
                try:
                    batch.append(self._q.get_nowait())
                except queue.Empty:
                    break
            self._post(batch)

    def _post(self, spans: list) -> None:
        body = json.dumps({
            "resourceSpans": [{
                "resource": {"attributes": [
                    _attr("service.name", _SERVICE)]},
                "scopeSpans": [{
                    "scope": {"name": "vllm_amd.tracing"},
                    "spans": spans,
                }],
            }]
        }).encode()
        req = urllib.request.Request(
            self.endpoint, data=body,
            headers={"Content-Type": "application/json"})
        try:
            urllib.request.urlopen(req, timeout=2.0).read()
        except Exception as e:  # noqa: BLE001
            logger.debug("OTLP export failed: %s", e)

    def shutdown(self, timeout: float = 3.0) -> None:
        self._shutdown.set()
        self._thread.join(timeout=timeout)
