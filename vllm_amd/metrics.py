"""Prometheus-format serving metrics.

Role of the reference's vllm/v1/metrics/loggers.py + prometheus export:
counters, gauges and histograms (TTFT, inter-token latency, e2e latency,
prompt/generation length) rendered in text exposition format at
/metrics. Plain Python, no client library — the exposition format is
three line shapes."""

from __future__ import annotations

import bisect
import threading
import time
from typing import Optional


class Histogram:
    def __init__(self, name: str, help_: str, buckets: list[float]):
        self.name = name
        self.help = help_
        self.buckets = sorted(buckets)
        self.counts = [0] * (len(self.buckets) + 1)  # +inf tail
        self.total = 0.0
        self.n = 0
        self._mu = threading.Lock()

    def observe(self, v: float) -> None:
        i = bisect.bisect_left(self.buckets, v)
        with self._mu:
            self.counts[i] += 1
            self.total += v
            self.n += 1

    def render(self) -> list[str]:
        lines = [f"# HELP {self.name} {self.help}",
                 f"# TYPE {self.name} histogram"]
        cum = 0
        with self._mu:
            for b, c in zip(self.buckets, self.counts):
                cum += c
                lines.append(f'{self.name}_bucket{{le="{b}"}} {cum}')
            cum += self.counts[-1]
            lines.append(f'{self.name}_bucket{{le="+Inf"}} {cum}')
            lines.append(f"{self.name}_sum {self.total}")
            lines.append(f"{self.name}_count {self.n}")
        return lines


_LATENCY_BUCKETS = [0.001, 0.005, 0.01, 0.02, 0.04, 0.06, 0.08, 0.1,
                    0.25, 0.5, 0.75, 1.0, 2.5, 5.0, 7.5, 10.0, 20.0,
                    40.0, 80.0]
_LEN_BUCKETS = [1, 2, 5, 10, 20, 50, 100, 200, 500, 1000, 2000, 5000,
                10000, 20000, 50000]


class ServerMetrics:
    """Per-server metric registry fed by the API handlers."""

    def __init__(self) -> None:
        self.ttft = Histogram(
            "vllm_amd:time_to_first_token_seconds",
            "Time from request arrival to first streamed token",
            _LATENCY_BUCKETS)
        self.itl = Histogram(
            "vllm_amd:time_per_output_token_seconds",
            "Inter-token latency of streamed tokens", _LATENCY_BUCKETS)
        self.e2e = Histogram(
            "vllm_amd:e2e_request_latency_seconds",
            "End-to-end request latency", _LATENCY_BUCKETS + [160.0, 640.0])
        self.prompt_len = Histogram(
            "vllm_amd:request_prompt_tokens",
            "Prompt length in tokens", _LEN_BUCKETS)
        self.gen_len = Histogram(
            "vllm_amd:request_generation_tokens",
            "Generated length in tokens", _LEN_BUCKETS)

    def render(self) -> list[str]:
        out: list[str] = []
        for h in (self.ttft, self.itl, self.e2e, self.prompt_len,
                  self.gen_len):
            out.extend(h.render())
        return out


class RequestTimer:
    """Tracks one request's TTFT / ITL / e2e and reports to the registry
    when the request finishes."""

    def __init__(self, metrics: Optional[ServerMetrics]):
        self.m = metrics
        self.t0 = time.monotonic()
        self.t_last: Optional[float] = None

    def on_tokens(self, n_new: int) -> None:
        if self.m is None or n_new <= 0:
            return
        now = time.monotonic()
        if self.t_last is None:
            # First delta: its arrival is the TTFT; extra tokens in the
            # same chunk carry no separable latency.
            self.m.ttft.observe(now - self.t0)
        else:
            dt = (now - self.t_last) / n_new
            for _ in range(n_new):
                self.m.itl.observe(dt)
        self.t_last = now

    def on_finish(self, prompt_tokens: int, gen_tokens: int) -> None:
        if self.m is None:
            return
        self.m.e2e.observe(time.monotonic() - self.t0)
        self.m.prompt_len.observe(prompt_tokens)
        self.m.gen_len.observe(gen_tokens)
