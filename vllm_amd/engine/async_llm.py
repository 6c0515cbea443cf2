"""AsyncLLM: asyncio front of the engine (role of vllm/v1/engine/
async_llm.py:72).

The engine loop runs on a dedicated thread (the GPU step is blocking);
requests arrive through a thread-safe queue and outputs are delivered to
per-request asyncio queues via call_soon_threadsafe. One engine step per
loop iteration — continuous batching keeps all active requests moving.
"""

from __future__ import annotations

import asyncio
import logging
import queue
import threading
import time
from typing import AsyncGenerator, Optional, Union

from vllm_amd.config import EngineConfig
from vllm_amd.engine.llm_engine import LLMEngine
from vllm_amd.outputs import RequestOutput
from vllm_amd.sampling_params import SamplingParams

logger = logging.getLogger(__name__)


class AsyncLLM:

    def __init__(self, config: EngineConfig):
        self.engine = LLMEngine(config)
        self.config = config
        # (request_id, prompt, params, loop, out_queue) | ("abort", rid)
        self._inbox: queue.Queue = queue.Queue()
        self._streams: dict[str, tuple[asyncio.AbstractEventLoop,
                                       asyncio.Queue]] = {}
        self._errors: dict[str, Exception] = {}
        self._shutdown = False
        self._engine_error: Optional[BaseException] = None
        self._counter = 0
        self._counter_lock = threading.Lock()
        self._thread = threading.Thread(
            target=self._run_loop, daemon=True, name="engine-loop"
        )
        self._thread.start()

    @property
    def tokenizer(self):
        return self.engine.tokenizer

    # ------------------------------------------------------------------
    def _run_loop(self) -> None:
        import time as _time

        last_log = _time.monotonic()
        tokens_since = 0
        while not self._shutdown:
            # Drain the inbox (block briefly when idle).
            block = not self.engine.has_unfinished_requests()
            while True:
                try:
                    item = self._inbox.get(timeout=0.005 if block else 0)
                except queue.Empty:
                    break
                block = False
                kind = item[0]
                if kind == "add":
                    _, rid, prompt, params, lora, loop, out_q = item
                    try:
                        self.engine.add_request(rid, prompt, params,
                                                lora=lora)
                        self._streams[rid] = (loop, out_q)
                    except Exception as e:  # noqa: BLE001
                        loop.call_soon_threadsafe(out_q.put_nowait, e)
                elif kind == "abort":
                    _, rid = item
                    if rid in self._streams:
                        self.engine.abort_request([rid])
                        self._streams.pop(rid, None)

            if not self.engine.has_unfinished_requests():
                continue
            try:
                outputs = self.engine.step()
            except Exception as e:  # noqa: BLE001
                logger.exception("engine step failed")
                from vllm_amd.executor.multiproc import EngineDeadError

                if isinstance(e, EngineDeadError):
                    # Unrecoverable: mark errored (health reports it) and
                    # fail all in-flight streams.
                    self._engine_error = e
                for rid, (loop, out_q) in self._streams.items():
                    loop.call_soon_threadsafe(out_q.put_nowait, e)
                self._streams.clear()
                if self._engine_error is not None:
                    return
                continue
            for out in outputs:
                if out.outputs:
                    tokens_since += len(out.outputs[0].token_ids) if \
                        out.finished else 0
                entry = self._streams.get(out.request_id)
                if entry is None:
                    continue
                loop, out_q = entry
                loop.call_soon_threadsafe(out_q.put_nowait, out)
                if out.finished:
                    self._streams.pop(out.request_id, None)

            # Periodic human stats line (role of the reference's
            # LoggingStatLogger).
            now = _time.monotonic()
            if now - last_log >= 10.0:
                s = self.stats()
                queries = s.get("prefix_cache_queries", 0)
                hit = (100.0 * s.get("prefix_cache_hits", 0) / queries
                       if queries else 0.0)
                logger.info(
                    "running=%d waiting=%d finished_toks/s=%.1f "
                    "kv_free=%d/%d prefix_hit=%.1f%%",
                    s.get("num_running", 0), s.get("num_waiting", 0),
                    tokens_since / (now - last_log),
                    s.get("kv_blocks_free", 0),
                    s.get("kv_blocks_total", 0), hit)
                last_log = now
                tokens_since = 0

    # ------------------------------------------------------------------
    def _next_id(self) -> str:
        with self._counter_lock:
            self._counter += 1
            return f"req-{self._counter}"

    async def generate(
        self,
        prompt: Union[str, list[int]],
        sampling_params: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
        lora: Optional[str] = None,
    ) -> AsyncGenerator[RequestOutput, None]:
        """Submit a request and stream RequestOutputs until finished."""
        rid = request_id or self._next_id()
        params = sampling_params or SamplingParams()
        loop = asyncio.get_running_loop()
        out_q: asyncio.Queue = asyncio.Queue()
        self._inbox.put(("add", rid, prompt, params, lora, loop, out_q))
        try:
            while True:
                item = await out_q.get()
                if isinstance(item, Exception):
                    raise item
                yield item
                if item.finished:
                    return
        finally:
            # Client disconnected / cancelled: abort in the engine.
            self._inbox.put(("abort", rid))

    async def abort(self, request_id: str) -> None:
        self._inbox.put(("abort", request_id))

    def start_profile(self) -> None:
        self.engine.engine_core.start_profile()

    def stop_profile(self):
        return self.engine.engine_core.stop_profile()

    def check_health(self) -> None:
        """Raise when the engine cannot serve (dead worker/engine proc or
        a fatal engine-loop error)."""
        if self._engine_error is not None:
            raise self._engine_error
        self.engine.check_health()

    def sleep(self, level: int = 1) -> None:
        """Release GPU memory (weights to host at level 1, discarded at
        level 2; KV pool freed). Only valid with no unfinished requests —
        the engine loop is parked in its idle poll then, so the call is
        safe from the API thread."""
        if self.engine.has_unfinished_requests():
            raise RuntimeError("cannot sleep with unfinished requests")
        self.engine.sleep(level)

    def wake_up(self) -> None:
        self.engine.wake_up()

    def is_sleeping(self) -> bool:
        return self.engine.is_sleeping()

    def stats(self) -> dict:
        sched = self.engine.engine_core.scheduler
        if sched is None:
            return {}
        return {
            "num_running": len(sched.running),
            "num_waiting": len(sched.waiting),
            "kv_blocks_total": self.engine.engine_core.num_gpu_blocks,
            "kv_blocks_free":
                sched.kv_cache_manager.block_pool.get_num_free_blocks(),
            "prefix_cache_queries": sched.prefix_cache_queries,
            "prefix_cache_hits": sched.prefix_cache_hits,
            "num_preemptions": sched.num_preemptions_total,
            "spec_tokens_drafted": sched.spec_stats_drafted,
            "spec_tokens_accepted": sched.spec_stats_accepted,
        }

    def shutdown(self) -> None:
        self._shutdown = True
        self._thread.join(timeout=5)
        self.engine.shutdown()
