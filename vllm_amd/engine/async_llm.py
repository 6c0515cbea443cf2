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
import os
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
                failed = list(self._streams.keys())
                for rid, (loop, out_q) in self._streams.items():
                    loop.call_soon_threadsafe(out_q.put_nowait, e)
                self._streams.clear()
                if self._engine_error is not None:
                    return
                # Drain the requests that were in flight when the step
                # crashed — otherwise a poisoned request re-crashes
                # every subsequent step and the loop wedges while
                # /health still reports green.
                try:
                    if failed:
                        self.engine.abort_request(failed)
                except Exception:  # noqa: BLE001
                    logger.exception("abort after step failure failed")
                if self.engine.has_unfinished_requests():
                    # Could not drain: stop pretending to be healthy.
                    self._engine_error = EngineDeadError(
                        "engine step failed and the scheduler could "
                        "not be drained; engine marked dead")
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
            "num_encoder_deferrals": sched.num_encoder_deferrals,
            "spec_tokens_drafted": sched.spec_stats_drafted,
            "spec_tokens_accepted": sched.spec_stats_accepted,
        }

    def shutdown(self) -> None:
        self._shutdown = True
        self._thread.join(timeout=5)
        self.engine.shutdown()


class DPAsyncLLM:
    """Serve-level data-parallel router (role of the reference's
    vllm/v1/engine/core_client.py DPLBAsyncMPClient: --data-parallel-size
    engine replicas behind one API server, least-loaded routing).

    Replica i owns GPUs [i*tp, (i+1)*tp): its config gets
    device_offset=i*tp (uniproc workers take the device directly from
    local_rank; multiproc workers add the offset per rank) and a unique
    worker_port so the replicas' TP process groups don't collide. Each
    replica runs its own engine loop; new requests go to the replica
    with the fewest in-flight requests. xGMI note: one replica's TP
    traffic stays on its own GPUs' links, so replicas don't contend."""

    def __init__(self, config: EngineConfig):
        import copy

        dp = config.parallel_config.data_parallel_size
        tp = config.parallel_config.tensor_parallel_size
        base_port = int(os.environ.get("VLLM_AMD_WORKER_PORT", "29533"))
        self.replicas: list[AsyncLLM] = []
        for i in range(dp):
            c = copy.deepcopy(config)
            c.parallel_config.data_parallel_size = 1
            c.parallel_config.device_offset = i * tp
            c.parallel_config.local_rank = i * tp
            c.parallel_config.worker_port = base_port + i
            self.replicas.append(AsyncLLM(c))
        self.config = config
        self._in_flight = [0] * dp
        self._counter = 0
        self._counter_lock = threading.Lock()

    @property
    def tokenizer(self):
        return self.replicas[0].tokenizer

    def _next_id(self) -> str:
        with self._counter_lock:
            self._counter += 1
            return f"req-{self._counter}"

    def _pick(self) -> int:
        """Least-loaded HEALTHY replica; a replica whose engine died
        (check_health raises) is routed around instead of failing every
        request that lands on it. All-dead falls through to replica 0
        so the caller gets the real error."""
        alive = []
        for i, r in enumerate(self.replicas):
            try:
                r.check_health()
                alive.append(i)
            except Exception:  # noqa: BLE001
                continue
        if not alive:
            alive = [0]
        return min(alive, key=lambda i: self._in_flight[i])

    async def generate(self, prompt, sampling_params=None,
                       request_id=None, lora=None):
        rid = request_id or self._next_id()
        i = self._pick()
        self._in_flight[i] += 1
        try:
            async for out in self.replicas[i].generate(
                    prompt, sampling_params, request_id=rid, lora=lora):
                yield out
        finally:
            self._in_flight[i] -= 1

    async def abort(self, request_id: str) -> None:
        for r in self.replicas:
            await r.abort(request_id)

    def start_profile(self) -> None:
        for r in self.replicas:
            r.start_profile()

    def stop_profile(self):
        return [r.stop_profile() for r in self.replicas]

    def check_health(self) -> None:
        for r in self.replicas:
            r.check_health()

    def sleep(self, level: int = 1) -> None:
        for r in self.replicas:
            r.sleep(level)

    def wake_up(self) -> None:
        for r in self.replicas:
            r.wake_up()

    def is_sleeping(self) -> bool:
        return any(r.is_sleeping() for r in self.replicas)

    def stats(self) -> dict:
        agg: dict = {}
        for r in self.replicas:
            for k, v in r.stats().items():
                agg[k] = agg.get(k, 0) + v
        agg["dp_in_flight"] = list(self._in_flight)
        return agg

    def shutdown(self) -> None:
        for r in self.replicas:
            r.shutdown()
