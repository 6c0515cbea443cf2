"""Engine-core process client (role of the reference's EngineCoreProc +
MPClient, vllm/v1/engine/core.py:1004 / core_client.py:503 — pickle
pipes instead of ZMQ+msgpack; same decoupling: the API process never
blocks on GPU steps, the engine process never parses HTTP).

The engine process runs a busy loop (schedule → execute → send outputs);
the client exposes the same surface as the in-process EngineCore so
LLMEngine/AsyncLLM work unchanged (`--multiprocess-engine`).
"""

from __future__ import annotations

import logging
import queue
import threading

import torch.multiprocessing as mp

from vllm_amd.config import EngineConfig
from vllm_amd.core.sched_output import EngineCoreOutput
from vllm_amd.request import Request

logger = logging.getLogger(__name__)


def _engine_proc_main(config: EngineConfig, conn) -> None:
    from vllm_amd.engine.core import EngineCore

    try:
        core = EngineCore(config)
        conn.send(("ready", core.num_gpu_blocks))
    except Exception as e:  # noqa: BLE001
        logger.exception("engine core init failed")
        conn.send(("init_error", repr(e)))
        return

    running = True
    while running:
        # Drain control messages; block when idle.
        while conn.poll(0 if core.has_unfinished_requests() else 0.005):
            try:
                msg = conn.recv()
            except EOFError:
                running = False
                break
            kind = msg[0]
            if kind == "add":
                core.add_request(msg[1])
            elif kind == "abort":
                core.abort_requests(msg[1])
            elif kind == "sleep":
                try:
                    core.sleep(msg[1])
                    conn.send(("slept", None))
                except Exception as e:  # noqa: BLE001
                    conn.send(("slept", repr(e)))
            elif kind == "wake_up":
                core.wake_up()
                conn.send(("woke", None))
            elif kind == "call":
                # Generic idle-time control RPC (update_weights,
                # save_sharded_state, start/stop_profile, ...).
                method, args = msg[1], msg[2]
                try:
                    result = getattr(core, method)(*args)
                    conn.send(("called", (None, result)))
                except Exception as e:  # noqa: BLE001
                    conn.send(("called", (repr(e), None)))
            elif kind == "shutdown":
                running = False
        if not running:
            break
        if not core.has_unfinished_requests():
            continue
        try:
            outputs = core.step()
        except Exception as e:  # noqa: BLE001
            logger.exception("engine step failed")
            conn.send(("error", repr(e)))
            continue
        if outputs:
            conn.send(("outputs", outputs))
            for out in outputs:
                if out.finished:
                    core.scheduler.release_request(out.req_id)
    core.shutdown()
    conn.send(("bye", None))


class EngineCoreClient:
    """Drop-in EngineCore replacement proxying to the engine process."""

    def __init__(self, config: EngineConfig):
        ctx = mp.get_context("spawn")
        self._conn, child = ctx.Pipe()
        self._proc = ctx.Process(
            target=_engine_proc_main, args=(config, child), daemon=True
        )
        self._proc.start()
        status, payload = self._conn.recv()
        if status != "ready":
            raise RuntimeError(f"engine core process failed: {payload}")
        self.num_gpu_blocks = payload
        self.is_driver = True
        self.scheduler = None  # lives in the engine process
        self._unfinished: set[str] = set()
        self._outq: "queue.Queue" = queue.Queue()
        self._ctrlq: "queue.Queue" = queue.Queue()
        self._sleeping = False
        self._recv_thread = threading.Thread(
            target=self._recv_loop, daemon=True, name="engine-core-recv"
        )
        self._alive = True
        self._recv_thread.start()

    def _recv_loop(self) -> None:
        while self._alive:
            try:
                if not self._conn.poll(0.05):
                    continue
                kind, payload = self._conn.recv()
            except (EOFError, OSError):
                return
            if kind == "outputs":
                self._outq.put(payload)
            elif kind in ("slept", "woke", "called"):
                self._ctrlq.put((kind, payload))
            elif kind == "error":
                self._outq.put(RuntimeError(payload))
            elif kind == "bye":
                return

    # ---- EngineCore surface -------------------------------------------
    def add_request(self, request: Request) -> None:
        self._unfinished.add(request.request_id)
        self._conn.send(("add", request))

    def abort_requests(self, request_ids: list[str]) -> None:
        for rid in request_ids:
            self._unfinished.discard(rid)
        self._conn.send(("abort", request_ids))

    def has_unfinished_requests(self) -> bool:
        return bool(self._unfinished)

    def sleep(self, level: int = 1) -> None:
        if self._unfinished:
            raise RuntimeError("cannot sleep with unfinished requests")
        self._conn.send(("sleep", level))
        kind, err = self._ctrlq.get(timeout=120)
        assert kind == "slept"
        if err:
            raise RuntimeError(err)
        self._sleeping = True

    def wake_up(self) -> None:
        self._conn.send(("wake_up",))
        kind, _ = self._ctrlq.get(timeout=300)
        assert kind == "woke"
        self._sleeping = False

    def is_sleeping(self) -> bool:
        return self._sleeping

    def _call(self, method: str, *args):
        self._conn.send(("call", method, args))
        kind, (err, result) = self._ctrlq.get(timeout=300)
        assert kind == "called"
        if err:
            raise RuntimeError(err)
        return result

    def update_weights(self, model_path: str) -> None:
        if self._unfinished:
            raise RuntimeError(
                "cannot update weights with unfinished requests")
        self._call("update_weights", model_path)

    def save_sharded_state(self, out_dir: str):
        return self._call("save_sharded_state", out_dir)

    def start_profile(self) -> None:
        self._call("start_profile")

    def stop_profile(self):
        return self._call("stop_profile")

    def check_health(self) -> None:
        from vllm_amd.executor.multiproc import EngineDeadError

        if not self._proc.is_alive():
            raise EngineDeadError(
                f"engine core process died (exit code "
                f"{self._proc.exitcode})")

    def step(self, timeout: float = 0.05) -> list[EngineCoreOutput]:
        """Dequeue one batch of outputs (the engine process steps on its
        own cadence; this just drains)."""
        try:
            item = self._outq.get(timeout=timeout)
        except queue.Empty:
            return []
        if isinstance(item, Exception):
            raise item
        outputs: list[EngineCoreOutput] = item
        for out in outputs:
            if out.finished:
                self._unfinished.discard(out.req_id)
        return outputs

    def shutdown(self) -> None:
        self._alive = False
        try:
            self._conn.send(("shutdown",))
        except (BrokenPipeError, OSError):
            pass
        self._proc.join(timeout=10)
        if self._proc.is_alive():
            self._proc.terminate()
