"""Multi-process executor: one worker process per GPU, spawned by the
engine (role of the reference's MultiprocExecutor,
vllm/v1/executor/multiproc_executor.py:108). Engine -> workers: a
single-writer shared-memory broadcast ring (executor/shm_queue.py, role
of shm_broadcast.py:464 MessageQueue) — each SchedulerOutput is pickled
ONCE, not once per worker. Workers -> engine: pipes (low volume; rank 0
ships the step result). The data plane between workers is RCCL over
xGMI.

The engine process owns the scheduler; worker rank 0 returns sampled
tokens. Workers execute steps asynchronously (a sender thread ships each
result when its GPU work completes), so the engine can schedule step N+1
while the workers run step N — the same one-step pipeline as the
in-process path.
"""

from __future__ import annotations

import logging
import os
import threading
from typing import Any

import torch.multiprocessing as mp

from vllm_amd.config import EngineConfig

logger = logging.getLogger(__name__)


def _worker_main(rank: int, config: EngineConfig, conn, master_port: int,
                 ring_name: str, num_readers: int, ring_size: int):
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(config.parallel_config.tensor_parallel_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(master_port)
    config.parallel_config.rank = rank
    config.parallel_config.local_rank = (
        config.parallel_config.device_offset + rank)
    config.parallel_config.world_size = \
        config.parallel_config.tensor_parallel_size

    from vllm_amd.worker.worker import Worker

    from vllm_amd.executor.shm_queue import ShmRing

    ring = ShmRing(num_readers, size=ring_size, name=ring_name,
                   create=False)

    worker = Worker(config)
    worker.init_device()
    worker.load_model()

    def orphan_check():
        # Engine death leaves readers spinning: exit when reparented.
        if os.getppid() == 1:
            raise EOFError("engine process died")

    send_lock = threading.Lock()

    def send(obj):
        with send_lock:
            conn.send(obj)

    pending = []  # queue of AsyncModelOutput futures (rank 0 only)
    pending_cv = threading.Condition()

    def sender_loop():
        while True:
            with pending_cv:
                while not pending:
                    pending_cv.wait()
                fut = pending.pop(0)
            if fut is None:
                return
            try:
                send(("ok", fut.result()))
            except Exception as e:  # noqa: BLE001
                send(("err", repr(e)))

    sender = None
    if rank == 0:
        sender = threading.Thread(target=sender_loop, daemon=True)
        sender.start()

    try:
        while True:
            msg = ring.read(rank, poll=orphan_check)
            kind = msg[0]
            if kind == "rpc":
                _, method, args, kwargs = msg
                try:
                    result = getattr(worker, method)(*args, **kwargs)
                    send(("ok", result))
                except Exception as e:  # noqa: BLE001
                    logger.exception("worker rpc %s failed", method)
                    send(("err", repr(e)))
            elif kind == "execute":
                so = msg[1]
                fut = worker.execute_model_async(so)
                if rank == 0:
                    with pending_cv:
                        pending.append(fut)
                        pending_cv.notify()
                else:
                    fut.result()
            elif kind == "execute_sync":
                so = msg[1]
                out = worker.execute_model(so)
                if rank == 0:
                    send(("ok", out))
            elif kind == "shutdown":
                if sender is not None:
                    with pending_cv:
                        pending.append(None)
                        pending_cv.notify()
                    sender.join(timeout=5)
                return
    except (EOFError, KeyboardInterrupt):
        pass
    finally:
        ring.close()


class _Future:
    def __init__(self, conn, executor=None):
        self._conn = conn
        self._executor = executor
        self._val = None
        self._done = False

    def result(self):
        if not self._done:
            if self._executor is not None:
                status, val = self._executor._recv(self._conn)
            else:
                status, val = self._conn.recv()
            if status == "err":
                raise RuntimeError(f"worker error: {val}")
            self._val = val
            self._done = True
        return self._val


class EngineDeadError(RuntimeError):
    """A worker (or the engine core) process died; the engine cannot make
    progress (role of the reference's EngineDeadError / fault sentinels).
    """


class MultiprocExecutor:
    """Engine-side handle to the worker processes."""

    def __init__(self, config: EngineConfig):
        self.config = config
        tp = config.parallel_config.tensor_parallel_size
        ctx = mp.get_context("spawn")
        port = (config.parallel_config.worker_port
                or int(os.environ.get("VLLM_AMD_WORKER_PORT", "29533")))
        from vllm_amd.executor.shm_queue import ShmRing

        ring_size = 8 << 20
        self.ring = ShmRing(tp, size=ring_size)
        self.conns = []
        self.procs = []
        for rank in range(tp):
            parent, child = ctx.Pipe()
            p = ctx.Process(
                target=_worker_main,
                args=(rank, config, child, port, self.ring.name, tp,
                      ring_size),
                daemon=True,
            )
            p.start()
            self.conns.append(parent)
            self.procs.append(p)

    def check_health(self) -> None:
        """Raise EngineDeadError if any worker process has died."""
        dead = [i for i, p in enumerate(self.procs) if not p.is_alive()]
        if dead:
            codes = [self.procs[i].exitcode for i in dead]
            raise EngineDeadError(
                f"worker process(es) {dead} died (exit codes {codes})")

    def _recv(self, conn):
        # Poll with a health check so a dead worker surfaces as a clean
        # EngineDeadError instead of a hang on the pipe.
        while not conn.poll(1.0):
            self.check_health()
        return conn.recv()

    def _send_all(self, msg) -> None:
        try:
            self.ring.write(msg, health_check=self.check_health)
        except EngineDeadError:
            raise
        except (ValueError, OSError) as e:
            self.check_health()
            raise EngineDeadError(f"shm ring write failed: {e}") from e

    # ---- control-plane RPC -------------------------------------------
    def collective_rpc(self, method: str, *args: Any, **kwargs: Any) -> list:
        self._send_all(("rpc", method, args, kwargs))
        results = []
        for conn in self.conns:
            try:
                status, val = self._recv(conn)
            except (EOFError, OSError) as e:
                self.check_health()
                raise EngineDeadError(f"worker pipe closed: {e}") from e
            if status == "err":
                raise RuntimeError(f"worker rpc {method} failed: {val}")
            results.append(val)
        return results

    def determine_num_kv_blocks(self) -> int:
        return min(self.collective_rpc("determine_num_kv_blocks"))

    def initialize_kv_cache(self, num_blocks: int) -> None:
        self.collective_rpc("initialize_kv_cache", num_blocks)

    def kv_cache_page_bytes(self) -> int:
        return self.collective_rpc("kv_cache_page_bytes")[0]

    def allocate_host_kv_pool(self, num_host_blocks: int) -> None:
        self.collective_rpc("allocate_host_kv_pool", num_host_blocks)

    def update_weights(self, model_path: str) -> None:
        self.collective_rpc("update_weights", model_path)

    def start_profile(self, out_dir: str) -> None:
        self.collective_rpc("start_profile", out_dir)

    def stop_profile(self) -> list:
        return self.collective_rpc("stop_profile")

    def save_sharded_state(self, out_dir: str) -> list:
        return self.collective_rpc("save_sharded_state", out_dir)

    def sleep(self, level: int = 1) -> None:
        self.collective_rpc("sleep", level)

    def wake_up(self) -> None:
        self.collective_rpc("wake_up")

    # ---- data plane ---------------------------------------------------
    def execute_model(self, so):
        self._send_all(("execute_sync", so))
        try:
            status, val = self._recv(self.conns[0])
        except (EOFError, OSError) as e:
            self.check_health()
            raise EngineDeadError(f"worker pipe closed: {e}") from e
        if status == "err":
            raise RuntimeError(f"worker step failed: {val}")
        return val

    def execute_model_async(self, so) -> _Future:
        self._send_all(("execute", so))
        return _Future(self.conns[0], self)

    def shutdown(self) -> None:
        try:
            self.ring.write(("shutdown",))
        except Exception:  # noqa: BLE001
            pass
        for p in self.procs:
            p.join(timeout=10)
            if p.is_alive():
                p.terminate()
        self.ring.close()
