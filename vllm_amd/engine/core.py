"""EngineCore: owns scheduler + worker(s); one step() = schedule ->
execute -> update (role of vllm/v1/engine/core.py:103).

Two launch modes:

- Single process (tp=1): scheduler and worker in-proc.
- SPMD (torchrun, world>1): every rank builds an EngineCore; rank 0 owns
  the scheduler and broadcasts each SchedulerOutput over the CPU (gloo)
  group; all ranks execute the model step collectively over RCCL/xGMI.
  Non-zero ranks drive their loop via run_spmd_worker_loop().
"""

from __future__ import annotations

import logging
import time
from typing import Optional

import torch

from vllm_amd.config import EngineConfig
from vllm_amd.core.sched_output import EngineCoreOutput, SchedulerOutput
from vllm_amd.core.scheduler import Scheduler
from vllm_amd.request import Request, RequestStatus
from vllm_amd.worker.worker import Worker

logger = logging.getLogger(__name__)


class EngineCore:

    def __init__(self, config: EngineConfig):
        import os

        self.config = config
        pp = config.parallel_config.pipeline_parallel_size
        if pp > 1:
            # v1 pipeline: synchronous lockstep; no spec decode/pooling.
            config.scheduler_config.async_scheduling = False
            if config.scheduler_config.num_speculative_tokens > 0:
                raise ValueError("speculative decoding requires pp=1")
        tp = config.parallel_config.tensor_parallel_size
        launched_world = int(os.environ.get("WORLD_SIZE", "1"))
        self._multiproc = tp > 1 and launched_world == 1

        # The engine's lockstep domain is the TP group: with tp == world this
        # is the whole job (classic SPMD TP); with tp < world each TP group
        # runs an independent engine replica (SPMD data parallelism — the
        # bench's dp mode; role of the reference's DP engine replicas).
        from vllm_amd.parallel.state import get_replica_group

        if self._multiproc:
            # Engine owns the scheduler; one spawned worker process per
            # GPU (RCCL group lives in the workers).
            from vllm_amd.executor.multiproc import MultiprocExecutor

            self.worker = MultiprocExecutor(config)
            self.world = get_replica_group()  # engine proc: group of 1
            self.is_driver = True
            num_blocks = self.worker.determine_num_kv_blocks()
            self.num_gpu_blocks = num_blocks
            self.worker.initialize_kv_cache(num_blocks)
        else:
            # In-process worker; SPMD under torchrun (one rank per GPU).
            self.worker = Worker(config)
            self.worker.init_device()
            self.worker.load_model()

            self.world = get_replica_group()
            self.is_driver = self.world.rank_in_group == 0

            # KV sizing must agree across ranks: min over ranks.
            num_blocks = self.worker.determine_num_kv_blocks()
            if self.world.world_size > 1:
                t = torch.tensor([num_blocks], dtype=torch.int64)
                if torch.cuda.is_available():
                    t = t.cuda()
                torch.distributed.all_reduce(
                    t, op=torch.distributed.ReduceOp.MIN,
                    group=self.world.device_group,
                )
                num_blocks = int(t.item())
            self.num_gpu_blocks = num_blocks
            self.worker.initialize_kv_cache(num_blocks)

        offload_gb = config.cache_config.cpu_offload_gb
        num_host_blocks = 0
        if offload_gb > 0:
            page = self.worker.kv_cache_page_bytes()
            num_host_blocks = int(offload_gb * (1 << 30) // page)
            self.worker.allocate_host_kv_pool(num_host_blocks)
        self.scheduler: Optional[Scheduler] = (
            Scheduler(config, num_gpu_blocks=num_blocks,
                      num_host_blocks=num_host_blocks)
            if self.is_driver
            else None
        )
        self.async_scheduling = config.scheduler_config.async_scheduling
        # One in-flight (SchedulerOutput, AsyncModelOutput) when pipelining.
        self._pending = None
        # KV-cache event stream for cache-aware routers (kv_events.py).
        self.kv_events = None
        ep = getattr(config.observability_config, "kv_events_endpoint", None)
        if ep and self.scheduler is not None:
            from vllm_amd.kv_events import KVEventPublisher
            self.kv_events = KVEventPublisher(ep)
            self.kv_events.attach(
                self.scheduler.kv_cache_manager.block_pool)

    # ------------------------------------------------------------------
    def add_request(self, request: Request) -> None:
        assert self.is_driver
        self.scheduler.add_request(request)

    def abort_requests(self, request_ids: list[str]) -> None:
        assert self.is_driver
        self.scheduler.finish_requests(
            request_ids, RequestStatus.FINISHED_ABORTED
        )

    def has_unfinished_requests(self) -> bool:
        return self.scheduler is not None and \
            self.scheduler.has_unfinished_requests()

    # ------------------------------------------------------------------
    # Sleep mode (role of the reference's engine sleep/wake_up): release
    # GPU memory between serving bursts. Level 1 offloads weights to host
    # RAM and frees the KV pool; level 2 discards weights too.
    def sleep(self, level: int = 1) -> None:
        if self.has_unfinished_requests():
            raise RuntimeError("cannot sleep with unfinished requests")
        if self._pending is not None:
            self._drain()
        self.worker.sleep(level)
        if self.scheduler is not None:
            # Cached prefix blocks point into the freed pool.
            self.scheduler.kv_cache_manager.reset_prefix_cache()
        self._sleeping = True

    def wake_up(self) -> None:
        self.worker.wake_up()
        self._sleeping = False

    def is_sleeping(self) -> bool:
        return getattr(self, "_sleeping", False)

    def update_weights(self, model_path: str) -> None:
        """In-place weight refresh (RL-style update). Engine must be
        idle — the next step serves the new weights."""
        if self.has_unfinished_requests():
            raise RuntimeError(
                "cannot update weights with unfinished requests")
        if self._pending is not None:
            self._drain()
        self.worker.update_weights(model_path)

    def save_sharded_state(self, out_dir: str):
        return self.worker.save_sharded_state(out_dir)

    def start_profile(self) -> None:
        self.worker.start_profile(
            self.config.observability_config.profile_dir)

    def stop_profile(self):
        return self.worker.stop_profile()

    def check_health(self) -> None:
        """Raise EngineDeadError if a worker process died (in-proc
        workers cannot die independently — no-op)."""
        if self._multiproc:
            self.worker.check_health()

    # ------------------------------------------------------------------
    def _drain(self) -> list[EngineCoreOutput]:
        if self._pending is None:
            return []
        so_prev, fut_prev = self._pending
        self._pending = None
        return self.scheduler.update_from_output(so_prev, fut_prev.result())

    def step(self) -> list[EngineCoreOutput]:
        """One engine iteration on the driver rank. With async scheduling
        the CPU schedules step N+1 while the GPU runs step N (placeholder
        tokens; results consumed one step later)."""
        assert self.is_driver
        scheduler_output = self.scheduler.schedule()
        if self.world.world_size > 1:
            self.world.broadcast_object(scheduler_output, src=0)
        pure_decode = (
            scheduler_output.total_num_scheduled_tokens > 0
            and scheduler_output.total_num_scheduled_tokens
            == len(scheduler_output.num_scheduled_tokens)
        )
        use_async = (self.async_scheduling and pure_decode
                     and not self.scheduler.has_guided_requests()
                     and not self.scheduler.has_pooling_requests())
        if not use_async:
            # Mixed/prefill/empty steps run synchronously: the runner's
            # slow path reads token values the pending step produces.
            outputs = self._drain()
            runner_output = self.worker.execute_model(scheduler_output)
            outputs += self.scheduler.update_from_output(
                scheduler_output, runner_output
            )
            return outputs
        fut = self.worker.execute_model_async(scheduler_output)
        outputs = self._drain()
        self._pending = (scheduler_output, fut)
        return outputs

    def step_worker(self) -> bool:
        """Non-driver ranks: receive ONE SchedulerOutput broadcast and
        execute it collectively. Returns False on the shutdown sentinel.
        Exactly one broadcast happens per driver step(), so counted
        lockstep loops (bench.py) stay in sync by construction."""
        assert not self.is_driver
        so = self.world.broadcast_object(None, src=0)
        if so is None:
            return False
        self.worker.execute_model(so)
        return True

    def run_spmd_worker_loop(self) -> None:
        """Non-driver ranks: receive scheduler outputs forever."""
        while self.step_worker():
            pass

    def shutdown(self) -> None:
        if self.kv_events is not None:
            self.kv_events.close()
        if self.is_driver and self._pending is not None:
            try:
                self._drain()
            except Exception:  # noqa: BLE001
                pass
        if self._multiproc:
            self.worker.shutdown()
        if self.is_driver and self.world.world_size > 1:
            self.world.broadcast_object(None, src=0)
