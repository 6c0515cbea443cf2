"""Worker: device init, model load, KV memory profiling, step execution.

Role of the reference's Worker (vllm/v1/worker/gpu_worker.py:128). One
worker per GPU; in SPMD mode (torchrun launch) every rank runs one and
rank 0 owns the scheduler.
"""

from __future__ import annotations

import gc
import logging

import os

import torch

from vllm_amd.config import EngineConfig
from vllm_amd.core.sched_output import ModelRunnerOutput, SchedulerOutput
from vllm_amd.worker.model_runner import ModelRunner

logger = logging.getLogger(__name__)


class Worker:

    def __init__(self, config: EngineConfig):
        self.config = config
        self.device: torch.device = None  # type: ignore
        self.runner: ModelRunner = None  # type: ignore

    def init_device(self) -> None:
        from vllm_amd.parallel.state import init_distributed

        pc = self.config.parallel_config
        if self.config.device_config.device == "cuda":
            torch.cuda.set_device(pc.local_rank)
            self.device = torch.device("cuda", pc.local_rank)
            # Debug sync-check (role of the reference's
            # with_gpu_sync_check, gpu_worker.py:1023): surface
            # accidental host synchronizations in the async decode path.
            # VLLM_AMD_SYNC_CHECK=warn|error.
            mode = os.environ.get("VLLM_AMD_SYNC_CHECK")
            if mode in ("warn", "error"):
                torch.cuda.set_sync_debug_mode(mode)
        else:
            self.device = torch.device("cpu")
        if pc.needs_distributed:
            init_distributed(
                tensor_parallel_size=pc.tensor_parallel_size,
                pipeline_parallel_size=pc.pipeline_parallel_size,
                backend=pc.distributed_backend,
            )
        torch.manual_seed(self.config.model_config.seed)
        self.runner = ModelRunner(self.config, self.device)

    def load_model(self) -> None:
        self.runner.load_model()

    def determine_num_kv_blocks(self) -> int:
        """Profile peak memory with a worst-case dummy forward, then size the
        KV pool into the remaining HBM (288 GB on MI355X)."""
        cache_cfg = self.config.cache_config
        if cache_cfg.num_gpu_blocks is not None:
            return cache_cfg.num_gpu_blocks
        if self.device.type != "cuda":
            return 4096  # CPU: arbitrary small pool
        torch.cuda.empty_cache()
        torch.cuda.reset_peak_memory_stats(self.device)
        free_before, total = torch.cuda.mem_get_info(self.device)
        self.runner.profile_run()
        torch.cuda.synchronize(self.device)
        peak = torch.cuda.max_memory_allocated(self.device)
        non_torch = (total - free_before) - torch.cuda.memory_allocated(
            self.device
        )
        usable = (
            total * cache_cfg.gpu_memory_utilization - peak - max(non_torch, 0)
        )
        page_bytes = self.runner.kv_cache_page_bytes()
        num_blocks = max(int(usable // page_bytes), 16)
        gc.collect()
        torch.cuda.empty_cache()
        logger.info(
            "KV sizing: total=%.1fGB peak_profile=%.1fGB page=%dKB "
            "-> %d blocks (%.1fGB, %d tokens)",
            total / 1e9, peak / 1e9, page_bytes // 1024, num_blocks,
            num_blocks * page_bytes / 1e9,
            num_blocks * cache_cfg.block_size,
        )
        return num_blocks

    def initialize_kv_cache(self, num_blocks: int) -> None:
        self.runner.allocate_kv_cache(num_blocks)

    def kv_cache_page_bytes(self) -> int:
        return self.runner.kv_cache_page_bytes()

    def allocate_host_kv_pool(self, num_host_blocks: int) -> None:
        self.runner.allocate_host_kv_pool(num_host_blocks)

    def update_weights(self, model_path: str) -> None:
        self.runner.update_weights(model_path)

    def start_profile(self, out_dir: str) -> None:
        self.runner.start_profile(out_dir)

    def stop_profile(self) -> str:
        return self.runner.stop_profile()

    def save_sharded_state(self, out_dir: str) -> str:
        return self.runner.save_sharded_state(out_dir)

    def sleep(self, level: int = 1) -> None:
        self.runner.sleep(level)

    def wake_up(self) -> None:
        self.runner.wake_up()

    def execute_model(self, so: SchedulerOutput) -> ModelRunnerOutput:
        return self.runner.execute_model(so)

    def execute_model_async(self, so: SchedulerOutput):
        """Launch a step; returns an AsyncModelOutput (result() waits for
        the sampled-token copy)."""
        return self.runner.execute_model_async(so)
