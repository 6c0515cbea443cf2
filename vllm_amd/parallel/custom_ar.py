"""Custom xGMI all-reduce / all-gather wrapper (csrc/comms.hip).

Role of the reference's vllm/distributed/device_communicators/
custom_all_reduce.py:74 (size-gated IPC all-reduce in front of NCCL),
MI355X-shaped: one process per GPU on one node, buffers exchanged as
hipIpc handles over the TP group's gloo CPU group, device-side flag
counters so hipGraph capture/replay of TP decode steps needs no host
involvement.

Dispatch policy (GroupCoordinator.all_reduce):
  <= max_bytes and 16B-aligned  -> custom kernel (one-shot <=512KB,
                                    two-shot above — chosen in C++)
  otherwise                     -> torch.distributed (RCCL)
"""

from __future__ import annotations

import logging
import os

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

# Covers decode all-reduces ([tokens, hidden] bf16; 1024x8192x2 = 16 MiB)
# and the logits all-gather chunk ([tokens, vocab/tp]; up to ~126 MiB at
# llama-3 vocab TP=2, batch 1024) so the whole TP decode step is
# graph-capturable. Buffer cost = 3x this per GPU — noise on 288 GB.
DEFAULT_MAX_BYTES = int(
    os.environ.get("VLLM_AMD_CAR_MAX_BYTES", 160 * 1024 * 1024))


class CustomCollectives:
    """Per-process handle to the comms.hip state (one TP group)."""

    def __init__(self, rank_in_group: int, world_size: int,
                 max_bytes: int = DEFAULT_MAX_BYTES):
        from vllm_amd.ops import hip_ops  # loads _C.so

        assert hip_ops is not None
        _C = torch.ops.vllm_amd
        self._C = _C
        self.world_size = world_size
        self.rank_in_group = rank_in_group
        self.max_bytes = max_bytes
        self.handle = _C.car_init(rank_in_group, world_size, max_bytes)
        self.disabled = False

    def connect(self, cpu_group) -> None:
        all_handles = [torch.empty_like(self.handle)
                       for _ in range(self.world_size)]
        dist.all_gather(all_handles, self.handle, group=cpu_group)
        self._C.car_connect(torch.stack(all_handles))

    def should_use(self, t: torch.Tensor) -> bool:
        if self.disabled or not t.is_cuda or not t.is_contiguous():
            return False
        if t.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            return False
        nbytes = t.numel() * t.element_size()
        return nbytes % 16 == 0 and nbytes <= self.max_bytes

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        self._C.car_all_reduce(t)
        return t

    def can_gather(self, t: torch.Tensor) -> bool:
        if self.disabled or not t.is_cuda:
            return False
        if t.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            return False
        nbytes = t.numel() * t.element_size()
        return nbytes % 16 == 0 and nbytes <= self.max_bytes

    def can_reduce_scatter(self, t: torch.Tensor) -> bool:
        if self.disabled or not t.is_cuda:
            return False
        if t.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            return False
        nbytes = t.numel() * t.element_size()
        return (nbytes % 16 == 0 and nbytes <= self.max_bytes
                and (nbytes // 16) % self.world_size == 0
                and t.shape[0] % self.world_size == 0)

    def reduce_scatter_rows(self, t: torch.Tensor) -> torch.Tensor:
        t = t.contiguous()
        out_shape = (t.shape[0] // self.world_size,) + tuple(t.shape[1:])
        out = torch.empty(out_shape, dtype=t.dtype, device=t.device)
        self._C.car_reduce_scatter(out.view(-1), t.view(-1))
        return out

    def all_gather_flat(self, t: torch.Tensor) -> torch.Tensor:
        """Returns [world, *t.shape]."""
        t = t.contiguous()
        out = torch.empty((self.world_size,) + tuple(t.shape),
                          dtype=t.dtype, device=t.device)
        self._C.car_all_gather(out.view(-1), t.view(-1))
        return out

    def error(self) -> int:
        return int(self._C.car_error())

    def destroy(self) -> None:
        if not self.disabled:
            self._C.car_destroy()
            self.disabled = True


def try_init_custom_collectives(rank_in_group: int, world_size: int,
                                cpu_group) -> "CustomCollectives | None":
    """Best-effort init; falls back to None (RCCL-only) on any failure.
    All ranks must agree — the boolean is all-reduced over the CPU group
    so no rank is left spinning on a peer that failed."""
    ok = True
    inst = None
    if os.environ.get("VLLM_AMD_DISABLE_CUSTOM_AR", "0") == "1":
        ok = False
    if not torch.cuda.is_available() or not (2 <= world_size <= 8):
        ok = False
    if ok:
        try:
            inst = CustomCollectives(rank_in_group, world_size)
        except Exception as e:  # noqa: BLE001
            logger.warning("custom all-reduce init failed (%s); "
                           "falling back to RCCL", e)
            ok = False
    # Agree before the handle exchange so no rank blocks in all_gather
    # against a peer whose car_init failed.
    flag = torch.tensor([1 if ok else 0], dtype=torch.int32)
    dist.all_reduce(flag, op=dist.ReduceOp.MIN, group=cpu_group)
    if int(flag.item()) == 0:
        if inst is not None:
            inst.destroy()
        return None
    try:
        inst.connect(cpu_group)
        ok = _self_test(inst, world_size)
    except Exception as e:  # noqa: BLE001
        # Handle-open failure can be rank-local (IPC limits); every rank
        # must agree again or peers would spin forever in the kernel.
        logger.warning("custom all-reduce connect failed (%s)", e)
        ok = False
    flag = torch.tensor([1 if ok else 0], dtype=torch.int32)
    dist.all_reduce(flag, op=dist.ReduceOp.MIN, group=cpu_group)
    if int(flag.item()) == 0:
        logger.warning("custom all-reduce self-test failed; using RCCL")
        inst.destroy()
        return None
    return inst


def _self_test(inst: CustomCollectives, world_size: int) -> bool:
    """Verify the IPC kernels against known sums on THIS topology before
    trusting them with traffic: one-shot, two-shot and all-gather sizes,
    three rounds each (parity/ack protocol), plus the device error flag.
    Rank- and position-dependent payloads catch peer-mapping swaps and
    chunk-offset bugs, not just visibility failures. A failure
    downgrades the whole group to RCCL instead of corrupting traffic."""
    r = inst.rank_in_group
    sum_r = world_size * (world_size - 1) / 2.0
    for nbytes in (4096, 1 << 20):  # one-shot and two-shot regimes
        n = nbytes // 4
        pos = torch.arange(n, dtype=torch.float32, device="cuda") % 17
        for round_i in range(3):
            t = pos + 1000.0 * r + round_i
            inst.all_reduce(t)
            torch.cuda.synchronize()
            expect = (pos + round_i) * world_size + 1000.0 * sum_r
            if not torch.allclose(t, expect):
                return False
    g = inst.all_gather_flat(
        torch.full((1024,), 1.0 + r, dtype=torch.float32, device="cuda"))
    torch.cuda.synchronize()
    for p in range(world_size):
        if not torch.allclose(g[p], torch.full_like(g[p], 1.0 + p)):
            return False
    return inst.error() == 0
