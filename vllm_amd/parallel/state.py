"""Distributed runtime state: TP groups over RCCL/xGMI.

Role of the reference's vllm/distributed/parallel_state.py:380
(GroupCoordinator / initialize_model_parallel), restructured for the
MI355X topology: one process per GPU, torch.distributed with the
"nccl" backend (RCCL on ROCm) over xGMI for the data plane, "gloo"
for CPU-only control paths and tests.

Single-process (tp=1, world=1) works with no init at all — every
collective becomes a no-op — so CPU unit tests and the offline LLM
path need no distributed bootstrap.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class GroupCoordinator:
    """One communication group (TP today; PP/DP/EP reuse the same shape)."""

    rank: int
    world_size: int
    rank_in_group: int
    device_group: Optional[dist.ProcessGroup] = None
    cpu_group: Optional[dist.ProcessGroup] = None
    # Custom xGMI collectives (parallel/custom_ar.py); size-gated in
    # front of RCCL, hipGraph-capturable. TP group only.
    comms: Optional[object] = None

    @property
    def is_first_rank(self) -> bool:
        return self.rank_in_group == 0

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.world_size == 1:
            return t
        if self.comms is not None and self.comms.should_use(t):
            return self.comms.all_reduce(t)
        if t.is_cuda:
            r = self._maybe_rccl()
            if r is not None:
                r.all_reduce(t)
                return t
        dist.all_reduce(t, group=self.device_group)
        return t

    def all_gather(self, t: torch.Tensor, dim: int = -1) -> torch.Tensor:
        if self.world_size == 1:
            return t
        if dim < 0:
            dim += t.dim()
        if self.comms is not None and self.comms.can_gather(t):
            out = self.comms.all_gather_flat(t)
        else:
            # Gather flat on dim 0 (collective layout), then view as
            # [world, *shape] to concat on `dim`.
            out = torch.empty(
                (self.world_size * t.shape[0],) + tuple(t.shape[1:]),
                dtype=t.dtype, device=t.device,
            )
            dist.all_gather_into_tensor(out, t.contiguous(),
                                        group=self.device_group)
            out = out.view((self.world_size,) + tuple(t.shape))
        if dim == 0:
            return out.reshape(-1, *t.shape[1:])
        pieces = out.unbind(0)
        return torch.cat(pieces, dim=dim)

    def reduce_scatter(self, t: torch.Tensor, dim: int = 0) -> torch.Tensor:
        if self.world_size == 1:
            return t
        assert dim == 0
        if self.comms is not None and self.comms.can_reduce_scatter(t):
            return self.comms.reduce_scatter_rows(t)
        out_shape = (t.shape[0] // self.world_size,) + tuple(t.shape[1:])
        out = torch.empty(out_shape, dtype=t.dtype, device=t.device)
        dist.reduce_scatter_tensor(out, t.contiguous(), group=self.device_group)
        return out

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.world_size == 1:
            return t
        dist.broadcast(t, src=src, group=self.device_group)
        return t

    def broadcast_object(self, obj=None, src: int = 0):
        if self.world_size == 1:
            return obj
        holder = [obj]
        dist.broadcast_object_list(holder, src=src,
                                   group=self.cpu_group or self.device_group)
        return holder[0]

    def all_to_all(self, t: torch.Tensor) -> torch.Tensor:
        """Equal-split all-to-all along dim 0 (EP dispatch/combine)."""
        if self.world_size == 1:
            return t
        out = torch.empty_like(t)
        dist.all_to_all_single(out, t.contiguous(), group=self.device_group)
        return out

    def barrier(self) -> None:
        if self.world_size > 1:
            dist.barrier(group=self.cpu_group or self.device_group)


# Module-level groups (initialized once per process).
_TP: Optional[GroupCoordinator] = None
_EP: Optional[GroupCoordinator] = None
_WORLD: Optional[GroupCoordinator] = None

_SINGLE = GroupCoordinator(rank=0, world_size=1, rank_in_group=0)
_PP: Optional[GroupCoordinator] = None
_REPLICA: Optional[GroupCoordinator] = None


def get_replica_group() -> GroupCoordinator:
    """Engine lockstep domain: the pp*tp model replica (== the TP group
    when pp == 1)."""
    if _REPLICA is not None:
        return _REPLICA
    return get_tp_group()


def get_pp_group() -> GroupCoordinator:
    return _PP if _PP is not None else _SINGLE


def get_pp_rank() -> int:
    return get_pp_group().rank_in_group


def get_pp_world_size() -> int:
    return get_pp_group().world_size


def is_first_pp_rank() -> bool:
    return get_pp_rank() == 0


def is_last_pp_rank() -> bool:
    return get_pp_rank() == get_pp_world_size() - 1


def pp_send(t: torch.Tensor, dst_in_group: int) -> None:
    g = get_pp_group()
    dist.send(t.contiguous(), dst=g.group_ranks[dst_in_group],
              group=g.device_group)


def pp_recv(t: torch.Tensor, src_in_group: int) -> None:
    g = get_pp_group()
    dist.recv(t, src=g.group_ranks[src_in_group], group=g.device_group)


def pp_broadcast_object(obj=None, src_in_group: int = 0):
    """Object broadcast within the PP group (src is a PP-group rank)."""
    g = get_pp_group()
    if g.world_size == 1:
        return obj
    return g.broadcast_object(obj, src=g.group_ranks[src_in_group])


def pp_layer_range(num_layers: int) -> tuple[int, int]:
    """[lo, hi) decoder layers owned by this PP stage (remainder layers
    go to the EARLY stages, which also hold the embedding)."""
    pp, r = get_pp_world_size(), get_pp_rank()
    per, extra = divmod(num_layers, pp)
    lo = r * per + min(r, extra)
    hi = lo + per + (1 if r < extra else 0)
    return lo, hi


def get_tp_group() -> GroupCoordinator:
    return _TP if _TP is not None else _SINGLE


def get_ep_group() -> GroupCoordinator:
    return _EP if _EP is not None else _SINGLE


def get_world_group() -> GroupCoordinator:
    return _WORLD if _WORLD is not None else _SINGLE


def get_tp_rank() -> int:
    return get_tp_group().rank_in_group


def get_tp_world_size() -> int:
    return get_tp_group().world_size


def is_initialized() -> bool:
    return _TP is not None


def init_distributed(
    tensor_parallel_size: int = 1,
    pipeline_parallel_size: int = 1,
    backend: str = "auto",
    rank: Optional[int] = None,
    world_size: Optional[int] = None,
    local_rank: Optional[int] = None,
) -> None:
    """Initialize torch.distributed and the model-parallel groups.

    Reads RANK/WORLD_SIZE/LOCAL_RANK/MASTER_* from env when launched by
    torchrun (the bench.py contract). world = TP in v1 (DP is one engine
    per replica; PP later).
    """
    global _TP, _EP, _WORLD
    rank = rank if rank is not None else int(os.environ.get("RANK", "0"))
    world_size = (
        world_size
        if world_size is not None
        else int(os.environ.get("WORLD_SIZE", "1"))
    )
    local_rank = (
        local_rank
        if local_rank is not None
        else int(os.environ.get("LOCAL_RANK", str(rank)))
    )

    if (world_size == 1 and tensor_parallel_size == 1
            and pipeline_parallel_size == 1):
        return  # single-process fast path; no distributed state at all

    if backend == "auto":
        backend = "nccl" if torch.cuda.is_available() else "gloo"

    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world_size
        )

    cpu_group = (
        dist.new_group(backend="gloo") if backend != "gloo" else None
    )

    model_world = tensor_parallel_size * pipeline_parallel_size
    assert world_size % model_world == 0
    # TP groups are contiguous ranks (adjacent ranks share xGMI hops);
    # PP groups stride across TP groups within one model replica:
    # rank = replica*pp*tp + pp_rank*tp + tp_rank.
    global _PP
    tp_group = None
    my_tp_ranks = None
    for start in range(0, world_size, tensor_parallel_size):
        ranks = list(range(start, start + tensor_parallel_size))
        g = dist.new_group(ranks, backend=backend)
        if rank in ranks:
            tp_group = g
            my_tp_ranks = ranks
    # Per-TP-group gloo group for the IPC handle exchange and object
    # collectives (the global gloo group would collide across replicas).
    tp_cpu_group = None
    if backend != "gloo" and tensor_parallel_size > 1:
        for start in range(0, world_size, tensor_parallel_size):
            ranks = list(range(start, start + tensor_parallel_size))
            g = dist.new_group(ranks, backend="gloo")
            if rank in ranks:
                tp_cpu_group = g
    _TP = GroupCoordinator(
        rank=rank,
        world_size=tensor_parallel_size,
        rank_in_group=my_tp_ranks.index(rank),
        device_group=tp_group,
        cpu_group=tp_cpu_group or cpu_group,
    )
    if (tensor_parallel_size > 1 and backend == "nccl"
            and torch.cuda.is_available()):
        from vllm_amd.parallel.custom_ar import try_init_custom_collectives
        _TP.comms = try_init_custom_collectives(
            _TP.rank_in_group, tensor_parallel_size,
            tp_cpu_group or cpu_group)
    if pipeline_parallel_size > 1:
        pp_group = None
        my_pp_ranks = None
        for rep in range(0, world_size, model_world):
            for t in range(tensor_parallel_size):
                ranks = [rep + p * tensor_parallel_size + t
                         for p in range(pipeline_parallel_size)]
                g = dist.new_group(ranks, backend=backend)
                if rank in ranks:
                    pp_group = g
                    my_pp_ranks = ranks
        # cpu_group stays None: object collectives must run on THIS
        # pp group (the global gloo group would collide across groups).
        _PP = GroupCoordinator(
            rank=rank,
            world_size=pipeline_parallel_size,
            rank_in_group=my_pp_ranks.index(rank),
            device_group=pp_group,
        )
        _PP.group_ranks = my_pp_ranks
        # The engine lockstep (scheduler-broadcast) domain with PP is the
        # whole model replica: the contiguous pp*tp block.
        global _REPLICA
        rep_group = None
        my_rep_ranks = None
        for rep in range(0, world_size, model_world):
            ranks = list(range(rep, rep + model_world))
            g = dist.new_group(ranks, backend=backend)
            cg = (dist.new_group(ranks, backend="gloo")
                  if backend != "gloo" else None)
            if rank in ranks:
                rep_group = g
                my_rep_ranks = ranks
                rep_cpu = cg
        _REPLICA = GroupCoordinator(
            rank=rank,
            world_size=model_world,
            rank_in_group=my_rep_ranks.index(rank),
            device_group=rep_group,
            cpu_group=rep_cpu,
        )
    _WORLD = GroupCoordinator(
        rank=rank,
        world_size=world_size,
        rank_in_group=rank,
        device_group=dist.group.WORLD,
        cpu_group=cpu_group,
    )
    # EP group == world group in v1 (experts sharded across all ranks).
    _EP = _WORLD


def destroy_distributed() -> None:
    global _TP, _EP, _WORLD, _PP, _REPLICA
    if _TP is not None and _TP.comms is not None:
        _TP.comms.destroy()
    _TP = _EP = _WORLD = _PP = _REPLICA = None
    if dist.is_initialized():
        dist.destroy_process_group()


# Convenience wrappers used by layers.
def tensor_model_parallel_all_reduce(t: torch.Tensor) -> torch.Tensor:
    return get_tp_group().all_reduce(t)


def sp_all_gather_rows(t: torch.Tensor) -> torch.Tensor:
    """Sequence parallelism: sharded rows [T/tp, H] -> full [T, H]."""
    return get_tp_group().all_gather(t, dim=0)


def sp_reduce_scatter_rows(t: torch.Tensor) -> torch.Tensor:
    """Sequence parallelism: partial full rows [T, H] -> reduced shard
    [T/tp, H] (this rank's contiguous row chunk)."""
    return get_tp_group().reduce_scatter(t, dim=0)


def tensor_model_parallel_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    return get_tp_group().all_gather(t, dim=dim)
