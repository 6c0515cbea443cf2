"""Direct ctypes binding to librccl (role of the reference's
vllm/distributed/device_communicators/pynccl_wrapper.py:167, which
dlopens the same C ABI — RCCL exports NCCL's symbols).

Why bypass torch.distributed: collectives issued directly on a chosen
HIP stream are hipGraph-CAPTURABLE and skip the ProcessGroup dispatch
layer. The decode hot path uses the custom IPC kernels (comms.hip) for
graph-captured TP steps; this layer covers sizes beyond the IPC buffer
and gives stream-targeted send/recv. Opt-in on the dispatch path via
VLLM_AMD_USE_RCCL_DIRECT=1 (torch.distributed remains the default
fallback).

The unique-id handshake rides the existing CPU (gloo) group.
"""

from __future__ import annotations

import ctypes
import logging
import os
from typing import Optional

import torch

logger = logging.getLogger(__name__)

# ncclDataType_t
NCCL_INT8 = 0
NCCL_UINT8 = 1
NCCL_INT32 = 2
NCCL_UINT32 = 3
NCCL_INT64 = 4
NCCL_UINT64 = 5
NCCL_FLOAT16 = 6
NCCL_FLOAT32 = 7
NCCL_FLOAT64 = 8
NCCL_BFLOAT16 = 9

# ncclRedOp_t
NCCL_SUM = 0
NCCL_PROD = 1
NCCL_MAX = 2
NCCL_MIN = 3
NCCL_AVG = 4

_DTYPE = {
    torch.int8: NCCL_INT8,
    torch.uint8: NCCL_UINT8,
    torch.int32: NCCL_INT32,
    torch.int64: NCCL_INT64,
    torch.float16: NCCL_FLOAT16,
    torch.float32: NCCL_FLOAT32,
    torch.float64: NCCL_FLOAT64,
    torch.bfloat16: NCCL_BFLOAT16,
}

NCCL_UNIQUE_ID_BYTES = 128


class ncclUniqueId(ctypes.Structure):
    _fields_ = [("internal", ctypes.c_byte * NCCL_UNIQUE_ID_BYTES)]


def _candidate_paths() -> list[str]:
    out = []
    env = os.environ.get("VLLM_AMD_RCCL_PATH")
    if env:
        out.append(env)
    torch_lib = os.path.join(os.path.dirname(torch.__file__), "lib",
                             "librccl.so")
    out += [torch_lib, "librccl.so", "/opt/rocm/lib/librccl.so"]
    return out


class RCCLLibrary:
    """Lazily-loaded librccl with the exported functions typed."""

    _FUNCS = [
        ("ncclGetErrorString", ctypes.c_char_p, [ctypes.c_int]),
        ("ncclGetUniqueId", ctypes.c_int, [ctypes.POINTER(ncclUniqueId)]),
        ("ncclCommInitRank", ctypes.c_int,
         [ctypes.POINTER(ctypes.c_void_p), ctypes.c_int, ncclUniqueId,
          ctypes.c_int]),
        ("ncclCommDestroy", ctypes.c_int, [ctypes.c_void_p]),
        ("ncclCommCount", ctypes.c_int,
         [ctypes.c_void_p, ctypes.POINTER(ctypes.c_int)]),
        ("ncclAllReduce", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int,
          ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclAllGather", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int,
          ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclReduceScatter", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int,
          ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclBroadcast", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int,
          ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclSend", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int, ctypes.c_int,
          ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclRecv", ctypes.c_int,
         [ctypes.c_void_p, ctypes.c_size_t, ctypes.c_int, ctypes.c_int,
          ctypes.c_void_p, ctypes.c_void_p]),
        ("ncclGroupStart", ctypes.c_int, []),
        ("ncclGroupEnd", ctypes.c_int, []),
    ]

    def __init__(self, path: Optional[str] = None):
        last_err = None
        self.lib = None
        for p in ([path] if path else _candidate_paths()):
            try:
                self.lib = ctypes.CDLL(p)
                self.path = p
                break
            except OSError as e:
                last_err = e
        if self.lib is None:
            raise OSError(f"librccl not found: {last_err}")
        for name, restype, argtypes in self._FUNCS:
            fn = getattr(self.lib, name)
            fn.restype = restype
            fn.argtypes = argtypes
            setattr(self, name, fn)

    def check(self, result: int) -> None:
        if result != 0:
            msg = self.ncclGetErrorString(result).decode()
            raise RuntimeError(f"RCCL error {result}: {msg}")


class RCCLCommunicator:
    """One communicator over a set of ranks; collectives issue on an
    explicit HIP stream (default: torch's current stream), which makes
    them hipGraph-capturable."""

    def __init__(self, rank: int, world_size: int, cpu_group=None,
                 unique_id: Optional[bytes] = None,
                 library: Optional[RCCLLibrary] = None):
        import torch.distributed as dist

        self.lib = library or RCCLLibrary()
        self.rank = rank
        self.world_size = world_size
        uid = ncclUniqueId()
        if unique_id is not None:
            ctypes.memmove(uid.internal, unique_id, NCCL_UNIQUE_ID_BYTES)
        else:
            if rank == 0:
                self.lib.check(self.lib.ncclGetUniqueId(ctypes.byref(uid)))
            if world_size > 1:
                t = torch.tensor(list(bytes(uid.internal)),
                                 dtype=torch.uint8)
                dist.broadcast(t, src=0, group=cpu_group)
                ctypes.memmove(uid.internal, bytes(t.tolist()),
                               NCCL_UNIQUE_ID_BYTES)
        self.comm = ctypes.c_void_p()
        self.lib.check(self.lib.ncclCommInitRank(
            ctypes.byref(self.comm), world_size, uid, rank))

    def _stream(self, stream) -> ctypes.c_void_p:
        if stream is None:
            stream = torch.cuda.current_stream()
        return ctypes.c_void_p(stream.cuda_stream)

    def all_reduce(self, t: torch.Tensor, op: int = NCCL_SUM,
                   stream=None) -> None:
        assert t.is_cuda and t.is_contiguous()
        self.lib.check(self.lib.ncclAllReduce(
            ctypes.c_void_p(t.data_ptr()), ctypes.c_void_p(t.data_ptr()),
            t.numel(), _DTYPE[t.dtype], op, self.comm, self._stream(stream)))

    def all_gather(self, out: torch.Tensor, t: torch.Tensor,
                   stream=None) -> None:
        assert out.numel() == t.numel() * self.world_size
        self.lib.check(self.lib.ncclAllGather(
            ctypes.c_void_p(t.data_ptr()), ctypes.c_void_p(out.data_ptr()),
            t.numel(), _DTYPE[t.dtype], self.comm, self._stream(stream)))

    def reduce_scatter(self, out: torch.Tensor, t: torch.Tensor,
                       op: int = NCCL_SUM, stream=None) -> None:
        assert t.numel() == out.numel() * self.world_size
        self.lib.check(self.lib.ncclReduceScatter(
            ctypes.c_void_p(t.data_ptr()), ctypes.c_void_p(out.data_ptr()),
            out.numel(), _DTYPE[t.dtype], op, self.comm,
            self._stream(stream)))

    def broadcast(self, t: torch.Tensor, root: int, stream=None) -> None:
        self.lib.check(self.lib.ncclBroadcast(
            ctypes.c_void_p(t.data_ptr()), ctypes.c_void_p(t.data_ptr()),
            t.numel(), _DTYPE[t.dtype], root, self.comm,
            self._stream(stream)))

    def send(self, t: torch.Tensor, peer: int, stream=None) -> None:
        self.lib.check(self.lib.ncclSend(
            ctypes.c_void_p(t.data_ptr()), t.numel(), _DTYPE[t.dtype], peer,
            self.comm, self._stream(stream)))

    def recv(self, t: torch.Tensor, peer: int, stream=None) -> None:
        self.lib.check(self.lib.ncclRecv(
            ctypes.c_void_p(t.data_ptr()), t.numel(), _DTYPE[t.dtype], peer,
            self.comm, self._stream(stream)))

    def destroy(self) -> None:
        if self.comm:
            self.lib.ncclCommDestroy(self.comm)
            self.comm = ctypes.c_void_p()
