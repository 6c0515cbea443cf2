"""Direct ctypes RCCL binding (parallel/rccl.py — reference
pynccl_wrapper.py role). CPU: library load + ABI surface; GPU: a
world-1 communicator's collectives are identities on real buffers."""

import pytest
import torch


def test_library_loads_and_uid():
    from vllm_amd.parallel.rccl import NCCL_UNIQUE_ID_BYTES, RCCLLibrary

    lib = RCCLLibrary()
    # every typed symbol resolved
    for name, _, _ in RCCLLibrary._FUNCS:
        assert getattr(lib, name) is not None
    from vllm_amd.parallel.rccl import ncclUniqueId
    uid = ncclUniqueId()
    import ctypes
    lib.check(lib.ncclGetUniqueId(ctypes.byref(uid)))
    assert len(bytes(uid.internal)) == NCCL_UNIQUE_ID_BYTES
    assert any(bytes(uid.internal))  # non-zero id


def test_dtype_map_covers_engine_dtypes():
    from vllm_amd.parallel import rccl

    for dt in (torch.bfloat16, torch.float16, torch.float32, torch.int64):
        assert dt in rccl._DTYPE


@pytest.mark.gpu
def test_world1_collectives_identity():
    from vllm_amd.parallel.rccl import RCCLCommunicator

    comm = RCCLCommunicator(0, 1)
    t = torch.randn(4096, device="cuda", dtype=torch.bfloat16)
    ref = t.clone()
    comm.all_reduce(t)
    torch.cuda.synchronize()
    assert torch.equal(t, ref)
    out = torch.empty_like(t)
    comm.all_gather(out, t)
    torch.cuda.synchronize()
    assert torch.equal(out, ref)
    rs = torch.empty(4096, device="cuda", dtype=torch.bfloat16)
    comm.reduce_scatter(rs, t)
    torch.cuda.synchronize()
    assert torch.equal(rs, ref)
    comm.destroy()


@pytest.mark.gpu
def test_world1_graph_capturable():
    """The point of the direct binding: collectives captured in a
    hipGraph replay correctly."""
    from vllm_amd.parallel.rccl import RCCLCommunicator

    comm = RCCLCommunicator(0, 1)
    x = torch.randn(1024, device="cuda", dtype=torch.float32)
    static = x.clone()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        comm.all_reduce(static)
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        comm.all_reduce(static)
        comm.all_reduce(static)
    static.copy_(x)
    g.replay()
    torch.cuda.synchronize()
    assert torch.allclose(static, x)  # world-1 sum = identity, twice
    comm.destroy()
