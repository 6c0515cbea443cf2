"""Shared-memory broadcast ring (executor/shm_queue.py) unit tests:
single-writer/multi-reader ordering, wrap-around, backpressure, and a
cross-process smoke (role of the reference's shm_broadcast tests)."""

import multiprocessing as mp

from vllm_amd.executor.shm_queue import ShmRing


def test_inproc_order_and_wrap():
    # Single-threaded test: keep (write burst + reader-1 lag) well under
    # the ring size or the writer would block on its own lagging reader.
    ring = ShmRing(2, size=4096)
    msgs = [{"i": i, "pad": "x" * (i * 7 % 100)} for i in range(200)]
    # Interleave: write a few, read from both readers (reader 1 lags).
    wrote = 0
    read0 = []
    read1 = []
    for batch in range(40):
        for _ in range(5):
            ring.write(msgs[wrote])
            wrote += 1
        while len(read0) < wrote:
            read0.append(ring.read(0))
        # reader 1 catches up every other batch (exercises backpressure)
        if batch % 2 == 1:
            while len(read1) < wrote:
                read1.append(ring.read(1))
    while len(read1) < wrote:
        read1.append(ring.read(1))
    assert read0 == msgs and read1 == msgs
    ring.close()


def test_record_too_large():
    ring = ShmRing(1, size=4096)
    try:
        import pytest

        with pytest.raises(ValueError):
            ring.write(b"y" * 8192)
    finally:
        ring.close()


def _reader_proc(name, size, n, rid, q):
    try:
        ring = ShmRing(2, size=size, name=name, create=False)
        out = [ring.read(rid) for _ in range(n)]
        ring.shm.close()
        q.put(("ok", out))
    except Exception as e:  # noqa: BLE001
        q.put(("err", repr(e)))


def test_cross_process_broadcast():
    ctx = mp.get_context("spawn")
    ring = ShmRing(2, size=1 << 16)
    n = 300
    q = ctx.Queue()
    procs = [ctx.Process(target=_reader_proc,
                         args=(ring.name, 1 << 16, n, r, q))
             for r in range(2)]
    for p in procs:
        p.start()
    msgs = [list(range(i % 50)) for i in range(n)]
    for m in msgs:
        ring.write(m)
    oks = 0
    for _ in range(2):
        status, payload = q.get(timeout=60)
        assert status == "ok", payload
        assert payload == msgs
        oks += 1
    for p in procs:
        p.join(timeout=30)
    ring.close()
    assert oks == 2
