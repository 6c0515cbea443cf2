"""Single-writer / multi-reader shared-memory broadcast ring.

Role of the reference's shm MessageQueue
(vllm/distributed/device_communicators/shm_broadcast.py:464): the
engine pickles each SchedulerOutput ONCE into a shared-memory ring and
every worker reads it in place — at TP=8 the pipe control plane costs 8
pickles + 8 socket writes per step, which shows up once decode steps are
a few ms.

Layout (bytes):
    [0:8)                     writer position (monotonic, u64)
    [64 + 64*r : 64 + 64*r+8) reader r position (monotonic, u64)
    [HDR:HDR+size)            ring payload area

Records: 8-byte little-endian length, then the pickle. A length of
0xFFFFFFFFFFFFFFFF is a wrap marker: both sides jump to the next ring
start. Positions are monotonic byte offsets (mod size for addressing),
so full/empty are unambiguous. The writer blocks (spin + short sleep)
while the slowest reader is more than one ring behind; readers spin hot
for a short window then back off to a 50 us sleep — decode steps arrive
every few ms, so the hot window catches the common case.
"""

from __future__ import annotations

import pickle
import struct
import time
from multiprocessing import shared_memory

_WRAP = 0xFFFFFFFFFFFFFFFF
_U64 = struct.Struct("<Q")
_HOT_SPIN_S = 200e-6
_IDLE_SLEEP_S = 50e-6


class ShmRing:
    HDR_READERS_OFF = 64
    READER_STRIDE = 64

    def __init__(self, num_readers: int, size: int = 8 << 20,
                 name: str | None = None, create: bool = True):
        self.num_readers = num_readers
        self.size = size
        self.data_off = self.HDR_READERS_OFF + self.READER_STRIDE * num_readers
        total = self.data_off + size
        if create:
            self.shm = shared_memory.SharedMemory(create=True, size=total)
            self.shm.buf[: self.data_off] = bytes(self.data_off)
        else:
            assert name is not None
            self.shm = shared_memory.SharedMemory(name=name)
        self.name = self.shm.name
        self._creator = create

    # -- position accessors (u64 in the header) -------------------------
    def _get(self, off: int) -> int:
        return _U64.unpack_from(self.shm.buf, off)[0]

    def _set(self, off: int, v: int) -> None:
        _U64.pack_into(self.shm.buf, off, v)

    def _reader_off(self, r: int) -> int:
        return self.HDR_READERS_OFF + self.READER_STRIDE * r

    # -- writer ----------------------------------------------------------
    def _min_reader(self) -> int:
        return min(self._get(self._reader_off(r))
                   for r in range(self.num_readers))

    def _wait_space(self, need: int, health_check=None) -> None:
        deadline = time.monotonic() + _HOT_SPIN_S
        while True:
            wpos = self._get(0)
            if wpos + need - self._min_reader() <= self.size:
                return
            if time.monotonic() > deadline:
                if health_check is not None:
                    health_check()
                time.sleep(_IDLE_SLEEP_S)
                deadline = time.monotonic() + _HOT_SPIN_S

    def write(self, obj, health_check=None) -> None:
        payload = pickle.dumps(obj, protocol=pickle.HIGHEST_PROTOCOL)
        rec = 8 + len(payload)
        if rec + 8 > self.size:
            raise ValueError(f"record too large for ring: {rec}")
        wpos = self._get(0)
        at = wpos % self.size
        room = self.size - at
        if rec + 8 > room:
            # Not enough contiguous room for record + a following header:
            # emit a wrap marker and restart at ring start.
            self._wait_space(room + rec, health_check)
            if room >= 8:
                _U64.pack_into(self.shm.buf, self.data_off + at, _WRAP)
            wpos += room
            at = 0
        self._wait_space(rec, health_check)
        base = self.data_off + at
        _U64.pack_into(self.shm.buf, base, len(payload))
        self.shm.buf[base + 8: base + 8 + len(payload)] = payload
        self._set(0, wpos + rec)

    # -- reader ----------------------------------------------------------
    def read(self, reader_id: int, poll=None):
        roff = self._reader_off(reader_id)
        rpos = self._get(roff)
        deadline = time.monotonic() + _HOT_SPIN_S
        while self._get(0) <= rpos:
            if time.monotonic() > deadline:
                if poll is not None:
                    poll()
                time.sleep(_IDLE_SLEEP_S)
                deadline = time.monotonic() + _HOT_SPIN_S
        at = rpos % self.size
        base = self.data_off + at
        room = self.size - at
        length = _U64.unpack_from(self.shm.buf, base)[0] if room >= 8 else _WRAP
        if length == _WRAP:
            rpos += room
            self._set(roff, rpos)
            return self.read(reader_id, poll)
        payload = bytes(self.shm.buf[base + 8: base + 8 + length])
        self._set(roff, rpos + 8 + length)
        return pickle.loads(payload)

    def close(self) -> None:
        try:
            self.shm.close()
            if self._creator:
                self.shm.unlink()
        except Exception:  # noqa: BLE001
            pass
