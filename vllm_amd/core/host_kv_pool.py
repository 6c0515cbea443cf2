"""Host (CPU RAM) tier for evicted prefix-cache KV blocks.

Role of the reference's CPU KV offloading (vllm kv_offload / CPUOffloading
connector), re-designed for this engine's prefix cache: when the GPU
BlockPool evicts a content-hashed free block to reuse it, the block's KV
is copied D2H into a pinned host pool keyed by the same content hash; a
later prefix-cache miss that hits the host tier materializes a fresh GPU
block with an H2D copy instead of recomputing the prefill.

This module is pure accounting (hash -> host slot with LRU); the actual
copies are ordered ops in SchedulerOutput executed by the model runner
before the forward (swap-outs strictly before swap-ins, so an
evict-then-hit within one schedule round is correct).
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Optional

from vllm_amd.core.kv_cache_utils import BlockHash


class HostKVPool:

    def __init__(self, num_slots: int) -> None:
        self.num_slots = num_slots
        self.free_slots = list(range(num_slots - 1, -1, -1))
        # hash value -> (full hash, slot). LRU order: oldest first.
        self.entries: "OrderedDict[int, tuple[BlockHash, int]]" = \
            OrderedDict()
        # Slots with a copy scheduled but not yet executed this round —
        # never evict these.
        self.in_flight: set[int] = set()

    def lookup(self, h: BlockHash) -> Optional[int]:
        ent = self.entries.get(h.value)
        if ent is None or ent[0].token_ids != h.token_ids:
            return None
        self.entries.move_to_end(h.value)
        return ent[1]

    def put(self, h: BlockHash) -> Optional[int]:
        """Reserve a slot for this hash (evicting the LRU entry if full).
        Returns None only when every slot is in flight."""
        old = self.entries.pop(h.value, None)
        if old is not None:
            slot = old[1]
        elif self.free_slots:
            slot = self.free_slots.pop()
        else:
            # Evict the least-recently-used entry whose slot is free to
            # reuse this round.
            slot = None
            for key, (eh, s) in self.entries.items():
                if s not in self.in_flight:
                    slot = s
                    del self.entries[key]
                    break
            if slot is None:
                return None
        self.entries[h.value] = (h, slot)
        self.in_flight.add(slot)
        return slot

    def end_round(self) -> None:
        self.in_flight.clear()

    def clear(self) -> None:
        self.entries.clear()
        self.free_slots = list(range(self.num_slots - 1, -1, -1))
        self.in_flight.clear()
