"""KV cache event publishing (role of the reference's
vllm/distributed/kv_events.py:560): external cache-aware routers
subscribe to the engine's prefix-cache state — which content-hashed
blocks are resident — to steer requests at the fleet level.

Wire format: newline-delimited JSON over a plain TCP socket (stdlib
only, same stance as tracing.py's OTLP exporter — no broker
dependency). Events are batched by a background thread; publishing is
strictly fire-and-forget and never blocks or fails the engine.

Event schema (one JSON object per line):
    {"event": "block_stored",  "block_hashes": [...], "token_ids_len":
     [...], "ts": ...}
    {"event": "block_removed", "block_hashes": [...], "ts": ...}
    {"event": "all_blocks_cleared", "ts": ...}
"""

from __future__ import annotations

import json
import logging
import queue
import socket
import threading
import time

logger = logging.getLogger(__name__)


class KVEventPublisher:
    """Batched background publisher; attach() wires it to a BlockPool."""

    def __init__(self, endpoint: str, flush_interval_s: float = 0.1,
                 max_queue: int = 65536):
        host, _, port = endpoint.rpartition(":")
        self.addr = (host or "127.0.0.1", int(port))
        self.q: "queue.Queue" = queue.Queue(maxsize=max_queue)
        self._stop = threading.Event()
        self.flush_interval_s = flush_interval_s
        self.dropped = 0
        self._thread = threading.Thread(target=self._run, daemon=True)
        self._thread.start()

    # -- producer side (engine thread; never blocks) ---------------------
    def _emit(self, obj: dict) -> None:
        obj["ts"] = time.time()
        try:
            self.q.put_nowait(obj)
        except queue.Full:
            self.dropped += 1

    def block_stored(self, block_hashes: list[int],
                     token_lens: list[int]) -> None:
        self._emit({"event": "block_stored", "block_hashes": block_hashes,
                    "token_ids_len": token_lens})

    def block_removed(self, block_hashes: list[int]) -> None:
        self._emit({"event": "block_removed", "block_hashes": block_hashes})

    def all_blocks_cleared(self) -> None:
        self._emit({"event": "all_blocks_cleared"})

    # -- consumer side ---------------------------------------------------
    def _run(self) -> None:
        sock = None
        while not self._stop.is_set():
            batch = []
            try:
                batch.append(self.q.get(timeout=self.flush_interval_s))
            except queue.Empty:
                continue
            while len(batch) < 4096:
                try:
                    batch.append(self.q.get_nowait())
                except queue.Empty:
                    break
            payload = "".join(
                json.dumps(o, separators=(",", ":")) + "\n" for o in batch
            ).encode()
            for _attempt in range(2):
                try:
                    if sock is None:
                        sock = socket.create_connection(self.addr,
                                                        timeout=2.0)
                    sock.sendall(payload)
                    break
                except OSError:
                    if sock is not None:
                        try:
                            sock.close()
                        except OSError:
                            pass
                    sock = None
        if sock is not None:
            try:
                sock.close()
            except OSError:
                pass

    def close(self) -> None:
        self._stop.set()
        self._thread.join(timeout=2.0)

    # -- wiring ----------------------------------------------------------
    def attach(self, block_pool) -> None:
        """Wrap the pool's cache/evict paths with event emission."""
        pub = self
        orig_cache = block_pool.cache_full_blocks
        orig_get_new = block_pool.get_new_blocks
        orig_reset = block_pool.reset_prefix_cache

        def cache_full_blocks(blocks, block_hashes, num_cached, num_full):
            orig_cache(blocks, block_hashes, num_cached, num_full)
            hs = [block_hashes[i].value
                  for i in range(num_cached, num_full)]
            if hs:
                pub.block_stored(
                    hs, [len(block_hashes[i].token_ids)
                         for i in range(num_cached, num_full)])

        def get_new_blocks(num_blocks):
            # Eviction happens inside get_new_blocks when a cached free
            # block is reused; detect by hash presence before/after.
            evicted: list[int] = []
            orig_on_evict = block_pool.on_evict

            def on_evict(block_hash, block_id):
                evicted.append(block_hash.value)
                if orig_on_evict is not None:
                    orig_on_evict(block_hash, block_id)

            block_pool.on_evict = on_evict
            try:
                out = orig_get_new(num_blocks)
            finally:
                block_pool.on_evict = orig_on_evict
            if evicted:
                pub.block_removed(evicted)
            return out

        def reset_prefix_cache():
            ok = orig_reset()
            if ok:
                pub.all_blocks_cleared()
            return ok

        block_pool.cache_full_blocks = cache_full_blocks
        block_pool.get_new_blocks = get_new_blocks
        block_pool.reset_prefix_cache = reset_prefix_cache
