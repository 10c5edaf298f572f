"""Dynamic micro-batching for the single-request embedding API.

The reference served `/text` from nine single-threaded CPU replicas
(deployments.yaml:6); one MI355X process caps out around 1/p50 ~ 180
req/s that way even though the bulk path does thousands/s. This batcher
lets concurrent `/text` callers share GPU batches: requests enqueue,
one collector thread drains up to ``max_batch`` items (waiting at most
``window_ms`` after the first), encodes them through the wrapper's
batched path (sort-by-length, padded, masked pool — the same path
``/texts`` uses), and wakes each caller with its row.

Enabled by CI_SERVE_BATCH_MS > 0 (the flask app then runs threaded).
With a single in-flight caller the cost is one condvar hop; under
concurrency the throughput approaches the bulk rate.
"""
from __future__ import annotations

import queue
import threading
from typing import Optional

import numpy as np

__all__ = ["MicroBatcher"]


class _Slot:
    __slots__ = ("text", "event", "result", "error")

    def __init__(self, text: str):
        self.text = text
        self.event = threading.Event()
        self.result: Optional[np.ndarray] = None
        self.error: Optional[BaseException] = None


class MicroBatcher:
    def __init__(self, wrapper, window_ms: float = 4.0, max_batch: int = 64):
        self.wrapper = wrapper
        self.window_s = window_ms / 1e3
        self.max_batch = max_batch
        self.q: "queue.Queue[_Slot]" = queue.Queue()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="embed-microbatcher")
        self._thread.start()
        self.batches = 0          # observability: batches executed
        self.batched_requests = 0

    def close(self) -> None:
        self._stop.set()
        sentinel = _Slot("")
        sentinel.event.set()
        self.q.put(sentinel)  # wake the collector

    def embed(self, text: str, timeout: float = 120.0) -> np.ndarray:
        slot = _Slot(text)
        self.q.put(slot)
        if not slot.event.wait(timeout):
            raise TimeoutError("embedding batch timed out")
        if slot.error is not None:
            raise slot.error
        return slot.result

    # --- collector ---------------------------------------------------------
    def _collect(self) -> list:
        try:
            first = self.q.get(timeout=0.5)
        except queue.Empty:
            return []
        batch = [first]
        # wait up to window_ms for co-arrivals, up to max_batch
        import time
        t0 = time.perf_counter()
        while len(batch) < self.max_batch:
            remaining = self.window_s - (time.perf_counter() - t0)
            if remaining <= 0:
                break
            try:
                batch.append(self.q.get(timeout=remaining))
            except queue.Empty:
                break
        return batch

    def _loop(self) -> None:
        while not self._stop.is_set():
            batch = self._collect()
            if not batch:
                continue
            batch = [s for s in batch if not s.event.is_set()]  # drop sentinels
            if not batch:
                continue
            try:
                embs = self.wrapper.texts_to_embedding(
                    [s.text for s in batch], bs=self.max_batch)
                for i, s in enumerate(batch):
                    s.result = embs[i: i + 1]  # (1, D) like get_pooled_features
            except BaseException as e:  # propagate per caller
                for s in batch:
                    s.error = e
            self.batches += 1
            self.batched_requests += len(batch)
            for s in batch:
                s.event.set()
