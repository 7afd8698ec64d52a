"""Prioritized GPU task pool: one worker thread per device, inference first.

The reference bridges 8 forked handler processes to a single runtime process
via hivemind task pools and MPFutures (server/task_pool.py:127-166); its
priorities are inference=1.0 < forward/backward=2.0 (task_prioritizer.py:15-20).
Here the server is ONE process (asyncio RPC front + this worker thread), so
the pool is a plain thread-safe priority queue — no shared-memory hops, no
cross-process KV descriptor pipes (SURVEY.md §7 step 3 collapse).
"""
from __future__ import annotations

import itertools
import queue
import threading
from concurrent.futures import Future
from typing import Callable, Optional

from bloombee_amd.utils.logging import get_logger

logger = get_logger(__name__)

PRIORITY_INFERENCE = 1.0
PRIORITY_TRAIN = 2.0


class TaskPool:
    def __init__(self, name: str = "gpu-worker"):
        self._q: "queue.PriorityQueue" = queue.PriorityQueue()
        self._seq = itertools.count()
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True, name=name)
        self._thread.start()

    def submit(self, fn: Callable, priority: float = PRIORITY_TRAIN) -> Future:
        fut: Future = Future()
        self._q.put((priority, next(self._seq), fn, fut))
        return fut

    def _run(self):
        while not self._stop.is_set():
            try:
                priority, _, fn, fut = self._q.get(timeout=0.2)
            except queue.Empty:
                continue
            if fut.set_running_or_notify_cancel():
                try:
                    fut.set_result(fn())
                except BaseException as e:  # noqa: BLE001 — delivered to caller
                    fut.set_exception(e)

    def shutdown(self):
        self._stop.set()
        self._thread.join(timeout=5)
