"""Cross-process mutual exclusion helpers (capability parity with the
reference's utils/lock.py Mutex/Lock; reference: utils/lock.py:32-63).

Unlike the reference — which plumbs a Mutex into learner/storage but never
locks it, leaving the shared-memory cursor as a by-convention race — the new
framework actually uses these around ring-buffer head/tail updates.
"""
from __future__ import annotations

from contextlib import contextmanager

import torch.multiprocessing as mp


class Mutex:
    """Binary semaphore with a contextmanager interface."""

    def __init__(self):
        self._sem = mp.Semaphore(1)

    @contextmanager
    def lock(self):
        self._sem.acquire()
        try:
            yield
        finally:
            self._sem.release()

    def get(self, q):
        with self.lock():
            return q.get()

    def put(self, q, item):
        with self.lock():
            q.put(item)


class Lock:
    """mp.Lock with the same contextmanager interface."""

    def __init__(self):
        self._lock = mp.Lock()

    @contextmanager
    def lock(self):
        self._lock.acquire()
        try:
            yield
        finally:
            self._lock.release()
