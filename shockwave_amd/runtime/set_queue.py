"""Queue over a set with targeted checkout.

Reference: scheduler/set_queue.py:1-63 — a thread-safe queue whose ``get``
can request a SPECIFIC item (worker id), blocking until it is present.
"""

from __future__ import annotations

import threading


class SetQueue:
    def __init__(self):
        self._set = set()
        self._cv = threading.Condition()

    def put(self, item):
        with self._cv:
            self._set.add(item)
            self._cv.notify_all()

    def get(self, item=None, timeout=None):
        with self._cv:
            if item is None:
                while not self._set:
                    if not self._cv.wait(timeout):
                        raise TimeoutError("SetQueue.get timed out")
                value = next(iter(self._set))
                self._set.remove(value)
                return value
            while item not in self._set:
                if not self._cv.wait(timeout):
                    raise TimeoutError(f"SetQueue.get({item}) timed out")
            self._set.remove(item)
            return item

    def get_nowait(self, item=None):
        with self._cv:
            if item is None:
                if not self._set:
                    raise KeyError("empty")
                value = next(iter(self._set))
                self._set.remove(value)
                return value
            if item not in self._set:
                raise KeyError(item)
            self._set.remove(item)
            return item

    def __contains__(self, item):
        with self._cv:
            return item in self._set

    def __len__(self):
        with self._cv:
            return len(self._set)
