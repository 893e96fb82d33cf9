"""Thread utilities (parity: reference utils/threads.py).

``RWLock`` (reference threads.py:5-57) and ``ThreadSafeCounter`` with a
``wait_gte`` end-of-batch barrier (reference threads.py:60-91).
"""
import threading
from contextlib import contextmanager


class RWLock:
    """Reader-writer lock: many concurrent readers, exclusive writers."""

    def __init__(self):
        self._cond = threading.Condition()
        self._readers = 0
        self._writer = False

    def acquire_read(self):
        with self._cond:
            while self._writer:
                self._cond.wait()
            self._readers += 1

    def release_read(self):
        with self._cond:
            self._readers -= 1
            if self._readers == 0:
                self._cond.notify_all()

    def acquire_write(self):
        with self._cond:
            while self._writer or self._readers > 0:
                self._cond.wait()
            self._writer = True

    def release_write(self):
        with self._cond:
            self._writer = False
            self._cond.notify_all()

    @contextmanager
    def lock_read(self):
        self.acquire_read()
        try:
            yield self
        finally:
            self.release_read()

    @contextmanager
    def lock_write(self):
        self.acquire_write()
        try:
            yield self
        finally:
            self.release_write()


class ThreadSafeCounter:
    """Monotonic counter with a blocking ``wait_gte`` threshold barrier."""

    def __init__(self, value: int = 0):
        self._value = value
        self._cond = threading.Condition()

    @property
    def value(self) -> int:
        with self._cond:
            return self._value

    def add(self, quantity: int = 1) -> None:
        with self._cond:
            self._value += quantity
            self._cond.notify_all()

    def set(self, value: int) -> None:
        with self._cond:
            self._value = value
            self._cond.notify_all()

    def wait_gte(self, threshold: int, timeout: float = None) -> bool:
        """Block until value >= threshold. Returns False on timeout."""
        with self._cond:
            return self._cond.wait_for(lambda: self._value >= threshold,
                                       timeout=timeout)
