"""Writer-priority reader-writer lock.

Parity with reference sparktorch/rw_lock.py:11-67: readers share, writers are
exclusive and take priority over newly arriving readers.  Used by the
parameter server when ``acquireLock=True``.
"""

from __future__ import annotations

import threading


class RWLock(object):
    """One mutex + two condition variables; writer-priority."""

    def __init__(self):
        self._lock = threading.Lock()
        self._readers_ok = threading.Condition(self._lock)
        self._writers_ok = threading.Condition(self._lock)
        self._active_readers = 0
        self._waiting_writers = 0
        self._writer_active = False

    def acquire_read(self) -> None:
        with self._lock:
            while self._writer_active or self._waiting_writers > 0:
                self._readers_ok.wait()
            self._active_readers += 1

    def acquire_write(self) -> None:
        with self._lock:
            self._waiting_writers += 1
            try:
                while self._writer_active or self._active_readers > 0:
                    self._writers_ok.wait()
            finally:
                self._waiting_writers -= 1
            self._writer_active = True

    def release(self) -> None:
        with self._lock:
            if self._writer_active:
                self._writer_active = False
            elif self._active_readers > 0:
                self._active_readers -= 1
            else:
                raise RuntimeError("release() called on an unheld RWLock")
            # Writer priority: wake writers first, readers only if no writer waits.
            if self._waiting_writers > 0:
                if not self._writer_active and self._active_readers == 0:
                    self._writers_ok.notify()
            else:
                self._readers_ok.notify_all()
