"""Timed readers-writer lock guarding state-dict access during live healing.

Reference parity: torchft/checkpointing/_rwlock.py:46-136. Writer-preferring:
once a writer waits, new readers block, so ``disallow_state_dict_read`` can't
be starved by the checkpoint HTTP server's readers.
"""

from __future__ import annotations

import threading
from contextlib import contextmanager
from typing import Generator


class RWLock:
    def __init__(self, timeout: float = -1) -> None:
        self._timeout = timeout
        self._cond = threading.Condition()
        self._readers = 0
        self._writer = False
        self._writers_waiting = 0

    def r_acquire(self, timeout: float | None = None) -> None:
        timeout = self._timeout if timeout is None else timeout
        with self._cond:
            deadline = None if timeout < 0 else timeout
            ok = self._cond.wait_for(
                lambda: not self._writer and self._writers_waiting == 0,
                timeout=deadline,
            )
            if not ok:
                raise TimeoutError(f"rwlock read acquire timed out after {timeout}s")
            self._readers += 1

    def r_release(self) -> None:
        with self._cond:
            assert self._readers > 0
            self._readers -= 1
            if self._readers == 0:
                self._cond.notify_all()

    def w_acquire(self, timeout: float | None = None) -> None:
        timeout = self._timeout if timeout is None else timeout
        with self._cond:
            self._writers_waiting += 1
            try:
                deadline = None if timeout < 0 else timeout
                ok = self._cond.wait_for(
                    lambda: not self._writer and self._readers == 0,
                    timeout=deadline,
                )
                if not ok:
                    raise TimeoutError(f"rwlock write acquire timed out after {timeout}s")
                self._writer = True
            finally:
                self._writers_waiting -= 1

    def w_release(self) -> None:
        with self._cond:
            assert self._writer
            self._writer = False
            self._cond.notify_all()

    @contextmanager
    def r_lock(self, timeout: float | None = None) -> Generator[None, None, None]:
        self.r_acquire(timeout=timeout)
        try:
            yield
        finally:
            self.r_release()

    @contextmanager
    def w_lock(self, timeout: float | None = None) -> Generator[None, None, None]:
        self.w_acquire(timeout=timeout)
        try:
            yield
        finally:
            self.w_release()
