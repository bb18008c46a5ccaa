"""User-space timeouts for futures, HIP streams and contexts.

Reference parity: torchft/futures.py (asyncio event-loop based
``_TimeoutManager`` with ``future_timeout`` / ``stream_timeout`` /
``context_timeout`` and a watchdog). Re-designed here as a single daemon
timer thread with a deadline heap — no asyncio — plus a small callback
executor so a slow abort callback (e.g. RCCL comm abort) can never wedge the
timer loop. A watchdog aborts the process if the timer thread itself stops
making progress (reference: TORCHFT_WATCHDOG_TIMEOUT_SEC).
"""

from __future__ import annotations

import heapq
import itertools
import os
import sys
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from contextlib import contextmanager
from datetime import timedelta
from typing import Callable, Generator, Optional, TypeVar

import torch

T = TypeVar("T")

WATCHDOG_TIMEOUT_SEC_ENV = "TORCHFT_WATCHDOG_TIMEOUT_SEC"


class _Timer:
    __slots__ = ("deadline", "seq", "fn", "cancelled")

    def __init__(self, deadline: float, seq: int, fn: Callable[[], None]) -> None:
        self.deadline = deadline
        self.seq = seq
        self.fn = fn
        self.cancelled = False

    def __lt__(self, other: "_Timer") -> bool:
        return (self.deadline, self.seq) < (other.deadline, other.seq)


class _TimerManager:
    """Singleton deadline-heap timer thread."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._cond = threading.Condition(self._lock)
        self._heap: list[_Timer] = []
        self._seq = itertools.count()
        self._thread: Optional[threading.Thread] = None
        self._watchdog_thread: Optional[threading.Thread] = None
        # Callbacks run off-thread so a wedged abort can't stall other timers.
        self._executor: Optional[ThreadPoolExecutor] = None
        self._last_tick = time.monotonic()

    def _ensure_started(self) -> None:
        if self._thread is not None:
            return
        self._executor = ThreadPoolExecutor(
            max_workers=4, thread_name_prefix="torchft_amd_timer_cb"
        )
        self._thread = threading.Thread(
            target=self._loop, daemon=True, name="torchft_amd_timer"
        )
        self._thread.start()
        self._watchdog_thread = threading.Thread(
            target=self._watchdog, daemon=True, name="torchft_amd_watchdog"
        )
        self._watchdog_thread.start()

    def _loop(self) -> None:
        while True:
            with self._cond:
                self._last_tick = time.monotonic()
                while not self._heap or self._heap[0].deadline > time.monotonic():
                    if self._heap:
                        wait = min(self._heap[0].deadline - time.monotonic(), 1.0)
                    else:
                        wait = 1.0
                    if wait > 0:
                        self._cond.wait(wait)
                    self._last_tick = time.monotonic()
                timer = heapq.heappop(self._heap)
            if not timer.cancelled:
                executor = self._executor
                assert executor is not None
                executor.submit(self._run_cb, timer.fn)

    @staticmethod
    def _run_cb(fn: Callable[[], None]) -> None:
        try:
            fn()
        except Exception:  # noqa: BLE001 - timer callbacks must never propagate
            import logging

            logging.getLogger(__name__).exception("timer callback failed")

    def _watchdog(self) -> None:
        # If the timer loop stops ticking (e.g. wedged by a native hang), the
        # whole FT protocol loses its error path — die loudly instead.
        timeout = float(os.environ.get(WATCHDOG_TIMEOUT_SEC_ENV, "30"))
        while True:
            time.sleep(timeout / 2)
            with self._lock:
                last = self._last_tick
            if time.monotonic() - last > timeout:
                print(
                    f"torchft_amd watchdog: timer loop wedged for >{timeout}s, exiting",
                    file=sys.stderr,
                    flush=True,
                )
                os._exit(1)

    def schedule(self, delay: timedelta, fn: Callable[[], None]) -> _Timer:
        with self._cond:
            self._ensure_started()
            timer = _Timer(
                time.monotonic() + delay.total_seconds(), next(self._seq), fn
            )
            heapq.heappush(self._heap, timer)
            self._cond.notify()
            return timer

    def cancel(self, timer: _Timer) -> None:
        with self._lock:
            timer.cancelled = True


_TIMER_MANAGER = _TimerManager()


def future_timeout(
    fut: torch.futures.Future[T], timeout: timedelta
) -> torch.futures.Future[T]:
    """Return a future that errors with TimeoutError if ``fut`` has not
    completed within ``timeout`` (reference: torchft/futures.py future_timeout)."""
    timed: torch.futures.Future[T] = torch.futures.Future()
    lock = threading.Lock()
    done = [False]

    def on_timeout() -> None:
        with lock:
            if done[0]:
                return
            done[0] = True
        timed.set_exception(TimeoutError(f"future timed out after {timeout}"))

    timer = _TIMER_MANAGER.schedule(timeout, on_timeout)

    def on_done(f: torch.futures.Future[T]) -> None:
        _TIMER_MANAGER.cancel(timer)
        with lock:
            if done[0]:
                return
            done[0] = True
        try:
            timed.set_result(f.value())
        except Exception as e:  # noqa: BLE001
            timed.set_exception(e)

    fut.add_done_callback(on_done)
    return timed


def stream_timeout(callback: Callable[[], None], timeout: timedelta) -> None:
    """Enqueue an event on the current HIP stream; if it has not completed by
    ``timeout``, invoke ``callback`` (typically ``pg.abort``).

    Reference: torchft/futures.py stream_timeout (CUDA event poll → abort).
    """
    if not torch.cuda.is_available():
        return
    event = torch.cuda.Event()
    event.record()

    def check() -> None:
        if not event.query():
            callback()

    _TIMER_MANAGER.schedule(timeout, check)


@contextmanager
def context_timeout(
    callback: Callable[[], None], timeout: timedelta
) -> Generator[None, None, None]:
    """Run ``callback`` if the with-block does not exit within ``timeout``."""
    timer = _TIMER_MANAGER.schedule(timeout, callback)
    try:
        yield
    finally:
        _TIMER_MANAGER.cancel(timer)
