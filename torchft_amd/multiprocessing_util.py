"""Pipe helpers for the subprocess-isolated process groups.

Reference parity: torchft/multiprocessing.py (_MonitoredPipe) and
multiprocessing_dummy_context.py (thread-backed mp context for tests).
"""

from __future__ import annotations

import queue
import threading
from multiprocessing.connection import Connection
from typing import Optional


class _MonitoredPipe:
    """Wraps a Connection with timeout-aware recv and exception unpickling."""

    def __init__(self, pipe: Connection) -> None:
        self._pipe = pipe

    def send(self, obj: object) -> None:
        self._pipe.send(obj)

    def recv(self, timeout: Optional[float] = None) -> object:
        if timeout is not None:
            if not self._pipe.poll(timeout):
                raise TimeoutError(f"pipe recv timed out after {timeout}s")
        out = self._pipe.recv()
        if isinstance(out, Exception):
            raise out
        return out

    def poll(self, timeout: float = 0.0) -> bool:
        return self._pipe.poll(timeout)

    def close(self) -> None:
        self._pipe.close()

    def closed(self) -> bool:
        return self._pipe.closed


class _ThreadPipeEnd:
    """One end of a thread-backed 'pipe' (queue pair) for dummy contexts."""

    def __init__(self, rx: "queue.Queue[object]", tx: "queue.Queue[object]") -> None:
        self._rx = rx
        self._tx = tx
        self._closed = False

    def send(self, obj: object) -> None:
        self._tx.put(obj)

    def recv(self) -> object:
        return self._rx.get()

    def poll(self, timeout: Optional[float] = None) -> bool:
        try:
            item = self._rx.get(timeout=timeout)
        except queue.Empty:
            return False
        # put it back — peek semantics
        self._rx.queue.appendleft(item)  # type: ignore[attr-defined]
        return True

    def close(self) -> None:
        self._closed = True

    @property
    def closed(self) -> bool:
        return self._closed


class _ThreadProcess:
    """threading.Thread with a Process-like interface (daemon, terminate)."""

    def __init__(self, target, args=(), daemon: bool = True) -> None:
        self._thread = threading.Thread(target=target, args=args, daemon=daemon)
        self.exitcode: Optional[int] = None

    def start(self) -> None:
        self._thread.start()

    def join(self, timeout: Optional[float] = None) -> None:
        self._thread.join(timeout)
        if not self._thread.is_alive():
            self.exitcode = 0

    def is_alive(self) -> bool:
        return self._thread.is_alive()

    def terminate(self) -> None:  # threads can't be killed; best-effort
        pass

    def kill(self) -> None:
        pass
