"""Stream/event helpers for the HIP (ROCm) runtime.

Reference parity: torchft/utils.py:17-67. On ROCm ``torch.cuda`` *is* HIP, so
the accelerator paths below drive HIP streams/events on MI355X.
"""

from contextlib import contextmanager, nullcontext
from typing import Generator, Optional

import torch


def get_stream_context(
    stream: Optional[torch.cuda.Stream] = None,
):
    """Return a context manager entering ``stream`` on the accelerator (HIP),
    or a null context on CPU-only hosts."""
    if torch.cuda.is_available() and stream is not None:
        return torch.cuda.stream(stream)
    return nullcontext()


def synchronize() -> None:
    """Synchronize the current accelerator (HIP device on MI355X)."""
    if torch.cuda.is_available():
        torch.cuda.synchronize()


def record_event(interprocess: bool = False) -> Optional[torch.cuda.Event]:
    """Record and return an event on the current HIP stream (None on CPU).

    ``interprocess=True`` creates a hipIpcEvent usable across processes (the
    Baby process-group isolation path).
    """
    if not torch.cuda.is_available():
        return None
    event = torch.cuda.Event(interprocess=interprocess)
    event.record()
    return event
