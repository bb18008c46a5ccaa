"""Checkpoint transport interface for live healing.

Reference parity: torchft/checkpointing/transport.py:14-69.
"""

from __future__ import annotations

from abc import ABC, abstractmethod
from datetime import timedelta
from typing import Generic, List, TypeVar

T = TypeVar("T")


class CheckpointTransport(Generic[T], ABC):
    @abstractmethod
    def metadata(self) -> str:
        """Metadata (e.g. an address) recovering replicas need to fetch the
        checkpoint from this process. Returned from every quorum request."""
        ...

    @abstractmethod
    def send_checkpoint(
        self, dst_ranks: List[int], step: int, state_dict: T, timeout: timedelta
    ) -> None:
        """Make ``state_dict`` for ``step`` available to ``dst_ranks``."""
        ...

    def disallow_checkpoint(self) -> None:
        """Called after the send is no longer safe (training will mutate the
        weights); transports that serve asynchronously must gate on this."""
        pass

    @abstractmethod
    def recv_checkpoint(
        self, src_rank: int, metadata: str, step: int, timeout: timedelta
    ) -> T:
        """Fetch the checkpoint for ``step`` from ``src_rank``."""
        ...

    def shutdown(self, wait: bool = True) -> None:
        pass
