"""Pre-completed Work objects (reference parity: torchft/work.py:15-26)."""

from datetime import timedelta
from typing import Optional

import torch
import torch.distributed as dist


class _DummyWork(dist._Work):
    """A Work that is already complete and resolves to ``result``."""

    def __init__(self, result: object) -> None:
        super().__init__()
        self.result_ = result
        self.future_: torch.futures.Future = torch.futures.Future()
        self.future_.set_result(result)

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        return True

    def get_future(self) -> torch.futures.Future:
        return self.future_
