"""Fault-tolerant DistributedDataParallel (reference parity: torchft/ddp.py).

The torch DDP reducer handles bucketing/overlap; cross-replica communication
goes through the Manager's fault-tolerant allreduce via a comm hook. A dummy
world-size-1 PG soaks up DDP's init broadcast (the Manager handles step-0
weight sync through the healing protocol instead).
"""

# NOTE: no `from __future__ import annotations` here — torch's
# register_comm_hook validates the hook's *runtime* annotations.
from typing import TYPE_CHECKING, cast

import torch
import torch.distributed as dist
from torch import nn
from torch.nn import parallel

from torchft_amd.process_group import ProcessGroupDummy

if TYPE_CHECKING:
    from torchft_amd.manager import Manager, _ManagedFuture


class DistributedDataParallel(parallel.DistributedDataParallel):
    """torch DDP patched for fault tolerance.

    Notes:
    * step-0 state sync happens through the Manager (init_sync), not DDP's
      internal broadcast.
    * ``find_unused_parameters=True`` pins the bucket layout: rebuilt buckets
      would diverge between recovering and healthy replicas.
    """

    def __init__(self, manager: "Manager", module: nn.Module, **kwargs: object) -> None:
        # dummy PG soaks up the init allreduce; real comms go via the hook
        pg = ProcessGroupDummy(0, 1)
        super().__init__(
            module,
            process_group=pg,
            # Forces the reducer to never rebuild buckets — rebuilt buckets
            # would diverge for recovering replicas.
            find_unused_parameters=True,
            **kwargs,  # pyre-ignore[6]
        )
        self.register_comm_hook(manager, self._comm_hook)

    @staticmethod
    def _comm_hook(
        state: "Manager", bucket: dist.GradBucket
    ) -> torch.futures.Future[torch.Tensor]:
        work = state.allreduce(bucket.buffer())
        ok = work.wait()
        fut = work.get_future()
        # return the materialized inner future — returning the lazy wrapper
        # hangs the reducer
        fut = cast("_ManagedFuture[torch.Tensor]", fut)
        if not ok or fut._fut is None:
            # collective errored: the error is tracked by the manager and the
            # step will be rejected at should_commit; hand the reducer a
            # completed future so backward finishes.
            done: torch.futures.Future[torch.Tensor] = torch.futures.Future()
            done.set_result(bucket.buffer())
            return done
        return fut._fut


class PureDistributedDataParallel(nn.Module):
    """Per-parameter post-accumulate-grad-hook DDP variant (simple, slow)."""

    def __init__(self, manager: "Manager", module: nn.Module) -> None:
        super().__init__()
        self.module = module

        def post_grad_hook(p: torch.Tensor) -> None:
            if p.grad is not None:
                manager.allreduce(p.grad)

        for p in module.parameters():
            p.register_post_accumulate_grad_hook(post_grad_hook)

    def forward(self, *args: object) -> object:
        return self.module(*args)
