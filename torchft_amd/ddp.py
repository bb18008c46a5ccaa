"""Fault-tolerant DistributedDataParallel.

Reuses torch's DDP reducer for bucketing and backward overlap; only the
cross-replica communication is rerouted, per gradient bucket, through the
Manager's fault-tolerant allreduce (reference semantics: torchft/ddp.py).

Two deliberate deviations from stock DDP:

* construction uses a world-size-1 dummy process group — DDP's built-in
  initial parameter broadcast is meaningless across elastic replica groups;
  step-0 weight agreement comes from the Manager's init-sync healing path;
* bucket rebuilding is disabled (``find_unused_parameters=True``) because a
  replica that joins later would rebuild different buckets than the
  incumbents and the flat allreduce payloads would no longer line up.
"""

# NOTE: no `from __future__ import annotations` — torch validates the comm
# hook's runtime annotations.
from typing import TYPE_CHECKING

import torch
import torch.distributed as dist
from torch import nn
from torch.nn import parallel

from torchft_amd.process_group import ProcessGroupDummy

if TYPE_CHECKING:
    from torchft_amd.manager import Manager


def _completed(value: torch.Tensor) -> torch.futures.Future:
    fut: torch.futures.Future = torch.futures.Future()
    fut.set_result(value)
    return fut


class DistributedDataParallel(parallel.DistributedDataParallel):
    """torch DDP with the reducer's allreduce replaced by the FT path."""

    def __init__(self, manager: "Manager", module: nn.Module, **kwargs: object) -> None:
        super().__init__(
            module,
            process_group=ProcessGroupDummy(0, 1),
            # pin the bucket layout (see module docstring)
            find_unused_parameters=True,
            **kwargs,  # pyre-ignore[6]
        )
        self.register_comm_hook(manager, self._comm_hook)

    @staticmethod
    def _comm_hook(
        state: "Manager", bucket: dist.GradBucket
    ) -> torch.futures.Future[torch.Tensor]:
        from torchft_amd.manager import _ManagedWork

        work = state.allreduce(bucket.buffer())
        ok = work.wait()
        if ok and isinstance(work, _ManagedWork):
            # hand the reducer the realized torch future (the lazy proxy
            # would hang it) — it resolves to the normalized bucket
            return work._realized()
        # launch failed or collective errored: the manager tracked the error
        # and should_commit will reject the step; give the reducer a
        # completed future so backward can finish
        return _completed(bucket.buffer())


class PureDistributedDataParallel(nn.Module):
    """Minimal per-parameter variant: one FT allreduce per gradient as it is
    accumulated. No bucketing, no overlap tuning — useful as a correctness
    oracle against the reducer-based implementation."""

    def __init__(self, manager: "Manager", module: nn.Module) -> None:
        super().__init__()
        self.module = module

        def reduce_grad(p: torch.Tensor) -> None:
            if p.grad is not None:
                # wait() here: the managed work's continuation ledger (the
                # AVG normalization) only replays once the work completes —
                # dropping the handle would leave the gradient un-averaged
                manager.allreduce(p.grad).wait()

        for p in module.parameters():
            p.register_post_accumulate_grad_hook(reduce_grad)

    def forward(self, *args: object) -> object:
        return self.module(*args)
