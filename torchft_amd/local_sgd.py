"""Fault-tolerant LocalSGD and (Streaming) DiLoCo.

Communication-reducing data parallelism across replica groups: inner steps
run locally; every ``sync_every`` steps the groups average parameters
(LocalSGD, https://arxiv.org/pdf/1805.09767) or outer-step on averaged
pseudo-gradients (DiLoCo https://arxiv.org/pdf/2311.08105; Streaming DiLoCo
https://arxiv.org/pdf/2501.18512 staggers per-fragment syncs so the outer
allreduce overlaps ``fragment_sync_delay`` inner steps).

Reference parity (semantics): torchft/local_sgd.py. MI355X notes: the outer
allreduce runs on a dedicated HIP stream and can use the fp8-quantized
path (``should_quantize=True``, CDNA4 kernels + alltoall/allgather across
all 7 xGMI links); backup parameters stage to pinned host memory by default
(288 GB HBM makes an on-device ``backup_device`` practical for ≤70B too);
bucketized allreduce defaults to 1 GiB flat buckets — large enough to hit
per-link peak bandwidth on xGMI rings.
"""

from __future__ import annotations

import logging
import math
import os
from contextlib import nullcontext
from types import TracebackType
from typing import Any, Dict, List, Optional, Tuple, Type

import torch
from torch import nn, optim
from torch.distributed.distributed_c10d import Work
from torch.utils.hooks import RemovableHandle

try:
    from torch.distributed.tensor import DTensor
except ImportError:  # pragma: no cover
    DTensor = None  # type: ignore[assignment]

from torchft_amd.manager import Manager

logger: logging.Logger = logging.getLogger(__name__)

USE_BUCKETIZATION_ENV: str = "TORCHFT_USE_BUCKETIZATION"


def split_into_fragments(model: nn.Module, num_fragments: int) -> List[nn.Module]:
    """Split a model into ``num_fragments`` pipeline-style fragments for
    Streaming DiLoCo (the reference cuts with torch.distributed.pipelining;
    here we cut by top-level children, balancing by parameter count).

    The fragments hold references to the original modules — training the
    model trains the fragments.
    """
    children = [m for m in model.children() if sum(p.numel() for p in m.parameters()) > 0]
    if len(children) < num_fragments:
        raise ValueError(
            f"model has {len(children)} parameterized top-level children, "
            f"cannot split into {num_fragments} fragments"
        )
    total = sum(p.numel() for m in children for p in m.parameters())
    target = total / num_fragments
    fragments: List[nn.Module] = []
    bucket: List[nn.Module] = []
    acc = 0
    for i, m in enumerate(children):
        bucket.append(m)
        acc += sum(p.numel() for p in m.parameters())
        remaining_needed = num_fragments - len(fragments) - 1
        if (acc >= target and remaining_needed > 0 and
                len(children) - i - 1 >= remaining_needed):
            fragments.append(nn.Sequential(*bucket) if len(bucket) > 1 else bucket[0])
            bucket, acc = [], 0
    fragments.append(nn.Sequential(*bucket) if len(bucket) > 1 else bucket[0])
    assert len(fragments) == num_fragments
    return fragments


def extract_local_tensor(t: torch.Tensor) -> torch.Tensor:
    """Cloned local representation of a (D)Tensor, detached from grads."""
    if DTensor is not None and isinstance(t, DTensor):
        new_tensor = t.to_local().clone()
    else:
        new_tensor = t.clone()
    new_tensor.grad = None
    return new_tensor


def _copy_back(p: torch.Tensor, src: torch.Tensor) -> None:
    """Copy a plain local tensor back into a parameter that may be a DTensor."""
    if DTensor is not None and isinstance(p, DTensor):
        p.data.copy_(
            DTensor.from_local(
                src, p.device_mesh, p.placements, shape=p.shape, stride=p.stride()
            )
        )
    else:
        p.data.copy_(src)


class LocalSGD:
    """Context manager syncing (averaging) model weights every ``sync_every``
    optimizer steps through the fault-tolerant Manager.

    Errors or membership changes between syncs discard the ``sync_every``
    step window; a new quorum forms on the next sync.
    """

    def __init__(
        self,
        manager: Manager,
        model: nn.Module,
        optimizer: optim.Optimizer,
        sync_every: int,
    ) -> None:
        super().__init__()
        self._manager = manager
        self._model = model
        self._local_optimizer = optimizer
        self._local_step = 0
        self._sync_every = sync_every
        assert sync_every >= 1, "sync_every must be greater than or equal to 1"
        self._hooks: List[RemovableHandle] = []

    def __enter__(self) -> "LocalSGD":
        self._hooks.append(
            self._local_optimizer.register_step_pre_hook(self._step_pre_hook)
        )
        self._hooks.append(
            self._local_optimizer.register_step_post_hook(self._step_post_hook)
        )
        return self

    def __exit__(
        self,
        exc_type: Optional[Type[BaseException]],
        exc_value: Optional[BaseException],
        traceback: Optional[TracebackType],
    ) -> bool:
        for hook in self._hooks:
            hook.remove()
        self._hooks.clear()
        return False

    def _step_pre_hook(
        self, _optim: optim.Optimizer, _args: Tuple[Any, ...], _kwargs: Dict[str, Any]
    ) -> None:
        # checkpoint serving may read the state dict concurrently
        self._manager.disallow_state_dict_read()

    def _step_post_hook(
        self, _optim: optim.Optimizer, _args: Tuple[Any, ...], _kwargs: Dict[str, Any]
    ) -> None:
        self._manager.allow_state_dict_read()
        self._local_step += 1
        if self._local_step >= self._sync_every:
            self.sync()

    def sync(self) -> None:
        """Averages model weights across the quorum."""
        self._manager.start_quorum()
        self._perform_sync()
        self._local_step = 0

    def _perform_sync(self) -> None:
        averaged_parameters = self._average()
        if self._manager.should_commit():
            for param, avg_param in zip(self._model.parameters(), averaged_parameters):
                _copy_back(param, avg_param)

    def _average(self) -> List[torch.Tensor]:
        works = []
        averaged_parameters = []
        for p in self._model.parameters():
            avg_param = extract_local_tensor(p)
            works.append(self._manager.allreduce(avg_param))
            averaged_parameters.append(avg_param)
        for work in works:
            work.wait()
        return averaged_parameters


class _StreamingDiLoCoFragment:
    """One model fragment in the staggered DiLoCo sync schedule.

    ``prepare_sync`` (at ``sync_every - fragment_sync_delay``) computes the
    pseudo-gradient (original − local) and launches the allreduce on a side
    HIP stream; ``perform_sync`` (at ``sync_every``) waits on the recorded
    event, restores the global parameters, runs the should_commit barrier,
    and on success outer-steps + merges by ``fragment_update_alpha``.
    """

    bucket_cap_mb: int = 1 * 1024 * 1024 * 1024
    use_bucketization: bool = False

    def __init__(
        self,
        manager: Manager,
        model_fragment: nn.Module,
        fragment_id: int,
        fragment_sync_offset: int,
        inner_optimizer: optim.Optimizer,
        outer_optimizer: optim.Optimizer,
        sync_every: int,
        backup_device: Optional[torch.device] = None,
        pin_memory: bool = True,
        use_bucketization: bool = False,
        bucket_cap_mb: Optional[int] = None,
        should_quantize: bool = False,
        fragment_sync_delay: int = 0,
        fragment_update_alpha: float = 0.0,
    ) -> None:
        if fragment_sync_offset > sync_every:
            raise ValueError("Fragment must be synced once before `sync_every` steps")

        self._fragment_id = fragment_id
        self._manager = manager
        self._model_fragment = model_fragment
        self._fragment_sync_offset = fragment_sync_offset
        self._local_optimizer = inner_optimizer
        self._sync_every = sync_every
        assert sync_every >= 1
        self._backup_device = backup_device
        self._pin_memory = pin_memory
        self._fragment_sync_delay = fragment_sync_delay
        self._fragment_update_alpha = fragment_update_alpha
        self._outer_optimizer = outer_optimizer

        self._allreduce_work: List[Work] = []
        # the outer allreduce runs on its own HIP stream so it overlaps the
        # following fragment_sync_delay inner steps
        self._stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if torch.cuda.is_available() else None
        )
        self._stop_event: Optional[torch.cuda.Event] = None

        if bucket_cap_mb is not None:
            self.bucket_cap_mb = int(bucket_cap_mb * 1024 * 1024)
        if os.getenv(USE_BUCKETIZATION_ENV, "False") == "True":
            self.use_bucketization = True
        else:
            self.use_bucketization = use_bucketization
        self.should_quantize = should_quantize

        self._grads: Dict[str, torch.Tensor] = {}
        # restore point in case the commit fails
        self.original_parameters: Dict[str, torch.Tensor] = {}
        # local params saved around restore so they can be merged back
        self._local_parameters: Dict[str, torch.Tensor] = {}

        for name, p in self._model_fragment.named_parameters():
            if DTensor is not None and isinstance(p, DTensor):
                p = extract_local_tensor(p.data)
            backup_device = self._backup_device or torch.device("cpu")
            t = torch.empty(*tuple(p.shape), dtype=p.dtype, device=backup_device)
            if (
                self._pin_memory
                and t.device == torch.device("cpu")
                and torch.cuda.is_available()
            ):
                t = t.pin_memory()
            self.original_parameters[name] = t

    def register_state_dict_fn(self) -> None:
        """Registers this fragment's backup params + outer optimizer state
        with the manager so recovering replicas heal them too."""
        fragment_key = f"StreamingDiLoCoFragment_{self._fragment_id}"

        def load_fn(state_dict: Dict[str, Dict[str, torch.Tensor]]) -> None:
            for name, param in state_dict["original_parameters"].items():
                if name in self.original_parameters:
                    self.original_parameters[name].copy_(param)
            self._outer_optimizer.load_state_dict(state_dict["outer_optimizer"])

        def save_fn() -> Dict[str, Dict[str, torch.Tensor]]:
            return {
                "outer_optimizer": self._outer_optimizer.state_dict(),
                "original_parameters": {
                    name: extract_local_tensor(param)
                    for name, param in self.original_parameters.items()
                },
            }

        self._manager.register_state_dict_fn(fragment_key, load_fn, save_fn)

    @torch.profiler.record_function("torchft_amd::local_sgd::save_parameters")
    def save_parameters(self) -> None:
        with torch.no_grad():
            for name, p in self._model_fragment.named_parameters():
                param_to_local = extract_local_tensor(p.data)
                self.original_parameters[name].copy_(param_to_local, non_blocking=True)

    def _save_local_parameters(self) -> None:
        with torch.no_grad():
            for name, p in self._model_fragment.named_parameters():
                self._local_parameters[name] = extract_local_tensor(p.data)

    @torch.profiler.record_function("torchft_amd::local_sgd::restore_parameters")
    def restore_parameters(self) -> None:
        with torch.no_grad():
            for name, p in self._model_fragment.named_parameters():
                if DTensor is not None and isinstance(p, DTensor):
                    p.data.copy_(
                        DTensor.from_local(
                            self.original_parameters[name],
                            p.device_mesh,
                            p.placements,
                            shape=p.shape,
                            stride=p.stride(),
                        ),
                        non_blocking=False,
                    )
                else:
                    p.data.copy_(self.original_parameters[name], non_blocking=False)

    def _save_grads(self) -> None:
        """pseudo-gradient = original (pre-window) params − local params"""
        with torch.no_grad():
            for name, p in self._model_fragment.named_parameters():
                if DTensor is not None and isinstance(p, DTensor):
                    local_param = p.to_local()
                else:
                    local_param = p
                pseudogradient = self.original_parameters[name].to(p.device) - local_param
                self._grads[name] = pseudogradient

    def _set_grads(self) -> None:
        with torch.no_grad():
            for name, p in self._model_fragment.named_parameters():
                if DTensor is not None and isinstance(p, DTensor):
                    p.grad = DTensor.from_local(
                        self._grads[name],
                        p.device_mesh,
                        p.placements,
                        shape=p.shape,
                        stride=p.stride(),
                    )
                else:
                    p.grad = self._grads[name]
                del self._grads[name]

    def _clear_local_parameters(self) -> None:
        self._local_parameters = {}

    def _merge_parameters(self) -> None:
        """p = (1-alpha) * global + alpha * local"""
        for name, p in self._model_fragment.named_parameters():
            if DTensor is not None and isinstance(p, DTensor):
                p.data.lerp_(
                    DTensor.from_local(
                        self._local_parameters[name],
                        p.device_mesh,
                        p.placements,
                        shape=p.shape,
                        stride=p.stride(),
                    ),
                    self._fragment_update_alpha,
                )
            else:
                p.data.lerp_(self._local_parameters[name], self._fragment_update_alpha)

    def wait(self) -> None:
        """Wait for the previously launched allreduce."""
        if len(self._allreduce_work) == 0:
            return
        if self._stream is not None:
            assert self._stop_event is not None
            self._stop_event.synchronize()
            self._stop_event = None
        self._allreduce_work = []

    @torch.profiler.record_function("torchft_amd::local_sgd::prepare_sync")
    def prepare_sync(self) -> None:
        """Compute pseudo-gradients and launch (but don't wait for) the
        allreduce on the side stream."""
        self._save_grads()
        assert len(self._allreduce_work) == 0
        if self._stream is not None:
            self._stream.wait_stream(torch.cuda.current_stream())
        with (
            torch.cuda.stream(self._stream) if self._stream is not None else nullcontext()
        ):
            self._average_grads()

    @torch.profiler.record_function("torchft_amd::local_sgd::perform_sync")
    def perform_sync(self) -> bool:
        """Wait for the allreduce, then commit (outer-step) or roll back."""
        assert len(self._allreduce_work) > 0

        with (
            torch.cuda.stream(self._stream) if self._stream is not None else nullcontext()
        ):
            for work in self._allreduce_work:
                work.wait()
            if self._stream is not None:
                self._stop_event = torch.cuda.Event()
                self._stop_event.record()

        self.wait()

        # save local params for merging, then roll back to the global state
        self._save_local_parameters()
        self.restore_parameters()

        # NOTE: with large fragment_sync_delay this can report success even
        # if the allreduce was aborted by a reconfiguration mid-flight (the
        # reference documents the same compromise to avoid extra aborts).
        should_commit = self._manager.should_commit()

        if should_commit:
            self._set_grads()
            self._outer_optimizer.step()
            self.save_parameters()
            self._merge_parameters()
        self._outer_optimizer.zero_grad()
        self._clear_local_parameters()
        return should_commit

    def _average_grads(self) -> None:
        if self.use_bucketization:
            self._allreduce_bucketized()
        else:
            self._allreduce_per_param()

    def _allreduce_per_param(self) -> None:
        for name, p in self._model_fragment.named_parameters():
            work = self._manager.allreduce(
                self._grads[name], should_quantize=self.should_quantize
            )
            self._allreduce_work.append(work)

    def _bucketize_and_allreduce(
        self, tensors: List[torch.Tensor], bucket_size_bytes: int
    ) -> None:
        """Pack tensors into flat buckets, allreduce each, scatter back via a
        future continuation."""
        if not tensors:
            return
        total_size = sum(t.numel() for t in tensors)
        dtype, device = tensors[0].dtype, tensors[0].device

        offset = 0
        flat_index = 0
        while offset < total_size:
            chunk_size = min(
                bucket_size_bytes // tensors[0].element_size(), total_size - offset
            )
            flat_buffer = torch.zeros(chunk_size, dtype=dtype, device=device)

            pack_offset = 0
            bucket_tensors: List[Tuple[torch.Tensor, int, int]] = []
            for t in tensors[flat_index:]:
                numel = t.numel()
                if pack_offset + numel > chunk_size:
                    break
                flat_buffer[pack_offset : pack_offset + numel].copy_(t.view(-1))
                bucket_tensors.append((t, pack_offset, numel))
                pack_offset += numel
                flat_index += 1

            work = self._manager.allreduce(
                flat_buffer, should_quantize=self.should_quantize
            )

            def callback(
                fut: torch.futures.Future[List[torch.Tensor]],
            ) -> List[torch.Tensor]:
                nonlocal bucket_tensors, flat_buffer
                for t, pack_offset, numel in bucket_tensors:
                    t.copy_(flat_buffer[pack_offset : pack_offset + numel].view_as(t))
                return []

            fut = work.get_future()
            fut = fut.then(callback)
            self._allreduce_work.append(work)
            offset += chunk_size

    def _allreduce_bucketized(self) -> None:
        grads = list(self._grads.values())
        assert len(grads) > 0, "No gradients to allreduce"
        self._bucketize_and_allreduce(grads, bucket_size_bytes=self.bucket_cap_mb)


class DiLoCo:
    """(Streaming) DiLoCo: inner steps local; every ``sync_every`` steps a
    fragment's pseudo-gradients average across replicas and an outer
    optimizer applies them, with per-fragment staggering so communication
    overlaps ``fragment_sync_delay`` inner steps.

    Requires ``use_async_quorum=False`` on the Manager (the sync schedule
    must be identical on every replica — see the deadlock note in
    ``_step_post_hook``).
    """

    def __init__(
        self,
        manager: Manager,
        model_fragments: List[nn.Module],
        inner_optimizer: optim.Optimizer,
        outer_optimizer: optim.Optimizer | List[optim.Optimizer],
        sync_every: int,
        backup_device: Optional[torch.device] = None,
        pin_memory: bool = True,
        use_bucketization: bool = False,
        bucket_cap_mb: Optional[int] = None,
        should_quantize: bool = False,
        fragment_sync_delay: int = 0,
        fragment_update_alpha: float = 0.0,
    ) -> None:
        if isinstance(outer_optimizer, list):
            assert len(outer_optimizer) == len(model_fragments), (
                "The number of outer optimizers must match the number of model fragments"
            )
        if manager._use_async_quorum:
            raise ValueError(
                "DiLoCo requires synchronous quorum; construct the Manager "
                "with use_async_quorum=False"
            )
        if sync_every < len(model_fragments):
            raise ValueError("Only 1 fragment can be synchronized at a time")
        if sync_every % len(model_fragments) != 0:
            raise ValueError("sync_every must divide the number of fragments")

        self._sync_every: int = sync_every // len(model_fragments)
        if fragment_sync_delay >= self._sync_every:
            raise ValueError("Fragment must be synced before it is reduced another time")
        if fragment_update_alpha < 0 or fragment_update_alpha > 1:
            raise ValueError("fragment_update_alpha must be between 0 and 1")

        super().__init__()
        self._manager = manager
        self._local_step = 0
        self._fragment_sync_delay = fragment_sync_delay
        self._hooks: List[RemovableHandle] = []
        self._local_optimizer = inner_optimizer

        self._fragments: List[_StreamingDiLoCoFragment] = [
            _StreamingDiLoCoFragment(
                manager,
                model_fragment,
                i,
                math.floor((sync_every / len(model_fragments)) * (i + 1)),
                inner_optimizer,
                (
                    outer_optimizer[i]
                    if isinstance(outer_optimizer, list)
                    else outer_optimizer
                ),
                sync_every,
                backup_device,
                pin_memory,
                use_bucketization,
                bucket_cap_mb,
                should_quantize,
                fragment_sync_delay,
                fragment_update_alpha,
            )
            for i, model_fragment in enumerate(model_fragments)
        ]

        assert fragment_sync_delay < sync_every // len(model_fragments)

        # copy params to the backup device before the first step
        self._save_parameters()
        self._register_state_dict_fn()

    def _register_state_dict_fn(self) -> None:
        for fragment in self._fragments:
            fragment.register_state_dict_fn()

    def _save_parameters(self) -> None:
        for fragment in self._fragments:
            fragment.save_parameters()

    def _restore_parameters(self) -> None:
        for fragment in self._fragments:
            fragment.restore_parameters()

    def __enter__(self) -> "DiLoCo":
        self._hooks.append(
            self._local_optimizer.register_step_pre_hook(self._step_pre_hook)
        )
        self._hooks.append(
            self._local_optimizer.register_step_post_hook(self._step_post_hook)
        )
        return self

    def _step_pre_hook(
        self, _optim: optim.Optimizer, _args: Tuple[Any, ...], _kwargs: Dict[str, Any]
    ) -> None:
        self._manager.disallow_state_dict_read()

    def __exit__(
        self,
        exc_type: Optional[Type[BaseException]],
        exc_value: Optional[BaseException],
        traceback: Optional[TracebackType],
    ) -> bool:
        for hook in self._hooks:
            hook.remove()
        self._hooks.clear()
        return False

    def _wait(self) -> None:
        for fragment in self._fragments:
            fragment.wait()

    def _current_fragment(self) -> int:
        """All replicas must pick the same fragment: derive it from the
        committed manager step, not local state."""
        step = self._manager.current_step()
        return step % len(self._fragments)

    def _step_post_hook(
        self, _optim: optim.Optimizer, _args: Tuple[Any, ...], _kwargs: Dict[str, Any]
    ) -> None:
        self._manager.allow_state_dict_read()

        # All nodes must send the same fragments in the same order, else:
        #   step 1: node A sends fragment 1, node B sends fragment 2
        #   step 2: node A waits for fragment 1, node B waits for fragment 2
        # -> deadlock. The schedule below is a pure function of the
        # committed step, so it is identical everywhere.
        self._local_step += 1

        if self._local_step == self._sync_every - self._fragment_sync_delay:
            # launch this window's fragment allreduce
            self._manager.start_quorum()
            fragment = self._current_fragment()
            logger.info(f"Preparing fragment={fragment} step={self._local_step}")
            self._fragments[fragment].prepare_sync()

        if self._local_step < self._sync_every:
            return

        if self._local_step == self._sync_every:
            fragment = self._current_fragment()
            logger.info(
                f"Syncing fragment={fragment} step={self._local_step} "
                f"manager_step={self._manager.current_step()}"
            )
            self._fragments[fragment].perform_sync()
            # on failure the fragment rolled back; we retry the window
            self._local_step = 0
            return

        raise AssertionError(
            f"{self._local_step=} should never be greater than {self._sync_every=}"
        )
