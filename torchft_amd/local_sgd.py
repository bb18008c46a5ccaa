"""Fault-tolerant LocalSGD and (Streaming) DiLoCo.

Communication-reducing data parallelism across replica groups: inner
optimizer steps stay local; every ``sync_every`` steps the replicas either
average parameters directly (LocalSGD, https://arxiv.org/pdf/1805.09767) or
outer-step on averaged pseudo-gradients (DiLoCo,
https://arxiv.org/pdf/2311.08105). Streaming DiLoCo
(https://arxiv.org/pdf/2501.18512) staggers the fragments so each outer
allreduce overlaps ``fragment_sync_delay`` inner steps.

Semantics are matched against the reference (torchft/local_sgd.py) — the
regression fixtures in tests/test_diloco_regression.py pin the numeric
trajectory — but the implementation is organized differently:

* a fragment's sync is an explicit three-phase object: ``stage()``
  (pseudo-gradients + allreduce launch on a side HIP stream), ``collect()``
  (drain the side stream), ``commit()`` (restore/outer-step/merge);
* the staggered schedule is computed by ``_SyncSchedule``, a pure function
  of the committed step — identical on every replica by construction (the
  cross-replica deadlock the reference warns about cannot be expressed);
* flat-bucket packing lives in ``_BucketPlan``, which binds scatter-back
  ranges per bucket (the reference's closure-cell bug — every callback
  seeing the last bucket — is structurally impossible) and refuses tensors
  larger than the bucket cap instead of silently skipping them.

MI355X notes: the outer allreduce can take the fp8-quantized path
(``should_quantize=True`` -> CDNA4 kernels + alltoall/allgather across all
7 xGMI links); backups stage to pinned host memory by default, and the
1 GiB default bucket is sized to saturate a per-link xGMI ring.
"""

from __future__ import annotations

import logging
import math
import os
from contextlib import nullcontext
from dataclasses import dataclass
from types import TracebackType
from typing import Any, Dict, Iterator, List, Optional, Tuple, Type

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


# ---------------------------------------------------------------------------
# (D)Tensor helpers
# ---------------------------------------------------------------------------


def extract_local_tensor(t: torch.Tensor) -> torch.Tensor:
    """Detached clone of the local shard of a (D)Tensor."""
    local = t.to_local() if (DTensor is not None and isinstance(t, DTensor)) else t
    out = local.clone()
    out.grad = None
    return out


def _like_param(p: torch.Tensor, local: torch.Tensor) -> torch.Tensor:
    """Lift a plain local tensor back to the parameter's type (DTensor-aware)."""
    if DTensor is not None and isinstance(p, DTensor):
        return DTensor.from_local(
            local, p.device_mesh, p.placements, shape=p.shape, stride=p.stride()
        )
    return local


def _copy_back(p: torch.Tensor, src: torch.Tensor) -> None:
    p.data.copy_(_like_param(p, src))


# ---------------------------------------------------------------------------
# fragment splitting
# ---------------------------------------------------------------------------


def _atomic_units(model: nn.Module) -> List[nn.Module]:
    """Parameterized split units: top-level children, with bare containers
    (ModuleList/Sequential/ModuleDict) flattened one level so a transformer
    whose blocks live in a single ModuleList still splits per block."""
    units: List[nn.Module] = []
    for child in model.children():
        if isinstance(child, (nn.ModuleList, nn.Sequential, nn.ModuleDict)):
            units.extend(c for c in child.children())
        else:
            units.append(child)
    return [u for u in units if sum(p.numel() for p in u.parameters()) > 0]


def split_into_fragments(model: nn.Module, num_fragments: int) -> List[nn.Module]:
    """Split ``model`` into ``num_fragments`` contiguous fragments balanced
    by parameter count, for Streaming DiLoCo.

    Fragments alias the original modules — training the model trains the
    fragments. Container children are flattened one level, so e.g. a Llama
    whose decoder blocks sit in one ``ModuleList`` splits at block
    granularity rather than counting the list as a single unit.
    (Reference uses torch.distributed.pipelining to cut:
    /root/reference/train_diloco.py:159-162.)
    """
    units = _atomic_units(model)
    if len(units) < num_fragments:
        raise ValueError(
            f"model has {len(units)} parameterized split units, "
            f"cannot split into {num_fragments} fragments"
        )
    total = sum(p.numel() for u in units for p in u.parameters())
    target = total / num_fragments

    fragments: List[nn.Module] = []
    bucket: List[nn.Module] = []
    acc = 0
    for i, unit in enumerate(units):
        bucket.append(unit)
        acc += sum(p.numel() for p in unit.parameters())
        still_needed = num_fragments - len(fragments) - 1
        units_left = len(units) - i - 1
        # close the bucket when it reached its share, or when every
        # remaining unit is needed to fill the remaining fragments
        must_close = still_needed > 0 and units_left == still_needed
        if must_close or (acc >= target and still_needed > 0 and units_left >= still_needed):
            fragments.append(nn.Sequential(*bucket) if len(bucket) > 1 else bucket[0])
            bucket, acc = [], 0
    fragments.append(nn.Sequential(*bucket) if len(bucket) > 1 else bucket[0])
    assert len(fragments) == num_fragments
    return fragments


# ---------------------------------------------------------------------------
# LocalSGD
# ---------------------------------------------------------------------------


class LocalSGD:
    """Context manager averaging model weights across the quorum every
    ``sync_every`` optimizer steps.

    Between syncs the replicas drift freely; an error or membership change
    discards the window and a fresh quorum forms at the next sync.
    """

    def __init__(
        self,
        manager: Manager,
        model: nn.Module,
        optimizer: optim.Optimizer,
        sync_every: int,
    ) -> None:
        assert sync_every >= 1, "sync_every must be greater than or equal to 1"
        self._manager = manager
        self._model = model
        self._local_optimizer = optimizer
        self._sync_every = sync_every
        self._local_step = 0
        self._hooks: List[RemovableHandle] = []

    def __enter__(self) -> "LocalSGD":
        opt = self._local_optimizer
        self._hooks += [
            opt.register_step_pre_hook(self._pre_step),
            opt.register_step_post_hook(self._post_step),
        ]
        return self

    def __exit__(
        self,
        exc_type: Optional[Type[BaseException]],
        exc_value: Optional[BaseException],
        traceback: Optional[TracebackType],
    ) -> bool:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
        return False

    def _pre_step(self, *_: object) -> None:
        # a checkpoint must not be served while the optimizer mutates state
        self._manager.disallow_state_dict_read()

    def _post_step(self, *_: object) -> None:
        self._manager.allow_state_dict_read()
        self._local_step += 1
        if self._local_step >= self._sync_every:
            self.sync()

    def sync(self) -> None:
        """Average the model weights across the quorum."""
        self._manager.start_quorum()
        averaged = self._allreduce_params()
        if self._manager.should_commit():
            for p, avg in zip(self._model.parameters(), averaged):
                _copy_back(p, avg)
        self._local_step = 0

    def _allreduce_params(self) -> List[torch.Tensor]:
        snapshots = [extract_local_tensor(p) for p in self._model.parameters()]
        pending = [self._manager.allreduce(t) for t in snapshots]
        for w in pending:
            w.wait()
        return snapshots


# ---------------------------------------------------------------------------
# bucketized allreduce
# ---------------------------------------------------------------------------


@dataclass
class _BucketSlot:
    tensor: torch.Tensor
    offset: int
    numel: int


class _BucketPlan:
    """Greedy contiguous packing of tensors into flat buckets of at most
    ``cap_bytes``. Each bucket owns its slot table, so scatter-back is bound
    per bucket by construction."""

    def __init__(self, tensors: List[torch.Tensor], cap_bytes: int) -> None:
        assert tensors, "no tensors to pack"
        self.dtype = tensors[0].dtype
        self.device = tensors[0].device
        cap_elems = max(1, cap_bytes // tensors[0].element_size())
        for t in tensors:
            if t.numel() > cap_elems:
                raise ValueError(
                    f"tensor with {t.numel()} elements exceeds the "
                    f"{cap_bytes}-byte bucket cap; raise bucket_cap_mb"
                )
        self.buckets: List[List[_BucketSlot]] = []
        current: List[_BucketSlot] = []
        used = 0
        for t in tensors:
            n = t.numel()
            if used + n > cap_elems and current:
                self.buckets.append(current)
                current, used = [], 0
            current.append(_BucketSlot(t, used, n))
            used += n
        if current:
            self.buckets.append(current)

    def flatten(self, slots: List[_BucketSlot]) -> torch.Tensor:
        size = slots[-1].offset + slots[-1].numel
        flat = torch.empty(size, dtype=self.dtype, device=self.device)
        for s in slots:
            flat[s.offset : s.offset + s.numel].copy_(s.tensor.reshape(-1))
        return flat

    @staticmethod
    def scatter(flat: torch.Tensor, slots: List[_BucketSlot]) -> None:
        for s in slots:
            s.tensor.copy_(flat[s.offset : s.offset + s.numel].view_as(s.tensor))


# ---------------------------------------------------------------------------
# DiLoCo fragments
# ---------------------------------------------------------------------------


class _StreamingDiLoCoFragment:
    """One model fragment in the staggered outer-sync schedule.

    Life cycle per outer window:
      ``stage()``   — at ``sync_every - fragment_sync_delay``: pseudo-grad =
                      original − local per parameter; allreduce launched on
                      a side HIP stream so it overlaps the following inner
                      steps;
      ``collect()`` — wait the allreduce works and fence the side stream;
      ``commit()``  — at ``sync_every``: save local params, restore the
                      pre-window globals, run the commit barrier; on success
                      outer-step on the averaged pseudo-grads, re-snapshot,
                      and lerp local progress back in by
                      ``fragment_update_alpha``.
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
        assert sync_every >= 1

        self._manager = manager
        self._model_fragment = model_fragment
        self._fragment_id = fragment_id
        self._fragment_sync_offset = fragment_sync_offset
        self._local_optimizer = inner_optimizer
        self._outer_optimizer = outer_optimizer
        self._sync_every = sync_every
        self._fragment_sync_delay = fragment_sync_delay
        self._fragment_update_alpha = fragment_update_alpha
        self.should_quantize = should_quantize

        if bucket_cap_mb is not None:
            self.bucket_cap_mb = int(bucket_cap_mb * 1024 * 1024)
        self.use_bucketization = (
            True
            if os.getenv(USE_BUCKETIZATION_ENV, "False") == "True"
            else use_bucketization
        )

        # side HIP stream: the outer allreduce overlaps inner compute
        self._stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if torch.cuda.is_available() else None
        )
        self._stream_fence: Optional[torch.cuda.Event] = None
        self._inflight: List[Work] = []

        self._pseudograds: Dict[str, torch.Tensor] = {}
        self._local_parameters: Dict[str, torch.Tensor] = {}
        # pre-window snapshot; also the rollback point on commit failure
        self.original_parameters: Dict[str, torch.Tensor] = {}
        backup = backup_device or torch.device("cpu")
        for name, p in self._model_fragment.named_parameters():
            shape_src = extract_local_tensor(p.data) if (
                DTensor is not None and isinstance(p, DTensor)
            ) else p
            t = torch.empty(
                *tuple(shape_src.shape), dtype=shape_src.dtype, device=backup
            )
            if pin_memory and t.device.type == "cpu" and torch.cuda.is_available():
                t = t.pin_memory()
            self.original_parameters[name] = t

    # -- snapshots ----------------------------------------------------------

    def _named_params(self) -> Iterator[Tuple[str, torch.Tensor]]:
        return self._model_fragment.named_parameters()

    @torch.profiler.record_function("torchft_amd::local_sgd::save_parameters")
    def save_parameters(self) -> None:
        with torch.no_grad():
            for name, p in self._named_params():
                self.original_parameters[name].copy_(
                    extract_local_tensor(p.data), non_blocking=True
                )

    @torch.profiler.record_function("torchft_amd::local_sgd::restore_parameters")
    def restore_parameters(self) -> None:
        with torch.no_grad():
            for name, p in self._named_params():
                p.data.copy_(
                    _like_param(p, self.original_parameters[name].to(p.device)),
                    non_blocking=False,
                )

    def register_state_dict_fn(self) -> None:
        """Expose backups + outer optimizer state through the manager so
        recovering replicas heal them along with the model."""
        key = f"StreamingDiLoCoFragment_{self._fragment_id}"

        def load_fn(state: Dict[str, Dict[str, torch.Tensor]]) -> None:
            for name, param in state["original_parameters"].items():
                if name in self.original_parameters:
                    self.original_parameters[name].copy_(param)
            self._outer_optimizer.load_state_dict(state["outer_optimizer"])

        def save_fn() -> Dict[str, Dict[str, torch.Tensor]]:
            return {
                "outer_optimizer": self._outer_optimizer.state_dict(),
                "original_parameters": {
                    name: extract_local_tensor(t)
                    for name, t in self.original_parameters.items()
                },
            }

        self._manager.register_state_dict_fn(key, load_fn, save_fn)

    # -- sync phases ---------------------------------------------------------

    @torch.profiler.record_function("torchft_amd::local_sgd::prepare_sync")
    def prepare_sync(self) -> None:
        """Phase 1: pseudo-gradients + allreduce launch on the side stream."""
        with torch.no_grad():
            for name, p in self._named_params():
                local = p.to_local() if (
                    DTensor is not None and isinstance(p, DTensor)
                ) else p
                self._pseudograds[name] = (
                    self.original_parameters[name].to(p.device) - local
                )

        assert not self._inflight
        if self._stream is not None:
            self._stream.wait_stream(torch.cuda.current_stream())
        with self._side_stream():
            if self.use_bucketization:
                self._launch_bucketized()
            else:
                for name in self._pseudograds:
                    self._inflight.append(
                        self._manager.allreduce(
                            self._pseudograds[name],
                            should_quantize=self.should_quantize,
                        )
                    )

    def _side_stream(self):
        return (
            torch.cuda.stream(self._stream)
            if self._stream is not None
            else nullcontext()
        )

    def _launch_bucketized(self) -> None:
        plan = _BucketPlan(list(self._pseudograds.values()), self.bucket_cap_mb)
        for slots in plan.buckets:
            flat = plan.flatten(slots)
            work = self._manager.allreduce(flat, should_quantize=self.should_quantize)

            # bind this bucket's (flat, slots) as defaults — each bucket's
            # callback must see its own ranges
            def unpack(value, flat=flat, slots=slots):  # noqa: B008
                _BucketPlan.scatter(flat, slots)
                return value

            work.get_future().then(lambda f, fn=unpack: fn(f.value()))
            self._inflight.append(work)

    def wait(self) -> None:
        """Drain a previously launched sync, if any."""
        if not self._inflight:
            return
        if self._stream is not None:
            assert self._stream_fence is not None
            self._stream_fence.synchronize()
            self._stream_fence = None
        self._inflight = []

    @torch.profiler.record_function("torchft_amd::local_sgd::perform_sync")
    def perform_sync(self) -> bool:
        """Phases 2+3: collect the allreduce, then commit or roll back."""
        assert self._inflight, "prepare_sync must run before perform_sync"

        with self._side_stream():
            for w in self._inflight:
                w.wait()
            if self._stream is not None:
                self._stream_fence = torch.cuda.Event()
                self._stream_fence.record()
        self.wait()

        # keep local progress for the merge, then roll back to the globals
        with torch.no_grad():
            self._local_parameters = {
                name: extract_local_tensor(p.data) for name, p in self._named_params()
            }
        self.restore_parameters()

        # NOTE: with a large fragment_sync_delay this can report success even
        # if the allreduce was aborted by a mid-flight reconfiguration; the
        # reference documents the same compromise to avoid extra aborts.
        committed = self._manager.should_commit()
        if committed:
            self._outer_step()
        self._outer_optimizer.zero_grad()
        self._local_parameters = {}
        return committed

    def _outer_step(self) -> None:
        """Install averaged pseudo-grads, outer-step, re-snapshot, merge."""
        with torch.no_grad():
            for name, p in self._named_params():
                p.grad = _like_param(p, self._pseudograds.pop(name))
        self._outer_optimizer.step()
        self.save_parameters()
        # p = (1-alpha) * global + alpha * local  (alpha=0 keeps the global)
        for name, p in self._named_params():
            p.data.lerp_(
                _like_param(p, self._local_parameters[name]),
                self._fragment_update_alpha,
            )


# alias kept for callers using the descriptive name
_Fragment = _StreamingDiLoCoFragment


# ---------------------------------------------------------------------------
# the staggered schedule
# ---------------------------------------------------------------------------


@dataclass(frozen=True)
class _SyncSchedule:
    """Pure schedule: which fragment phase fires at a given local step.

    Both decisions depend only on (local_step, window, delay) plus the
    committed manager step — values every replica agrees on — so all
    replicas stage and collect the same fragment in the same order. A
    divergent order would deadlock (A waits fragment 1 while B waits
    fragment 2); with a shared pure schedule that cannot happen.
    """

    window: int  # inner steps per fragment window
    delay: int  # overlap distance

    def stages_now(self, local_step: int) -> bool:
        return local_step == self.window - self.delay

    def commits_now(self, local_step: int) -> bool:
        return local_step == self.window


class DiLoCo:
    """(Streaming) DiLoCo driver.

    Hooks the inner optimizer; every ``sync_every // num_fragments`` steps
    one fragment stages its outer allreduce, ``fragment_sync_delay`` steps
    later it commits. Requires the Manager in synchronous-quorum mode: the
    schedule must be a pure function of the committed step on every replica.
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
        n = len(model_fragments)
        if isinstance(outer_optimizer, list) and len(outer_optimizer) != n:
            raise AssertionError(
                "The number of outer optimizers must match the number of fragments"
            )
        if manager._use_async_quorum:
            raise ValueError(
                "DiLoCo requires synchronous quorum; construct the Manager "
                "with use_async_quorum=False"
            )
        if sync_every < n:
            raise ValueError("Only 1 fragment can be synchronized at a time")
        if sync_every % n != 0:
            raise ValueError("sync_every must divide the number of fragments")
        window = sync_every // n
        if fragment_sync_delay >= window:
            raise ValueError("Fragment must be synced before it is reduced again")
        if not 0.0 <= fragment_update_alpha <= 1.0:
            raise ValueError("fragment_update_alpha must be between 0 and 1")

        self._manager = manager
        self._local_optimizer = inner_optimizer
        self._local_step = 0
        self._schedule = _SyncSchedule(window=window, delay=fragment_sync_delay)
        self._hooks: List[RemovableHandle] = []

        self._fragments: List[_StreamingDiLoCoFragment] = [
            _StreamingDiLoCoFragment(
                manager,
                frag,
                i,
                math.floor(window * (i + 1)),
                inner_optimizer,
                outer_optimizer[i]
                if isinstance(outer_optimizer, list)
                else outer_optimizer,
                sync_every,
                backup_device,
                pin_memory,
                use_bucketization,
                bucket_cap_mb,
                should_quantize,
                fragment_sync_delay,
                fragment_update_alpha,
            )
            for i, frag in enumerate(model_fragments)
        ]

        # snapshot before the first inner step, and expose to healing
        for f in self._fragments:
            f.save_parameters()
            f.register_state_dict_fn()

    # context manager: install the inner-optimizer hooks
    def __enter__(self) -> "DiLoCo":
        opt = self._local_optimizer
        self._hooks += [
            opt.register_step_pre_hook(self._pre_step),
            opt.register_step_post_hook(self._post_step),
        ]
        return self

    def __exit__(
        self,
        exc_type: Optional[Type[BaseException]],
        exc_value: Optional[BaseException],
        traceback: Optional[TracebackType],
    ) -> bool:
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
        return False

    def _pre_step(self, *_: object) -> None:
        self._manager.disallow_state_dict_read()

    def _current_fragment(self) -> int:
        # pure function of the COMMITTED step -> identical on all replicas
        return self._manager.current_step() % len(self._fragments)

    def _wait(self) -> None:
        for f in self._fragments:
            f.wait()

    def _restore_parameters(self) -> None:
        for f in self._fragments:
            f.restore_parameters()

    def _save_parameters(self) -> None:
        for f in self._fragments:
            f.save_parameters()

    def _post_step(self, *_: object) -> None:
        self._manager.allow_state_dict_read()
        self._local_step += 1

        if self._schedule.stages_now(self._local_step):
            self._manager.start_quorum()
            frag = self._current_fragment()
            logger.info(f"staging fragment={frag} local_step={self._local_step}")
            self._fragments[frag].prepare_sync()

        if self._schedule.commits_now(self._local_step):
            frag = self._current_fragment()
            logger.info(
                f"committing fragment={frag} local_step={self._local_step} "
                f"manager_step={self._manager.current_step()}"
            )
            self._fragments[frag].perform_sync()
            # on failure the fragment rolled back; the window replays
            self._local_step = 0
        elif self._local_step > self._schedule.window:
            raise AssertionError(
                f"{self._local_step=} overran the window {self._schedule.window=}"
            )
