"""Reconfigurable process groups for fault-tolerant training on MI355X.

Quorum membership changes at step granularity, so communicators must be
rebuildable at step granularity too. On MI355X the production backend is
RCCL over xGMI (torch's ``"nccl"`` backend IS RCCL on ROCm); rebuilds are
abort-and-recreate with non-blocking communicator init, and every issued
collective carries a user-space deadline whose expiry calls ``abort()`` —
a dead peer unblocks the survivors instead of wedging or killing them.

Semantics matched against the reference (torchft/process_group.py: ABC
:131-399, wrapper :402-640, Gloo :643-711, NCCL :780-892, Dummy :1005-1134,
error-swallowing :1176-1249, fake :1252-1317, managed :1320-1353) but the
structure here is different by design:

* every collective funnels through ONE dispatch point (``_issue``) that
  applies the timeout policy, records the op in a flight-recorder ring
  buffer, and attaches the deadline guard — there is no per-backend hook
  hierarchy;
* each ``ProcessGroupWrapper`` keeps an ``_OpLog`` of the last N issued
  collectives (op, shapes, bytes, quorum); ``abort()`` dumps it as JSON to
  ``TORCHFT_ABORT_DUMP_DIR`` so a wedged RCCL collective on a production
  job is debuggable post-mortem (flight-recorder analog of
  torchft/manager.py:815-824).

Subprocess-isolated ("Baby") variants live in ``baby_process_group.py``.
"""

from __future__ import annotations

import json
import logging
import os
import threading
import time
import warnings
from collections import deque
from contextlib import contextmanager
from datetime import timedelta
from typing import TYPE_CHECKING, Callable, Generator, List, Optional, Union

import torch
import torch.distributed as dist
from torch.distributed import (
    PrefixStore,
    ProcessGroup as BaseProcessGroup,
    Store,
    TCPStore,
)
from torch.distributed.distributed_c10d import (
    AllgatherOptions,
    AllreduceCoalescedOptions,
    AllreduceOptions,
    AllToAllOptions,
    BarrierOptions,
    BroadcastOptions,
    ReduceOp,
    ReduceScatterOptions,
    Work,
)
from torch.futures import Future

from torchft_amd.futures import context_timeout, stream_timeout
from torchft_amd.utils import synchronize
from torchft_amd.work import _DummyWork

if TYPE_CHECKING:
    from torchft_amd.manager import Manager

logger: logging.Logger = logging.getLogger(__name__)

ABORT_DUMP_DIR_ENV = "TORCHFT_ABORT_DUMP_DIR"


def create_store_client(
    store_addr: str, timeout: timedelta, cache: Optional[dict] = None
) -> Store:
    """``host:port/prefix`` -> PrefixStore over a TCPStore client.

    With ``cache`` (a per-process-group dict), the underlying TCPStore
    connection is reused across reconfigures: a quorum change rotates only
    the prefix, so re-dialing the store server would pay TCP connect +
    handshake for nothing. The cache must never be shared between ranks (a
    TCPStore client is not safe for concurrent use from thread-ranks), so
    it is owned by the ProcessGroup instance, not the module.
    """
    host, _, rest = store_addr.partition(":")
    port, _, prefix = rest.partition("/")
    key = (host, int(port))
    store = cache.get(key) if cache is not None else None
    if store is None:
        store = TCPStore(
            host_name=host,
            port=int(port),
            is_master=False,
            wait_for_workers=False,
            timeout=timeout,
        )
        if cache is not None:
            cache[key] = store
    return PrefixStore(prefix, store)


# ---------------------------------------------------------------------------
# flight recorder
# ---------------------------------------------------------------------------


def _tensor_brief(t: object) -> object:
    if isinstance(t, torch.Tensor):
        return [str(t.dtype).replace("torch.", ""), list(t.shape)]
    if isinstance(t, (list, tuple)):
        return [_tensor_brief(x) for x in t]
    return None


class _OpLog:
    """Ring buffer of recently issued collectives.

    Cheap enough for the hot path (a dict append, no device work); dumped to
    disk when the group aborts so the post-mortem shows exactly which
    collective wedged and what was in flight.
    """

    def __init__(self, capacity: int = 64) -> None:
        self._ring: deque = deque(maxlen=capacity)
        self._seq = 0
        self._mu = threading.Lock()

    def record(self, op: str, tensors: object, quorum_id: object) -> int:
        with self._mu:
            self._seq += 1
            self._ring.append(
                {
                    "seq": self._seq,
                    "op": op,
                    "tensors": _tensor_brief(tensors),
                    "quorum_id": quorum_id,
                    "time": time.time(),
                    "status": "issued",
                }
            )
            return self._seq

    def mark(self, seq: int, status: str) -> None:
        with self._mu:
            for rec in reversed(self._ring):
                if rec["seq"] == seq:
                    rec["status"] = status
                    return

    def snapshot(self) -> List[dict]:
        with self._mu:
            return [dict(r) for r in self._ring]

    def dump(self, tag: str) -> Optional[str]:
        """Write the ring to ``TORCHFT_ABORT_DUMP_DIR`` (no-op if unset)."""
        dump_dir = os.environ.get(ABORT_DUMP_DIR_ENV)
        if not dump_dir:
            return None
        try:
            os.makedirs(dump_dir, exist_ok=True)
            path = os.path.join(dump_dir, f"oplog_{tag}_{os.getpid()}.json")
            with open(path, "w") as f:
                json.dump(self.snapshot(), f, indent=1)
            return path
        except OSError as e:  # never let diagnostics kill the abort path
            logger.warning(f"op-log dump failed: {e}")
            return None


# ---------------------------------------------------------------------------
# the reconfigurable ProcessGroup surface
# ---------------------------------------------------------------------------


class ProcessGroup(BaseProcessGroup):
    """torch ProcessGroup extended with quorum-driven reconfiguration.

    Adds ``configure()`` (rebuild against a fresh per-quorum store prefix),
    ``abort()``, ``errored()`` and ``set_timeout()``, plus registration into
    the c10d registry so functional collectives and DeviceMesh can address
    the group.
    """

    def __init__(self, *args: object, **kwargs: object) -> None:
        super().__init__(*args, **kwargs)  # pyre-ignore[6]
        self._group_name: Optional[str] = None

    def _absent(self, op: str) -> NotImplementedError:
        return NotImplementedError(f"{type(self).__name__} does not implement {op}")

    # collective surface — concrete groups override what they support
    def allgather(
        self,
        output_tensors: List[List[torch.Tensor]],
        input_tensor: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        raise self._absent("allgather")

    def allgather_into_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        raise self._absent("allgather_into_tensor_coalesced")

    def allreduce(
        self,
        tensors: List[torch.Tensor],
        opts: Union[AllreduceOptions, ReduceOp],
    ) -> Work:
        raise self._absent("allreduce")

    def allreduce_coalesced(
        self,
        tensors: List[torch.Tensor],
        opts: AllreduceCoalescedOptions,
    ) -> Work:
        raise self._absent("allreduce_coalesced")

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        raise self._absent("alltoall_base")

    def barrier(self, opts: BarrierOptions) -> Work:
        raise self._absent("barrier")

    def broadcast(
        self, tensor_list: List[torch.Tensor], opts: BroadcastOptions
    ) -> Work:
        raise self._absent("broadcast")

    def broadcast_one(self, tensor: torch.Tensor, root: int) -> Work:
        opts = BroadcastOptions()
        opts.rootRank = root
        return self.broadcast([tensor], opts)

    def recv(self, tensors: List[torch.Tensor], src_rank: int, tag: int) -> Work:
        raise self._absent("recv")

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: ReduceScatterOptions,
    ) -> Work:
        raise self._absent("reduce_scatter")

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> Work:
        raise self._absent("reduce_scatter_tensor_coalesced")

    def send(self, tensors: List[torch.Tensor], dst_rank: int, tag: int) -> Work:
        raise self._absent("send")

    # reconfiguration surface
    def configure(
        self,
        store_addr: str,
        replica_id: str,
        rank: int,
        world_size: int,
        quorum_id: Optional[int] = None,
        group_rank: Optional[int] = None,
        group_world_size: Optional[int] = None,
        global_ranks: Optional[list[int]] = None,
    ) -> None:
        """Rebuild the communicator against ``store_addr`` (a per-quorum
        ``host:port/prefix``). Blocks until the new communicator exists."""
        raise self._absent("configure")

    def size(self) -> int:
        raise self._absent("size")

    def getBackendName(self) -> str:
        raise self._absent("getBackendName")

    def abort(self) -> None:
        pass

    def shutdown(self) -> None:
        pass

    def errored(self) -> Optional[Exception]:
        """Async error requiring reconfiguration, if any."""
        return None

    def set_timeout(self, timeout: timedelta) -> None:
        raise self._absent("set_timeout")

    # c10d registry integration (functional collectives / DeviceMesh).
    # Elastic worlds can't be described to DeviceMesh, so the group is
    # registered under world size 1 and its real size changes underneath.
    def _register(self, name: str) -> str:
        group_name = f"{self.getBackendName()}:{name}"

        def factory(
            prefix_store: PrefixStore, rank: int, world_size: int, timeout: float
        ) -> "ProcessGroup":
            return self

        devices = ["cpu", "cuda"] if torch.cuda.is_available() else ["cpu"]
        dist.Backend.register_backend(group_name, factory, devices=devices)
        return group_name

    def register(self, name: str) -> "ProcessGroup":
        """Register with the global c10d registry; call at most once."""
        return dist.new_group(
            ranks=[dist.get_rank()],
            backend=self._register(name),
            group_desc=f"{self.getBackendName()}:{name}",
            timeout=timedelta(seconds=60.0),
        )

    @property
    def group_name(self) -> str:
        if self._group_name is None:
            raise ValueError("ProcessGroup name not set")
        return self._group_name

    def _set_group_name(self, name: str) -> None:
        self._group_name = name

    def unregister(self) -> None:
        dist.destroy_process_group(self)

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}()"


class ProcessGroupWrapper(ProcessGroup):
    """Epoch-style reconfiguration around a torch process group.

    Every ``configure()`` starts a new communicator epoch: the previous
    communicator is aborted, a store client is opened against the quorum's
    prefix, and ``_build()`` creates the replacement. All collectives pass
    through ``_issue()``, the single point where the timeout policy, the
    flight-recorder record and the deadline guard are applied.
    """

    def __init__(
        self,
        timeout: timedelta = timedelta(seconds=60),
        pg: Optional[ProcessGroup] = None,
    ) -> None:
        super().__init__(0, 1)
        self._inner: Optional[BaseProcessGroup] = pg
        self._timeout = timeout
        self._replica_id: Optional[str] = None
        self._rank: Optional[int] = None
        self._quorum_id: Optional[int] = None
        self._group_rank: Optional[int] = None
        self._group_world_size: Optional[int] = None
        self._global_ranks: Optional[list[int]] = None
        self._store_cache: dict = {}
        self._reaper: Optional[threading.Thread] = None
        self._oplog = _OpLog()
        self.errors_logger: logging.Logger = logging.getLogger("torchft_errors")

    # -- epoch management --------------------------------------------------

    def configure(
        self,
        store_addr: str,
        replica_id: str,
        rank: int,
        world_size: int,
        quorum_id: Optional[int] = None,
        group_rank: Optional[int] = None,
        group_world_size: Optional[int] = None,
        global_ranks: Optional[list[int]] = None,
    ) -> None:
        self._replica_id = replica_id
        self._rank = rank
        self._quorum_id = quorum_id
        self._group_rank = group_rank
        self._group_world_size = group_world_size
        self._global_ranks = global_ranks

        inner = self._inner
        if isinstance(inner, ProcessGroup):
            # wrapping an already-reconfigurable group: delegate the epoch
            inner.configure(
                store_addr,
                replica_id,
                rank,
                world_size,
                quorum_id,
                group_rank,
                group_world_size,
                global_ranks,
            )
            return

        # Retire the old communicator in the BACKGROUND: on MI355X the RCCL
        # comm abort alone measures ~507 ms of the ~550 ms reconfigure
        # (scripts/measure_reconfigure.py) while building the replacement
        # takes ~45 ms — and the two touch disjoint communicator objects
        # (the caller already synchronized the device, so nothing is in
        # flight on the old one). The previous teardown is joined first so
        # aborts never pile up.
        if self._reaper is not None:
            self._reaper.join()
            self._reaper = None
        old, self._inner = self._inner, None
        if old is not None:
            self._reaper = threading.Thread(
                target=self._retire, args=(old,), name="torchft-pg-reaper"
            )
            self._reaper.start()

        store = create_store_client(
            store_addr, timeout=self._timeout, cache=self._store_cache
        )
        self._inner = self._build(store, rank, world_size)

    def _build(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        raise self._absent("_build")

    def abort(self, errored: bool = True) -> None:
        if errored:
            self.errors_logger.info(
                "",
                extra={
                    "job_id": os.environ.get("JOB_ID", "unknown"),
                    "replica_id": self._replica_id,
                    "rank": self._rank,
                    "quorum_id": self._quorum_id,
                    "error": "process_group_abort",
                },
            )
            path = self._oplog.dump(
                f"{self._replica_id or 'r'}_{self._rank or 0}_q{self._quorum_id}"
            )
            if path:
                logger.error(f"aborting; op log dumped to {path}")

        inner, self._inner = self._inner, None
        if inner is not None:
            self._retire(inner)

    @staticmethod
    def _retire(inner: BaseProcessGroup) -> None:
        """Tear down a communicator (comm abort on the device backend)."""
        if hasattr(inner, "abort"):
            inner.abort()
            return
        # stock torch pg: abort the device backend if one exists
        try:
            backend = (
                inner._get_backend(torch.device("cuda"))
                if torch.cuda.is_available()
                else None
            )
        except RuntimeError:
            backend = None
        if backend is not None and hasattr(backend, "abort"):
            backend.abort()

    def shutdown(self) -> None:
        if self._reaper is not None:
            self._reaper.join()
            self._reaper = None
        self._inner = None

    def set_timeout(self, timeout: timedelta) -> None:
        self._timeout = timeout

    @property
    def parent(self) -> BaseProcessGroup:
        assert self._inner is not None, "process group not initialized"
        return self._inner

    # -- single dispatch point ---------------------------------------------

    def _prepare_opts(self, opts: object) -> object:
        """Timeout-policy hook applied to every op's options."""
        return opts

    def _attach_deadline(self, work: Work, opts: object) -> Work:
        """Deadline-policy hook applied to every returned work."""
        return work

    @contextmanager
    def _launch_guard(self) -> Generator[None, None, None]:
        """Guards the (possibly blocking) op launch itself."""
        yield

    def _issue(
        self, op: str, tensors: object, opts: object, launch: Callable[[object], Work]
    ) -> Work:
        seq = self._oplog.record(op, tensors, self._quorum_id)
        try:
            with self._launch_guard():
                work = launch(self._prepare_opts(opts))
        except Exception:
            self._oplog.mark(seq, "launch_failed")
            raise
        self._oplog.mark(seq, "in_flight")
        return self._attach_deadline(work, opts)

    # -- collective surface -------------------------------------------------

    def allgather(
        self,
        output_tensors: List[List[torch.Tensor]],
        input_tensor: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        return self._issue(
            "allgather",
            input_tensor,
            opts,
            lambda o: self.parent.allgather(output_tensors, input_tensor, o),
        )

    def allgather_into_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        return self._issue(
            "allgather_into_tensor_coalesced",
            input_tensors,
            opts,
            lambda o: self.parent.allgather_into_tensor_coalesced(
                output_tensors, input_tensors, o
            ),
        )

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        return self._issue(
            "allreduce", tensors, opts, lambda o: self.parent.allreduce(tensors, o)
        )

    def allreduce_coalesced(
        self, tensors: List[torch.Tensor], opts: Union[AllreduceOptions, ReduceOp]
    ) -> Work:
        return self._issue(
            "allreduce_coalesced",
            tensors,
            opts,
            lambda o: self.parent.allreduce_coalesced(tensors, o),
        )

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        return self._issue(
            "alltoall_base",
            input_buffer,
            opts,
            lambda o: self.parent.alltoall_base(
                output_buffer, input_buffer, output_split_sizes, input_split_sizes, o
            ),
        )

    def barrier(self, opts: Optional[BarrierOptions] = None) -> Work:
        return self._issue(
            "barrier", None, opts, lambda o: self.parent.barrier(o)
        )

    def broadcast(self, tensor_list: List[torch.Tensor], opts: object) -> Work:
        return self._issue(
            "broadcast",
            tensor_list,
            opts,
            lambda o: self.parent.broadcast(tensor_list, o),
        )

    def recv(self, tensors: List[torch.Tensor], src_rank: int, tag: int) -> Work:
        return self._issue(
            "recv", tensors, None, lambda o: self.parent.recv(tensors, src_rank, tag)
        )

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: object,
    ) -> Work:
        return self._issue(
            "reduce_scatter",
            input_tensors,
            opts,
            lambda o: self.parent.reduce_scatter(output_tensors, input_tensors, o),
        )

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> Work:
        return self._issue(
            "reduce_scatter_tensor_coalesced",
            input_tensors,
            opts,
            lambda o: self.parent.reduce_scatter_tensor_coalesced(
                output_tensors, input_tensors, o
            ),
        )

    def send(self, tensors: List[torch.Tensor], dst_rank: int, tag: int) -> Work:
        return self._issue(
            "send", tensors, None, lambda o: self.parent.send(tensors, dst_rank, tag)
        )

    def size(self) -> int:
        return self.parent.size()

    def getBackendName(self) -> str:
        inner = self._inner
        if isinstance(inner, ProcessGroup):
            return inner.getBackendName()
        raise self._absent("getBackendName")

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}(pg={self._inner})"


class ProcessGroupGloo(ProcessGroupWrapper):
    """Reconfigurable Gloo group — CPU tests and the CPU control path."""

    def _build(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        from torch.distributed import ProcessGroupGloo as TorchGloo

        pg = BaseProcessGroup(store, rank, world_size)
        pg._set_default_backend(BaseProcessGroup.BackendType.GLOO)
        backend = TorchGloo(store, rank, world_size, self._timeout)
        backend._set_sequence_number_for_group()
        if self._global_ranks:
            backend.options.global_ranks_in_group = self._global_ranks
        if self._group_rank and self._group_world_size:
            backend.options.group_name = (
                f"torchft_quorum_{self._quorum_id}_rank_"
                f"{self._group_rank % self._group_world_size}"
            )
        pg._register_backend(
            torch.device("cpu"), BaseProcessGroup.BackendType.GLOO, backend
        )
        if torch.cuda.is_available():
            pg._register_backend(
                torch.device("cuda"), BaseProcessGroup.BackendType.GLOO, backend
            )
        return pg

    def getBackendName(self) -> str:
        return "torchft-gloo"

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: ReduceScatterOptions,
    ) -> None:
        raise RuntimeError("ProcessGroupGloo does not support reduce_scatter.")

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> None:
        raise RuntimeError(
            "ProcessGroupGloo does not support reduce_scatter_tensor_coalesced."
        )


class _DeadlineWork(Work):
    """Work that aborts its process group if completion misses the deadline.

    ``wait()`` guards both the CPU-side wait (which may block, e.g. a
    barrier) and the HIP-stream completion; either missing the deadline
    calls ``pg.abort()``, which unblocks the stream with an error that
    surfaces through ``errored()`` at commit time.
    """

    def __init__(self, pg: ProcessGroup, work: Work, deadline: timedelta) -> None:
        super().__init__()
        self._pg = pg
        self._work = work
        self._deadline = deadline

    def _expired(self) -> None:
        logger.error(f"collective missed {self._deadline} deadline; aborting pg")
        self._pg.abort()

    @contextmanager
    def _guard(self, deadline: timedelta) -> Generator[None, None, None]:
        with context_timeout(self._expired, deadline):
            yield
        stream_timeout(self._expired, deadline)

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        effective = timeout or self._deadline
        with self._guard(effective):
            if self._work is not None and not self._work.wait():
                return False
            if timeout is not None:
                torch.cuda.synchronize()
        return True

    def get_future(self) -> Future:
        fut = self._work.get_future()

        def enforce(f: Future) -> None:
            try:
                with self._guard(self._deadline):
                    f.wait()
            except Exception as e:  # noqa: BLE001
                logger.error(f"deadline enforcement failed: {e}")

        fut.add_done_callback(enforce)
        return fut


class ProcessGroupRCCL(ProcessGroupWrapper):
    """Reconfigurable RCCL process group — the MI355X production backend.

    Communicators are created non-blocking and torn down with comm-abort on
    reconfigure; the c10d watchdog timeout is cleared and replaced by the
    user-space ``_DeadlineWork`` guard so a dead peer aborts the group
    instead of crashing the trainer.

    xGMI note: each MI355X GPU drives 7 point-to-point links (~153 GB/s
    each). RCCL engages all of them given correct topology; nothing here
    pins channels — callers choose bucket sizes for per-link bandwidth.
    """

    def __init__(self, timeout: timedelta = timedelta(seconds=60.0)) -> None:
        super().__init__(timeout)
        self._use_abort: bool = torch.cuda.nccl.version() >= (2, 25)
        self._errored: Optional[Exception] = None

        env = "TORCH_NCCL_NONBLOCKING_TIMEOUT"
        if env not in os.environ:
            warnings.warn(
                f"{env} is not set, defaulting to {timeout}. If any "
                "nonblocking RCCL operations have already run this may "
                "result in the default timeout of 30 minutes and hangs."
            )
            os.environ[env] = str(timeout.total_seconds())

    # timeout policy: hand deadlines to user space, not the c10d watchdog
    def _prepare_opts(self, opts: object) -> object:
        if self._use_abort and hasattr(opts, "timeout"):
            opts.timeout = AllgatherOptions().timeout
        return opts

    def _attach_deadline(self, work: Work, opts: object) -> Work:
        if not self._use_abort:
            return work
        deadline = self._timeout
        if hasattr(opts, "timeout") and opts.timeout.total_seconds() > 0:
            deadline = opts.timeout
        return _DeadlineWork(self, work, deadline)

    @contextmanager
    def _launch_guard(self) -> Generator[None, None, None]:
        deadline = self._timeout

        def expired() -> None:
            logger.error(f"launch missed {deadline} deadline; aborting pg")
            self.abort()

        with context_timeout(expired, deadline):
            yield

    def _build(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        from torch.distributed import ProcessGroupNCCL as TorchRCCL

        self._errored = None

        opts = TorchRCCL.Options()
        opts.config.blocking = False
        if self._global_ranks:
            opts.global_ranks_in_group = self._global_ranks
        if self._group_rank and self._group_world_size:
            opts.group_name = (
                f"torchft_quorum_{self._quorum_id}_rank_"
                f"{self._group_rank % self._group_world_size}"
            )

        pg = BaseProcessGroup(store, rank, world_size)
        pg._set_default_backend(BaseProcessGroup.BackendType.NCCL)
        backend = TorchRCCL(store, rank, world_size, opts)
        backend._set_sequence_number_for_group()
        # pre-establish the communicator so the first collective after
        # reconfigure doesn't pay rendezvous latency
        backend.eager_connect_single_device(
            torch.device(torch.cuda.current_device())
        )
        pg._register_backend(
            torch.device("cuda"), BaseProcessGroup.BackendType.NCCL, backend
        )
        return pg

    def abort(self, errored: bool = True) -> None:
        # mark the error first: the abort unblocks in-flight streams and
        # errored() must already report by then
        self._errored = RuntimeError("aborted")
        super().abort(errored=errored)

    def errored(self) -> Optional[Exception]:
        synchronize()  # surface any async stream error first
        return self._errored

    def getBackendName(self) -> str:
        return "torchft-rccl"


# torch calls the accelerator backend NCCL; on ROCm it is RCCL.
ProcessGroupNCCL = ProcessGroupRCCL


class ProcessGroupDummy(ProcessGroup):
    """World-size-1 loopback group: inputs land in outputs, every op
    completes immediately. Soaks up DDP's init broadcast; test double."""

    def __init__(self, rank: int, world: int) -> None:
        super().__init__(rank, world)
        assert rank == 0 and world == 1
        self._rank = rank
        self._world = world
        self.wait_count = 0
        self.get_future_count = 0
        self.configure_count = 0
        self._work: List[Work] = []

    def configure(
        self,
        store_addr: str,
        replica_id: str,
        rank: int,
        world_size: int,
        quorum_id: Optional[int] = None,
        group_rank: Optional[int] = None,
        group_world_size: Optional[int] = None,
        global_ranks: Optional[list[int]] = None,
    ) -> None:
        self.configure_count += 1

    def _loopback(self, result: object) -> Work:
        w = _DummyWork(result)
        self._work.append(w)
        return w

    def allgather(self, output_tensors, input_tensor, opts: object) -> Work:
        for o, i in zip(output_tensors[0], input_tensor):
            o.copy_(i)
        return self._loopback(output_tensors)

    def allgather_into_tensor_coalesced(
        self, output_tensors, input_tensors, opts: AllgatherOptions
    ) -> Work:
        for o, i in zip(output_tensors, input_tensors):
            o.copy_(i)
        return self._loopback(output_tensors)

    def allreduce(self, tensors, opts: object) -> Work:
        return self._loopback(tensors)

    def allreduce_coalesced(self, tensors, opts: object) -> Work:
        return self._loopback(tensors)

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        output_buffer.copy_(input_buffer)
        return self._loopback([output_buffer])

    def barrier(self, opts: Optional[BarrierOptions] = None) -> Work:
        return _DummyWork(None)

    def broadcast(self, tensor_list, opts: object) -> Work:
        return self._loopback(tensor_list)

    def recv(self, tensors, src_rank: int, tag: int) -> Work:
        return _DummyWork(None)

    def reduce_scatter(self, output_tensors, input_tensors, opts: object) -> Work:
        for o, i in zip(output_tensors, input_tensors[0]):
            o.copy_(i)
        return self._loopback(output_tensors)

    def reduce_scatter_tensor_coalesced(
        self, output_tensors, input_tensors, opts: ReduceScatterOptions
    ) -> Work:
        for o, i in zip(output_tensors, input_tensors):
            o.copy_(i)
        return self._loopback(output_tensors)

    def send(self, tensors, dst_rank: int, tag: int) -> Work:
        return _DummyWork(None)

    def size(self) -> int:
        return self._world

    def getBackendName(self) -> str:
        return "torchft-dummy"


class _ErrorSwallowingWork(Work):
    def __init__(
        self,
        pg: "ErrorSwallowingProcessGroupWrapper",
        work: Work,
        default_result: object,
    ) -> None:
        super().__init__()
        self._pg = pg
        self._work = work
        self._default_result = default_result

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        try:
            self._work.wait()
        except Exception as e:  # noqa: BLE001
            self._pg.report_error(e)
        return True

    def get_future(self) -> Future:
        def absorb(f: Future) -> object:
            try:
                return f.value()
            except Exception as e:  # noqa: BLE001
                logger.exception(f"swallowed collective error: {e}")
                self._pg.report_error(e)
                return self._default_result

        return self._work.get_future().then(absorb)


class ErrorSwallowingProcessGroupWrapper(ProcessGroupWrapper):
    """Collective errors become dummy successes plus a sticky error flag;
    after the first error every op is skipped until ``configure``."""

    def __init__(self, pg: ProcessGroup) -> None:
        super().__init__(pg=pg)
        self._error: Optional[Exception] = None

    def configure(self, *args: object, **kwargs: object) -> None:
        self._error = None
        super().configure(*args, **kwargs)  # type: ignore[arg-type]

    def report_error(self, e: Exception) -> None:
        self._error = e

    def error(self) -> Optional[Exception]:
        return self._error

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        if self._error is not None:
            return _DummyWork(tensors)
        try:
            return _ErrorSwallowingWork(self, super().allreduce(tensors, opts), tensors)
        except Exception as e:  # noqa: BLE001
            self.report_error(e)
            return _DummyWork(tensors)


class FakeProcessGroupWrapper(ProcessGroupWrapper):
    """Test-only fault injection: the next op's future raises on demand."""

    def __init__(self, pg: ProcessGroup) -> None:
        super().__init__(pg=pg)
        self._future_error: Optional[Exception] = None

    def configure(self, *args: object, **kwargs: object) -> None:
        self._future_error = None
        super().configure(*args, **kwargs)  # type: ignore[arg-type]

    def report_future_error(self, e: Exception) -> None:
        self._future_error = e

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        work = super().allreduce(tensors, opts)
        injected, self._future_error = self._future_error, None
        if injected is None:
            return work

        def detonate(fut: Future) -> List[torch.Tensor]:
            raise injected

        poisoned_fut = work.get_future().then(detonate)

        class _PoisonedWork(Work):
            def wait(self, timeout: Optional[timedelta] = None) -> bool:
                work.wait()
                raise injected

            def get_future(self) -> Future:
                return poisoned_fut

        return _PoisonedWork()


class ManagedProcessGroup(ProcessGroupWrapper):
    """Presents a Manager as a PG so stock torch DDP/FSDP hooks route
    through the fault-tolerant allreduce; ``size()`` is the live quorum
    participant count."""

    def __init__(self, manager: "Manager") -> None:
        super().__init__(pg=manager._pg)
        self._manager = manager

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        assert len(tensors) == 1
        if isinstance(opts, ReduceOp):
            return self._manager.allreduce(tensors[0], reduce_op=opts)
        if isinstance(opts, AllreduceOptions):
            return self._manager.allreduce(tensors[0], reduce_op=opts.reduceOp)
        raise AssertionError(f"unsupported allreduce opts: {opts!r}")

    def size(self) -> int:
        return self._manager.num_participants()

    def getBackendName(self) -> str:
        return self._manager._pg.getBackendName()
