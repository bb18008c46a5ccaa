"""Reconfigurable process groups for fault-tolerant training on MI355X.

The fault-tolerance protocol requires communicators that can be aborted and
rebuilt at step granularity when quorum membership changes. On MI355X the
production backend is RCCL over xGMI (PyTorch's ``"nccl"`` backend IS RCCL on
ROCm); reconfiguration is abort-and-recreate with non-blocking communicator
init plus user-space timeouts that call ``abort()`` so a wedged collective can
never take down the trainer.

Reference parity (semantics, not code): torchft/process_group.py —
``ProcessGroup`` ABC with ``configure/abort/errored/set_timeout`` (:131-399),
``ProcessGroupWrapper`` (:402-640), ``ProcessGroupGloo`` (:643-711),
``ProcessGroupNCCL`` (:780-892), ``ProcessGroupDummy`` (:1005-1134),
``ErrorSwallowingProcessGroupWrapper`` (:1176-1249), ``FakeProcessGroupWrapper``
(:1252-1317), ``ManagedProcessGroup`` (:1320-1353).

Subprocess-isolated ("Baby") variants live in ``baby_process_group.py``.
"""

from __future__ import annotations

import logging
import os
import warnings
from contextlib import contextmanager
from datetime import timedelta
from typing import TYPE_CHECKING, Generator, List, Optional, TypeVar, Union

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

T = TypeVar("T")


def create_store_client(store_addr: str, timeout: timedelta) -> Store:
    """Create a PrefixStore(TCPStore) client from ``host:port/prefix``."""
    host, _, rest = store_addr.partition(":")
    port, _, prefix = rest.partition("/")
    store = TCPStore(
        host_name=host,
        port=int(port),
        is_master=False,
        wait_for_workers=False,
        timeout=timeout,
    )
    return PrefixStore(prefix, store)


class ProcessGroup(BaseProcessGroup):
    """Abstract reconfigurable process group.

    Adds ``configure()`` (rebuild the communicator against a fresh store
    prefix for a new quorum), ``abort()``, ``errored()`` and ``set_timeout()``
    on top of the stock torch collective surface, plus registration into the
    c10d backend registry so functional collectives / DeviceMesh work.
    """

    def __init__(self, *args: object, **kwargs: object) -> None:
        super().__init__(*args, **kwargs)  # pyre-ignore[6]
        self._group_name: Optional[str] = None

    # -- collective surface (implemented by subclasses) --------------------

    def allgather(
        self,
        output_tensors: List[List[torch.Tensor]],
        input_tensor: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def allgather_into_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def allreduce(
        self,
        tensors: List[torch.Tensor],
        opts: Union[AllreduceOptions, ReduceOp],
    ) -> Work:
        raise NotImplementedError("not implemented")

    def allreduce_coalesced(
        self,
        tensors: List[torch.Tensor],
        opts: AllreduceCoalescedOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def barrier(self, opts: BarrierOptions) -> Work:
        raise NotImplementedError("not implemented")

    def broadcast(self, tensor_list: List[torch.Tensor], opts: BroadcastOptions) -> Work:
        raise NotImplementedError("not implemented")

    def broadcast_one(self, tensor: torch.Tensor, root: int) -> Work:
        opts = BroadcastOptions()
        opts.rootRank = root
        return self.broadcast([tensor], opts)

    def recv(self, tensors: List[torch.Tensor], src_rank: int, tag: int) -> Work:
        raise NotImplementedError("not implemented")

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: ReduceScatterOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> Work:
        raise NotImplementedError("not implemented")

    def send(self, tensors: List[torch.Tensor], dst_rank: int, tag: int) -> Work:
        raise NotImplementedError("not implemented")

    # -- reconfiguration ---------------------------------------------------

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
        """Rebuild the communicator. ``store_addr`` must be a unique prefixed
        store address per quorum (``host:port/prefix``). Blocks until the new
        communicator exists; raises on failure."""
        raise NotImplementedError("not implemented")

    def size(self) -> int:
        raise NotImplementedError("not implemented")

    def getBackendName(self) -> str:
        raise NotImplementedError("not implemented")

    def _register(self, name: str) -> str:
        group_name = f"{self.getBackendName()}:{name}"

        # Resizable worlds don't fit DeviceMesh, so register as world-size-1.
        def create_pg(
            prefix_store: PrefixStore, rank: int, world_size: int, timeout: float
        ) -> "ProcessGroup":
            return self

        devices = ["cpu"]
        if torch.cuda.is_available():
            devices.append("cuda")
        dist.Backend.register_backend(group_name, create_pg, devices=devices)
        return group_name

    def register(self, name: str) -> "ProcessGroup":
        """Register with the c10d global registry (enables functional
        collectives). Call at most once."""
        group_name = self._register(name)
        return dist.new_group(
            ranks=[dist.get_rank()],
            backend=group_name,
            group_desc=group_name,
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

    def abort(self) -> None:
        pass

    def shutdown(self) -> None:
        pass

    def errored(self) -> Optional[Exception]:
        """Async error that requires reconfiguration, if any."""
        return None

    def set_timeout(self, timeout: timedelta) -> None:
        raise NotImplementedError("set_timeout not implemented")

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}()"


class ProcessGroupWrapper(ProcessGroup):
    """Reconfiguration by abort-and-recreate around any inner process group.

    ``configure()`` aborts the old backend, creates a fresh TCPStore client
    against the per-quorum prefix and calls ``_create_pg``. Subclasses hook
    ``_create_pg`` / ``_wrap_work`` / ``_opts_hook`` / ``_run_context``.
    """

    def __init__(
        self,
        timeout: timedelta = timedelta(seconds=60),
        pg: Optional[ProcessGroup] = None,
    ) -> None:
        super().__init__(0, 1)
        self._pg: Optional[BaseProcessGroup] = pg
        self._timeout = timeout
        self._replica_id: Optional[str] = None
        self._rank: Optional[int] = None
        self._quorum_id: Optional[int] = None
        self._group_rank: Optional[int] = None
        self._group_world_size: Optional[int] = None
        self._global_ranks: Optional[list[int]] = None
        self.errors_logger: logging.Logger = logging.getLogger("torchft_errors")

    def getBackendName(self) -> str:
        pg = self._pg
        if isinstance(pg, ProcessGroup):
            return pg.getBackendName()
        raise NotImplementedError("not implemented")

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
        pg = self._pg
        self._replica_id = replica_id
        self._quorum_id = quorum_id
        self._group_rank = group_rank
        self._group_world_size = group_world_size
        self._rank = rank
        self._global_ranks = global_ranks
        if isinstance(pg, ProcessGroup):
            pg.configure(
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

        self.abort(errored=False)
        store = create_store_client(store_addr, timeout=self._timeout)
        self._pg = self._create_pg(store, rank, world_size)

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
        pg = self._pg
        if pg is not None:
            if hasattr(pg, "abort"):
                pg.abort()
            else:
                backend = None
                try:
                    if torch.cuda.is_available():
                        backend = pg._get_backend(torch.device("cuda"))
                except RuntimeError:
                    backend = None
                if backend is not None and hasattr(backend, "abort"):
                    backend.abort()
            self._pg = None

    def shutdown(self) -> None:
        self._pg = None

    def _create_pg(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        raise NotImplementedError("not implemented")

    def _wrap_work(self, work: Work, opts: object) -> Work:
        return work

    def _opts_hook(self, opts: T) -> T:
        return opts

    @contextmanager
    def _run_context(self) -> Generator[None, None, None]:
        yield

    def set_timeout(self, timeout: timedelta) -> None:
        self._timeout = timeout

    # -- collectives: forward to the inner pg through the hooks ------------

    def allgather(
        self,
        output_tensors: List[List[torch.Tensor]],
        input_tensor: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.allgather(output_tensors, input_tensor, self._opts_hook(opts)),
                opts,
            )

    def allgather_into_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.allgather_into_tensor_coalesced(
                    output_tensors, input_tensors, self._opts_hook(opts)
                ),
                opts,
            )

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        with self._run_context():
            return self._wrap_work(self.parent.allreduce(tensors, self._opts_hook(opts)), opts)

    def allreduce_coalesced(
        self, tensors: List[torch.Tensor], opts: Union[AllreduceOptions, ReduceOp]
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.allreduce_coalesced(tensors, self._opts_hook(opts)), opts
            )

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.alltoall_base(
                    output_buffer,
                    input_buffer,
                    output_split_sizes,
                    input_split_sizes,
                    self._opts_hook(opts),
                ),
                opts,
            )

    def barrier(self, opts: Optional[BarrierOptions] = None) -> Work:
        with self._run_context():
            return self._wrap_work(self.parent.barrier(self._opts_hook(opts)), opts)

    def broadcast(self, tensor_list: List[torch.Tensor], opts: object) -> Work:
        with self._run_context():
            return self._wrap_work(self.parent.broadcast(tensor_list, self._opts_hook(opts)), opts)

    def recv(self, tensors: List[torch.Tensor], src_rank: int, tag: int) -> Work:
        with self._run_context():
            return self._wrap_work(self.parent.recv(tensors, src_rank, tag), None)

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: object,
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.reduce_scatter(output_tensors, input_tensors, self._opts_hook(opts)),
                opts,
            )

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> Work:
        with self._run_context():
            return self._wrap_work(
                self.parent.reduce_scatter_tensor_coalesced(
                    output_tensors, input_tensors, self._opts_hook(opts)
                ),
                opts,
            )

    def send(self, tensors: List[torch.Tensor], dst_rank: int, tag: int) -> Work:
        with self._run_context():
            return self._wrap_work(self.parent.send(tensors, dst_rank, tag), None)

    def size(self) -> int:
        return self.parent.size()

    @property
    def parent(self) -> BaseProcessGroup:
        assert self._pg is not None, "process group not initialized"
        return self._pg

    def __repr__(self) -> str:
        return f"{self.__class__.__name__}(pg={self._pg})"


class ProcessGroupGloo(ProcessGroupWrapper):
    """Reconfigurable Gloo process group (CPU tests and CPU fallback)."""

    def _create_pg(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        from torch.distributed import ProcessGroupGloo as BaseProcessGroupGloo

        pg = BaseProcessGroup(store, rank, world_size)
        pg._set_default_backend(BaseProcessGroup.BackendType.GLOO)
        backend_class = BaseProcessGroupGloo(store, rank, world_size, self._timeout)
        backend_class._set_sequence_number_for_group()

        if self._global_ranks:
            backend_class.options.global_ranks_in_group = self._global_ranks
        if self._group_rank and self._group_world_size:
            backend_class.options.group_name = (
                f"torchft_quorum_{self._quorum_id}_rank_"
                f"{self._group_rank % self._group_world_size}"
            )

        pg._register_backend(torch.device("cpu"), BaseProcessGroup.BackendType.GLOO, backend_class)
        if torch.cuda.is_available():
            pg._register_backend(
                torch.device("cuda"), BaseProcessGroup.BackendType.GLOO, backend_class
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
        raise RuntimeError("ProcessGroupGloo does not support reduce_scatter_tensor_coalesced.")


class _WorkAcceleratorTimeout(Work):
    """Wraps a Work with a user-space HIP-stream timeout that aborts the PG
    instead of letting the RCCL watchdog crash the process."""

    def __init__(self, pg: ProcessGroup, work: Work, timeout: timedelta) -> None:
        super().__init__()
        self._pg = pg
        self._work = work
        self._timeout = timeout

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        async_timeout = timeout or self._timeout
        with self._stream_timeout(self._pg, async_timeout):
            if self._work is not None:
                if not self._work.wait():
                    return False
            if timeout is not None:
                torch.cuda.synchronize()
            return True

    @classmethod
    @contextmanager
    def _stream_timeout(
        cls, pg: ProcessGroup, timeout: timedelta
    ) -> Generator[None, None, None]:
        def callback() -> None:
            logger.error(f"aborting after {timeout}!")
            pg.abort()

        # .wait() itself must be cancellable (e.g. a blocking barrier) ...
        with context_timeout(callback, timeout):
            yield
        # ... and the HIP stream must complete within the timeout too.
        stream_timeout(callback, timeout)

    def get_future(self) -> Future[object]:
        fut = self._work.get_future()

        def done_callback(fut: Future[object]) -> None:
            try:
                with self._stream_timeout(self._pg, self._timeout):
                    fut.wait()
            except Exception as e:  # noqa: BLE001
                logger.error(f"done callback failed: {e}")

        fut.add_done_callback(done_callback)
        return fut


class ProcessGroupRCCL(ProcessGroupWrapper):
    """Reconfigurable RCCL process group — the MI355X production backend.

    torch's ``"nccl"`` backend IS RCCL on ROCm. Communicators are created
    non-blocking and aborted with the RCCL comm-abort path on reconfigure;
    per-op timeouts are cleared and replaced with user-space timeouts that
    call ``abort()`` so a dead peer unblocks the survivors without killing
    them (reference semantics: torchft/process_group.py:780-892).

    xGMI note: each MI355X GPU has 7 point-to-point xGMI links (~153 GB/s
    each); RCCL engages them all given correct topology, so the wrapper adds
    no channel pinning — bucket sizing for per-link bandwidth is handled by
    the callers (Manager/DiLoCo defaults).
    """

    def __init__(self, timeout: timedelta = timedelta(seconds=60.0)) -> None:
        super().__init__(timeout)
        self._use_abort: bool = torch.cuda.nccl.version() >= (2, 25)
        self._errored: Optional[Exception] = None

        NONBLOCKING_TIMEOUT_ENV = "TORCH_NCCL_NONBLOCKING_TIMEOUT"
        if NONBLOCKING_TIMEOUT_ENV not in os.environ:
            warnings.warn(
                f"{NONBLOCKING_TIMEOUT_ENV} is not set, defaulting to {timeout}. "
                "If any nonblocking RCCL operations have already run this may "
                "result in the default timeout of 30 minutes and hangs on error."
            )
            os.environ[NONBLOCKING_TIMEOUT_ENV] = str(timeout.total_seconds())

    def _opts_hook(self, opts: T) -> T:
        if not self._use_abort:
            return opts
        # Clear the c10d watchdog timeout; our user-space timeout aborts
        # instead of crashing.
        if hasattr(opts, "timeout"):
            opts.timeout = AllgatherOptions().timeout
        return opts

    def _wrap_work(self, work: Work, opts: object) -> Work:
        if not self._use_abort:
            return work
        timeout = self._timeout
        if hasattr(opts, "timeout") and opts.timeout.total_seconds() > 0:
            timeout = opts.timeout
        return _WorkAcceleratorTimeout(self, work, timeout)

    @contextmanager
    def _run_context(self) -> Generator[None, None, None]:
        timeout: timedelta = self._timeout

        def callback() -> None:
            logger.error(f"aborting after {timeout}!")
            self.abort()

        with context_timeout(callback, timeout):
            yield

    def _create_pg(self, store: Store, rank: int, world_size: int) -> BaseProcessGroup:
        from torch.distributed import ProcessGroupNCCL as BaseProcessGroupRCCL

        self._errored = None

        opts = BaseProcessGroupRCCL.Options()
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
        backend_class = BaseProcessGroupRCCL(store, rank, world_size, opts)
        backend_class._set_sequence_number_for_group()
        # Pre-establish the communicator on the local device so the first
        # collective after reconfigure doesn't pay rendezvous latency.
        backend_class.eager_connect_single_device(torch.device(torch.cuda.current_device()))
        pg._register_backend(torch.device("cuda"), BaseProcessGroup.BackendType.NCCL, backend_class)
        return pg

    def abort(self, errored: bool = True) -> None:
        # Set the error before aborting so errored() reports correctly once
        # the abort unblocks the stream.
        self._errored = RuntimeError("aborted")
        super().abort(errored=errored)

    def errored(self) -> Optional[Exception]:
        synchronize()  # ensure in-flight work surfaced any async error
        return self._errored

    def getBackendName(self) -> str:
        return "torchft-rccl"


# The reference's name for the accelerator PG; on ROCm it is the same thing.
ProcessGroupNCCL = ProcessGroupRCCL


class ProcessGroupDummy(ProcessGroup):
    """World-size-1 no-op PG: copies inputs to outputs and succeeds.

    Soaks up DDP's init broadcast and serves as a test double.
    """

    def __init__(self, rank: int, world: int) -> None:
        super().__init__(rank, world)
        assert rank == 0
        assert world == 1
        self._rank = rank
        self._world = world
        self.wait_count = 0
        self.get_future_count = 0
        self._work: List[Work] = []
        self.configure_count = 0

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

    def allgather(
        self,
        output_tensors: List[List[torch.Tensor]],
        input_tensor: List[torch.Tensor],
        opts: object,
    ) -> Work:
        for o, i in zip(output_tensors[0], input_tensor):
            o.copy_(i)
        res = _DummyWork(output_tensors)
        self._work.append(res)
        return res

    def allgather_into_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: AllgatherOptions,
    ) -> Work:
        for o, i in zip(output_tensors, input_tensors):
            o.copy_(i)
        res = _DummyWork(output_tensors)
        self._work.append(res)
        return res

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        res = _DummyWork(tensors)
        self._work.append(res)
        return res

    def allreduce_coalesced(
        self, tensors: List[torch.Tensor], opts: Union[AllreduceOptions, ReduceOp]
    ) -> Work:
        res = _DummyWork(tensors)
        self._work.append(res)
        return res

    def alltoall_base(
        self,
        output_buffer: torch.Tensor,
        input_buffer: torch.Tensor,
        output_split_sizes: List[int],
        input_split_sizes: List[int],
        opts: AllToAllOptions,
    ) -> Work:
        output_buffer.copy_(input_buffer)
        res = _DummyWork([output_buffer])
        self._work.append(res)
        return res

    def barrier(self, opts: Optional[BarrierOptions] = None) -> Work:
        return _DummyWork(None)

    def broadcast(self, tensor_list: List[torch.Tensor], opts: object) -> Work:
        res = _DummyWork(tensor_list)
        self._work.append(res)
        return res

    def recv(self, tensors: List[torch.Tensor], src_rank: int, tag: int) -> Work:
        return _DummyWork(None)

    def reduce_scatter(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[List[torch.Tensor]],
        opts: object,
    ) -> Work:
        for o, i in zip(output_tensors, input_tensors[0]):
            o.copy_(i)
        res = _DummyWork(output_tensors)
        self._work.append(res)
        return res

    def reduce_scatter_tensor_coalesced(
        self,
        output_tensors: List[torch.Tensor],
        input_tensors: List[torch.Tensor],
        opts: ReduceScatterOptions,
    ) -> Work:
        for o, i in zip(output_tensors, input_tensors):
            o.copy_(i)
        res = _DummyWork(output_tensors)
        self._work.append(res)
        return res

    def send(self, tensors: List[torch.Tensor], dst_rank: int, tag: int) -> Work:
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

    def get_future(self) -> Future[object]:
        fut = self._work.get_future()

        def callback(fut: Future[List[torch.Tensor]]) -> object:
            try:
                return fut.value()
            except Exception as e:  # noqa: BLE001
                logger.exception(f"got exception in future -- skipping remaining: {e}")
                self._pg.report_error(e)
                return self._default_result

        return fut.then(callback)


class ErrorSwallowingProcessGroupWrapper(ProcessGroupWrapper):
    """Converts collective errors into dummy successes plus a sticky error
    flag; after the first error all ops are skipped until ``configure``."""

    def __init__(self, pg: ProcessGroup) -> None:
        super().__init__(pg=pg)
        self._error: Optional[Exception] = None

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
        self._error = None
        super().configure(
            store_addr,
            replica_id,
            rank,
            world_size,
            quorum_id,
            group_rank,
            group_world_size,
            global_ranks,
        )

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
    """Test-only fault injection: makes the next op's future raise."""

    def __init__(self, pg: ProcessGroup) -> None:
        super().__init__(pg=pg)
        self._future_error: Optional[Exception] = None

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
        self._future_error = None
        super().configure(
            store_addr,
            replica_id,
            rank,
            world_size,
            quorum_id,
            group_rank,
            group_world_size,
            global_ranks,
        )

    def report_future_error(self, e: Exception) -> None:
        self._future_error = e

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        work = super().allreduce(tensors, opts)
        if self._future_error is None:
            return work

        future_error, self._future_error = self._future_error, None
        assert future_error is not None

        inner_fut = work.get_future()

        def callback(fut: Future[List[torch.Tensor]]) -> List[torch.Tensor]:
            raise future_error

        errored_fut = inner_fut.then(callback)

        class _FakeErrorWork(Work):
            def __init__(self) -> None:
                super().__init__()

            def wait(self, timeout: Optional[timedelta] = None) -> bool:
                work.wait()
                raise future_error

            def get_future(self) -> Future[object]:
                return errored_fut

        return _FakeErrorWork()


class ManagedProcessGroup(ProcessGroupWrapper):
    """Adapts a Manager into a PG so stock torch DDP/FSDP can use the
    fault-tolerant allreduce; ``size()`` reports quorum participants."""

    def __init__(self, manager: "Manager") -> None:
        super().__init__(pg=manager._pg)
        self._manager = manager

    def allreduce(self, tensors: List[torch.Tensor], opts: object) -> Work:
        assert len(tensors) == 1
        if isinstance(opts, ReduceOp):
            return self._manager.allreduce(tensors[0], reduce_op=opts)
        if isinstance(opts, AllreduceOptions):
            return self._manager.allreduce(tensors[0], reduce_op=opts.reduceOp)
        raise AssertionError("unreachable")

    def size(self) -> int:
        return self._manager.num_participants()

    def getBackendName(self) -> str:
        return self._manager._pg.getBackendName()
