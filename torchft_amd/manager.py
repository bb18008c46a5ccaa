"""Per-rank fault-tolerant training control plane for MI355X.

The ``Manager`` drives the per-step protocol: (async) quorum formation
against the C++ lighthouse/manager services, RCCL communicator
reconfiguration on membership change, live healing of joined/behind replicas
over a checkpoint transport on a dedicated HIP recovery stream, error
tracking, and the all-ranks ``should_commit`` barrier that gates every
optimizer step.

Reference parity (semantics): torchft/manager.py — ctor (:162-364),
``allreduce`` (:410-493), ``report_error/errored`` (:495-514),
``wrap_future`` (:516-558), ``start_quorum``/``_async_quorum`` (:560-813),
``should_commit`` (:855-943), state dict registry (:380-399),
``_ManagedWork``/``_ManagedFuture`` lazy callback chain (:1080-1363).
"""

from __future__ import annotations

import concurrent.futures
import logging
import os
import socket
import traceback
import uuid
import weakref
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta
from enum import Enum
from typing import TYPE_CHECKING, Callable, Dict, Optional, TypeVar, cast

import torch
import torch.distributed as dist
from torch.distributed import ReduceOp, TCPStore
from torch.distributed.distributed_c10d import AllreduceOptions, Work

from torchft_amd._ftcore import ManagerClient, ManagerServer
from torchft_amd.checkpointing import CheckpointTransport, HTTPTransport
from torchft_amd.checkpointing._rwlock import RWLock
from torchft_amd.futures import future_timeout
from torchft_amd.utils import get_stream_context, synchronize
from torchft_amd.work import _DummyWork

if TYPE_CHECKING:
    from torchft_amd.process_group import ProcessGroup

T = TypeVar("T")
S = TypeVar("S")

logger: logging.Logger = logging.getLogger(__name__)

MANAGER_ADDR_KEY: str = "manager_addr"
REPLICA_ID_KEY: str = "replica_id"
MANAGER_PORT_ENV: str = "TORCHFT_MANAGER_PORT"
LIGHTHOUSE_ADDR_ENV: str = "TORCHFT_LIGHTHOUSE"
TIMEOUT_SEC_ENV: str = "TORCHFT_TIMEOUT_SEC"
QUORUM_TIMEOUT_SEC_ENV: str = "TORCHFT_QUORUM_TIMEOUT_SEC"
CONNECT_TIMEOUT_SEC_ENV: str = "TORCHFT_CONNECT_TIMEOUT_SEC"
QUORUM_RETRIES_ENV: str = "TORCHFT_QUORUM_RETRIES"


def get_timeout(timeout_sec_env: Optional[str], default: timedelta) -> timedelta:
    if timeout_sec_env is not None:
        return timedelta(seconds=int(timeout_sec_env))
    return default


def extract_trailing_digits(s: str) -> int:
    """Extract trailing digits of a string, e.g. ``"replica_5"`` → 5."""
    i = len(s)
    while i > 0 and s[i - 1].isdigit():
        i -= 1
    return int(s[i:]) if i < len(s) else 0


class WorldSizeMode(Enum):
    """How the world size is handled when replicas join/leave.

    DYNAMIC: the world size may change between steps; batch size (and thus
        learning dynamics) varies with membership.
    FIXED_WITH_SPARES: at most ``min_replica_size`` replicas participate
        numerically; extras run but their gradients are discarded.
    """

    DYNAMIC = 0
    FIXED_WITH_SPARES = 1


class ExceptionWithTraceback(Exception):
    def __init__(self, e: Exception) -> None:
        self.original_exception = e
        self.stack_trace: str = traceback.format_exc()
        super().__init__(f"{e}\n{self.stack_trace}")


class Manager:
    """Fault-tolerant training loop manager (one per rank).

    Requires the replica group's TCPStore (MASTER_ADDR/PORT or
    store_addr/store_port) to already exist — torchrun provides it.

    NOTE: when saving periodic checkpoints you must save/restore the
    Manager's ``state_dict`` as well to avoid synchronization issues.
    """

    def __init__(
        self,
        pg: "ProcessGroup",
        load_state_dict: Optional[Callable[[T], None]],
        state_dict: Optional[Callable[[], T]],
        min_replica_size: int,
        use_async_quorum: bool = True,
        timeout: timedelta = timedelta(seconds=60),
        quorum_timeout: timedelta = timedelta(seconds=60),
        connect_timeout: timedelta = timedelta(seconds=60),
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
        world_size_mode: WorldSizeMode = WorldSizeMode.DYNAMIC,
        store_addr: Optional[str] = None,
        store_port: Optional[int] = None,
        lighthouse_addr: Optional[str] = None,
        replica_id: Optional[str] = None,
        port: Optional[int] = None,
        hostname: Optional[str] = None,
        heartbeat_interval: timedelta = timedelta(milliseconds=100),
        checkpoint_transport: Optional[CheckpointTransport[Dict[str, T]]] = None,
        init_sync: bool = True,
        max_retries: Optional[int] = None,
        quorum_retries: int = 0,
        should_quantize: bool = False,
    ) -> None:
        self.quorum_logger: logging.Logger = logging.getLogger("torchft_quorums")
        self.commits_logger: logging.Logger = logging.getLogger("torchft_commits")
        self.errors_logger: logging.Logger = logging.getLogger("torchft_errors")

        self._load_state_dict_fns: Dict[str, Callable[[object], None]] = {}
        self._user_state_dicts: Dict[str, Callable[[], object]] = {}

        self._replica_id = replica_id

        # Guards state-dict reads (checkpoint serving) vs training mutation.
        self._state_dict_lock = RWLock(timeout=timeout.total_seconds())

        if load_state_dict and state_dict:
            self.register_state_dict_fn("default", load_state_dict, state_dict)

        self._pending_state_dict: Optional[Dict[str, object]] = None
        self._use_async_quorum = use_async_quorum

        self._timeout = get_timeout(os.environ.get(TIMEOUT_SEC_ENV), timeout)
        self._quorum_timeout = get_timeout(
            os.environ.get(QUORUM_TIMEOUT_SEC_ENV), quorum_timeout
        )
        self._connect_timeout = get_timeout(
            os.environ.get(CONNECT_TIMEOUT_SEC_ENV), connect_timeout
        )

        self._default_should_quantize = should_quantize
        self._replica_world_size_mode = world_size_mode
        self._init_sync = init_sync
        self._max_retries = max_retries
        self._commit_failures = 0
        self._quorum_retries: int = int(
            os.environ.get(QUORUM_RETRIES_ENV, str(quorum_retries))
        )

        store_addr = store_addr or os.environ["MASTER_ADDR"]
        store_port = store_port or int(os.environ["MASTER_PORT"])
        self._group_rank: int = rank if rank is not None else int(os.environ["RANK"])
        group_rank = self._group_rank
        self._group_world_size: int = world_size or int(os.environ["WORLD_SIZE"])
        self._min_replica_size = min_replica_size

        if checkpoint_transport is None:
            checkpoint_transport = HTTPTransport[Dict[str, T]](
                timeout=timeout, num_chunks=0
            )
        self._checkpoint_transport: CheckpointTransport[Dict[str, T]] = (
            checkpoint_transport
        )

        self._executor = ThreadPoolExecutor(
            max_workers=1, thread_name_prefix="async_quorum"
        )
        self._quorum_future: Optional[concurrent.futures.Future] = None

        self._store = TCPStore(
            host_name=store_addr,
            port=store_port,
            is_master=False,
            wait_for_workers=False,
        )
        self._pg = pg
        self._manager: Optional[ManagerServer] = None

        # Healing runs on its own HIP stream so checkpoint send/recv overlap
        # with compute; should_commit syncs on the recorded recovery event.
        self._recovery_stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if torch.cuda.is_available() else None
        )
        self._recovery_event: Optional[torch.cuda.Event] = None

        if hostname is None:
            hostname = socket.gethostname()
            try:
                socket.getaddrinfo(hostname, None)
            except socket.gaierror:
                hostname = "127.0.0.1"

        if self._group_rank == 0:
            if port is None:
                port = int(os.environ.get(MANAGER_PORT_ENV, 0))
            bind = f"0.0.0.0:{port}"
            lighthouse_addr = lighthouse_addr or os.environ[LIGHTHOUSE_ADDR_ENV]

            # Unique suffix so a fast restart of the same replica name is a
            # distinct member at the lighthouse.
            new_uuid = str(uuid.uuid4())
            replica_id = new_uuid if not replica_id else f"{replica_id}:{new_uuid}"
            self._manager = ManagerServer(
                replica_id=replica_id,
                lighthouse_addr=lighthouse_addr,
                hostname=hostname,
                bind=bind,
                store_addr=f"{store_addr}:{store_port}",
                world_size=self._group_world_size,
                heartbeat_interval=heartbeat_interval,
                connect_timeout=connect_timeout,
                quorum_retries=self._quorum_retries,
            )
            self._store.set(MANAGER_ADDR_KEY, self._manager.address())
            self._store.set(REPLICA_ID_KEY, replica_id)

        addr = self._store.get(MANAGER_ADDR_KEY).decode("utf-8")
        self._client = ManagerClient(addr, connect_timeout=connect_timeout)

        replica_id = self._store.get(REPLICA_ID_KEY).decode("utf-8")
        self._logger = _ManagerLogger(
            manager=self, replica_id=replica_id or "", group_rank=group_rank
        )

        self._step = 0
        self._quorum_id = -1
        self._errored: Optional[ExceptionWithTraceback] = None
        self._healing = False
        self._batches_committed = 0

        self._participating_replica_rank: Optional[int] = None
        self._participating_replica_world_size: int = 0
        self._is_state_dict_read_allowed = True

        self._global_rank: int = (
            self._group_rank
            if self._replica_id is None
            else (
                extract_trailing_digits(self._replica_id) * self._group_world_size
                + self._group_rank
            )
        )

    # -- state-dict registry ------------------------------------------------

    def allow_state_dict_read(self) -> None:
        if self._is_state_dict_read_allowed:
            return
        self._is_state_dict_read_allowed = True
        self._state_dict_lock.w_release()

    def disallow_state_dict_read(self) -> None:
        if not self._is_state_dict_read_allowed:
            return
        self._is_state_dict_read_allowed = False
        self._state_dict_lock.w_acquire()

    def register_state_dict_fn(
        self,
        key: str,
        load_state_dict: Callable[[T], None],
        state_dict: Callable[[], T],
    ) -> None:
        assert key not in self._load_state_dict_fns
        assert key not in self._user_state_dicts
        self._load_state_dict_fns[key] = cast(Callable[[object], None], load_state_dict)
        self._user_state_dicts[key] = state_dict

    def set_state_dict_fns(
        self, load_state_dict: Callable[[T], None], state_dict: Callable[[], T]
    ) -> None:
        self._logger.warn(
            "`set_state_dict_fns` is deprecated, use `register_state_dict_fn`"
        )
        self.register_state_dict_fn("set_state_dict_fns", load_state_dict, state_dict)

    def shutdown(self, wait: bool = True) -> None:
        self._checkpoint_transport.shutdown(wait=wait)
        if self._manager is not None:
            self._manager.shutdown()
        self._executor.shutdown(wait=wait)

    # -- the hot path --------------------------------------------------------

    @torch.profiler.record_function("torchft_amd::manager::allreduce")
    def allreduce(
        self,
        tensor: torch.Tensor,
        should_quantize: Optional[bool] = None,
        reduce_op: ReduceOp = ReduceOp.AVG,
    ) -> Work:
        """Fault-tolerant allreduce; AVG scales by 1/num_participants.

        On error the returned work completes successfully with the tensor
        untouched (and possibly corrupted — zero before reuse); the error is
        tracked and surfaces at ``should_commit``.

        ``should_quantize=True`` routes through the fp8 quantized allreduce
        (CDNA4 HIP kernels + alltoall/allgather over all 7 xGMI links).
        """
        if should_quantize is None:
            should_quantize = self._default_should_quantize

        if self.errored():
            return _DummyWork(tensor)

        self.wait_quorum()
        num_participants: int = self.num_participants()

        if not self.is_participating():
            tensor.zero_()

        pg_reduce_op = reduce_op
        if reduce_op == ReduceOp.AVG:
            if not torch.is_floating_point(tensor):
                raise ValueError(
                    "average reduce op is only supported for floating point tensors"
                )
            pg_reduce_op = ReduceOp.SUM

        try:
            if should_quantize and torch.cuda.is_available():
                from torchft_amd.collectives import allreduce_quantized

                work = allreduce_quantized(
                    [tensor], pg_reduce_op, self._pg, torch.cuda.current_stream()
                )
            else:
                opts = AllreduceOptions()
                opts.reduceOp = pg_reduce_op
                work = self._pg.allreduce([tensor], opts)

            # grad normalization as a continuation on the future chain
            def callback(fut: torch.futures.Future[torch.Tensor]) -> torch.Tensor:
                nonlocal tensor
                if reduce_op == ReduceOp.AVG:
                    tensor /= num_participants
                return tensor

            managed_work = _ManagedWork(self, work, tensor)
            fut = cast(torch.futures.Future[torch.Tensor], managed_work.get_future())
            fut = fut.then(callback)
            return managed_work
        except Exception as e:  # noqa: BLE001
            self._logger.exception(f"got exception in all reduce -- skipping remaining: {e}")
            self.report_error(e)
            return _DummyWork(tensor)

    def report_error(self, e: Exception) -> None:
        """Mark this step as failed (gradients must be discarded)."""
        self._errored = ExceptionWithTraceback(e)

    def errored(self) -> Optional[ExceptionWithTraceback]:
        return self._errored

    def wrap_future(
        self,
        fut: torch.futures.Future[T],
        default: T,
        timeout: Optional[timedelta] = None,
    ) -> torch.futures.Future[T]:
        """Swallow errors on ``fut``, report them, and complete with
        ``default`` instead; also applies a timeout."""
        fut = future_timeout(fut, timeout or self._timeout)

        stream: Optional[torch.cuda.Stream] = (
            torch.cuda.current_stream() if torch.cuda.is_available() else None
        )

        def callback(fut: torch.futures.Future[T]) -> T:
            nonlocal default, stream
            with get_stream_context(stream):
                try:
                    return fut.value()
                except Exception as e:  # noqa: BLE001
                    self._logger.exception(
                        f"got exception in future -- skipping remaining: {e}"
                    )
                    self.report_error(e)
                    return default

        return fut.then(callback)

    def start_quorum(
        self,
        allow_heal: bool = True,
        shrink_only: bool = False,
        timeout: Optional[timedelta] = None,
    ) -> None:
        """Compute a new quorum (async by default) and ready the manager for
        a new step. Call before the forward pass; the quorum overlaps with it."""
        # wait for a previous quorum to complete
        if self._quorum_future is not None:
            self._quorum_future.result()

        self._errored = None
        self._healing = False

        self._quorum_future = self._executor.submit(
            self._async_quorum,
            allow_heal=allow_heal,
            shrink_only=shrink_only,
            quorum_timeout=timeout or self._quorum_timeout,
            curr_device=(
                torch.cuda.current_device() if torch.cuda.is_available() else -1
            ),
        )
        if not self._use_async_quorum:
            self.wait_quorum()
            if self._healing:
                # eagerly apply so the forward pass runs on healed weights
                self._apply_pending_state_dict()
                self._healing = False

    @torch.profiler.record_function("torchft_amd::manager::wait_quorum")
    def wait_quorum(self) -> None:
        assert self._quorum_future is not None, "must call start_quorum before wait_quorum"
        self._quorum_future.result()

    @torch.profiler.record_function("torchft_amd::manager::_async_quorum")
    def _async_quorum(
        self,
        allow_heal: bool,
        shrink_only: bool,
        quorum_timeout: timedelta,
        curr_device: int,
    ) -> None:
        if curr_device >= 0 and torch.cuda.is_available():
            torch.cuda.set_device(curr_device)

        quorum = self._client._quorum(
            group_rank=self._group_rank,
            step=self._step,
            checkpoint_metadata=self._checkpoint_transport.metadata(),
            shrink_only=shrink_only,
            timeout=quorum_timeout,
            init_sync=self._init_sync,
            commit_failures=self._commit_failures,
        )

        quorum_id = quorum.quorum_id
        replica_rank = quorum.replica_rank
        replica_world_size = quorum.replica_world_size
        recover_src_manager_address = quorum.recover_src_manager_address
        store_address = quorum.store_address
        max_step = quorum.max_step
        max_replica_rank = quorum.max_replica_rank
        max_replica_world_size = quorum.max_world_size
        heal = quorum.heal
        replica_ids = quorum.replica_ids

        ranks_in_quorum = [
            extract_trailing_digits(rid.split(":")[0]) * self._group_world_size
            + self._group_rank
            for rid in replica_ids
        ]

        # Async quorum: only the already-up-to-date replicas participate this
        # step (healing ones catch up); sync quorum: everyone.
        self._participating_replica_rank, self._participating_replica_world_size = (
            (max_replica_rank, max_replica_world_size)
            if self._use_async_quorum or not allow_heal
            else (replica_rank, replica_world_size)
        )

        if self._replica_world_size_mode == WorldSizeMode.FIXED_WITH_SPARES:
            self._participating_replica_world_size = min(
                self._participating_replica_world_size, self._min_replica_size
            )
            if (
                self._participating_replica_rank is not None
                and self._participating_replica_rank >= self._min_replica_size
            ):
                self._participating_replica_rank = None

        if quorum_id != self._quorum_id:
            self.quorum_logger.info(
                "",
                extra={
                    "job_id": os.environ.get("JOB_ID", "unknown"),
                    "replica_id": self._replica_id,
                    "rank": self._group_rank,
                    "quorum_id": quorum_id,
                    "step": max_step,
                },
            )
            store_prefixed_addr = f"{store_address}/torchft/{quorum_id}/{self._group_rank}"
            self._logger.info(f"reconfiguring for {quorum_id=} {store_prefixed_addr=}")
            try:
                self._quorum_id = quorum_id
                # RCCL comm abort + re-init; must not race in-flight work.
                if torch.cuda.is_available():
                    torch.cuda.synchronize()
                with torch.profiler.record_function(
                    "torchft_amd::manager::pg::configure"
                ):
                    self._pg.configure(
                        store_prefixed_addr,
                        self._replica_id if self._replica_id is not None else "0",
                        replica_rank,
                        replica_world_size,
                        quorum_id,
                        self._group_rank,
                        self._group_world_size,
                        ranks_in_quorum,
                    )
            except Exception as e:  # noqa: BLE001
                self._logger.exception(f"got exception in pg configure: {e}")
                self.report_error(e)
                return

        if allow_heal:
            # Recovery runs on the dedicated HIP recovery stream so the
            # checkpoint copies overlap with compute.
            with get_stream_context(self._recovery_stream):
                try:
                    if quorum.recover_dst_replica_ranks:
                        self._logger.info(
                            f"peers need recovery from us {quorum.recover_dst_replica_ranks}"
                        )
                        with torch.profiler.record_function(
                            "torchft_amd::manager::send_checkpoint"
                        ):
                            self._checkpoint_transport.send_checkpoint(
                                dst_ranks=quorum.recover_dst_replica_ranks,
                                step=max_step,
                                state_dict=self._manager_state_dict(),
                                timeout=self._timeout,
                            )

                    if heal:
                        self._healing = True
                        self._logger.info(
                            f"healing required, fetching checkpoint metadata from "
                            f"{recover_src_manager_address=} {max_step=}"
                        )
                        primary_client = ManagerClient(
                            recover_src_manager_address,
                            connect_timeout=self._connect_timeout,
                        )
                        checkpoint_metadata = primary_client._checkpoint_metadata(
                            self._group_rank, timeout=self._timeout
                        )
                        recover_src_replica_rank = quorum.recover_src_replica_rank
                        assert recover_src_replica_rank is not None, (
                            "must have a recover rank when healing"
                        )
                        self._logger.info(
                            f"fetching checkpoint from {recover_src_replica_rank=} "
                            f"with {checkpoint_metadata=}"
                        )
                        # stage the user state dict; applied from the main
                        # thread only (at should_commit or sync start_quorum)
                        self._pending_state_dict = self._checkpoint_transport.recv_checkpoint(
                            src_rank=recover_src_replica_rank,
                            metadata=checkpoint_metadata,
                            step=max_step,
                            timeout=self._timeout,
                        )
                        self.load_state_dict(self._pending_state_dict["torchft"])
                        self._step = max_step
                except Exception as e:  # noqa: BLE001
                    self._logger.exception(f"got exception in recovery: {e}")
                    self.report_error(e)

                self._recovery_event = (
                    torch.cuda.current_stream().record_event()
                    if self._recovery_stream is not None
                    else None
                )

    def _apply_pending_state_dict(self) -> None:
        assert self._healing, "must be in healing state"
        assert self._quorum_future is not None, "must call step before should_commit"
        self._quorum_future.result()

        pending_state_dict = self._pending_state_dict
        if pending_state_dict is None:
            assert self.errored(), "checkpoint was not staged and no error occured"
            return

        self._logger.info("applying pending state dict")
        assert len(self._load_state_dict_fns) > 0, "user load_state_dict is not initialized."
        pending_user_state_dict = cast(Dict[str, object], pending_state_dict["user"])
        for key, load_fn in self._load_state_dict_fns.items():
            load_fn(pending_user_state_dict[key])
        self._pending_state_dict = None
        self._logger.info("Loaded state dict.")

    @torch.profiler.record_function("torchft_amd::manager::should_commit")
    def should_commit(self, timeout: Optional[timedelta] = None) -> bool:
        """All-ranks barrier deciding whether to step the optimizer.

        Must be called after backward and before optimizer.step(); the
        optimizer may only step when this returns True.
        """
        # recovery must be complete before committing
        if self._recovery_event is not None:
            self._recovery_event.synchronize()
            self._recovery_event = None

        if torch.cuda.is_available():
            synchronize()

        if err := self._pg.errored():
            self.report_error(err)

        if self._healing:
            self._apply_pending_state_dict()

        enough_replicas = self.num_participants() >= self._min_replica_size
        local_should_commit = enough_replicas and self._errored is None
        should_commit = self._client.should_commit(
            self._group_rank,
            self._step,
            local_should_commit,
            timeout=timeout or self._timeout,
        )
        self._logger.info(
            f"should_commit={should_commit} enough_replicas={enough_replicas}, "
            f"errored={self._errored}"
        )

        self.commits_logger.info(
            "",
            extra={
                "job_id": os.environ.get("JOB_ID", "unknown"),
                "replica_id": self._replica_id,
                "rank": self._group_rank,
                "quorum_id": self._quorum_id,
                "step": self._step,
                "commit_result": should_commit,
            },
        )

        self._checkpoint_transport.disallow_checkpoint()

        if should_commit:
            self._step += 1
            self._batches_committed += self.num_participants()
            self._commit_failures = 0
        else:
            self._commit_failures += 1
            if self._max_retries is not None and self._commit_failures > self._max_retries:
                msg = (
                    f"should_commit failed {self._commit_failures} times consecutively, "
                    f"exceeding max_retries={self._max_retries}"
                )
                self._logger.exception(msg)
                raise RuntimeError(msg)

        return should_commit

    # -- state ----------------------------------------------------------------

    def load_state_dict(self, state_dict: Dict[str, int]) -> None:
        self._step = state_dict["step"]
        self._batches_committed = state_dict["batches_committed"]

    def _manager_state_dict(self) -> Dict[str, object]:
        with self._state_dict_lock.r_lock():
            assert len(self._user_state_dicts) > 0, "user state_dict is not initialized."
            return {
                "user": {key: fn() for key, fn in self._user_state_dicts.items()},
                "torchft": self.state_dict(),
            }

    def state_dict(self) -> Dict[str, int]:
        return {"step": self._step, "batches_committed": self._batches_committed}

    def current_step(self) -> int:
        return self._step

    def batches_committed(self) -> int:
        return self._batches_committed

    def participating_rank(self) -> Optional[int]:
        if self._quorum_future is None:
            return None
        self.wait_quorum()
        return self._participating_replica_rank

    def num_participants(self) -> int:
        if self._quorum_future is None:
            return 0
        self.wait_quorum()
        assert self._participating_replica_world_size >= 0, "internal error"
        return self._participating_replica_world_size

    def is_participating(self) -> bool:
        if self._participating_replica_rank is None:
            return False
        if self._healing:
            assert self._use_async_quorum
            return False
        return True


class _ManagerLogger:
    def __init__(self, manager: Manager, replica_id: str, group_rank: int) -> None:
        self._logger: logging.Logger = logging.getLogger(__name__)
        self._replica_id = replica_id
        self._group_rank = group_rank
        self._manager = manager

    def prefix(self) -> str:
        return f"[{self._replica_id}/{self._group_rank} - step {self._manager.current_step()}]"

    def info(self, msg: str) -> None:
        self._logger.info(f"{self.prefix()} {msg}")

    def warn(self, msg: str) -> None:
        self._logger.warning(f"{self.prefix()} {msg}")

    def exception(self, msg: str) -> None:
        self._logger.exception(f"{self.prefix()} {msg}")


class _SimpleFuture(torch.futures.Future[T]):
    """Wraps a pre-determined value for use in the _ManagedFuture callback
    chain without blocking the CPU on ``value()`` of a real future."""

    def __init__(self, value: object) -> None:
        super().__init__()
        self._value = value

    def value(self) -> object:
        return self._value

    def then(self, callback: Callable) -> torch.futures.Future:
        raise NotImplementedError("callback-chain value wrapper only")

    def wait(self) -> object:
        raise NotImplementedError("callback-chain value wrapper only")

    def done(self) -> bool:
        raise NotImplementedError("callback-chain value wrapper only")

    def add_done_callback(self, callback: Callable) -> None:
        raise NotImplementedError("callback-chain value wrapper only")

    def set_result(self, result: object) -> None:
        raise NotImplementedError("callback-chain value wrapper only")

    def set_exception(self, result: object) -> None:
        raise NotImplementedError("callback-chain value wrapper only")


class _ManagedFuture(torch.futures.Future[T]):
    """Lazy future chaining bound to the issuing HIP stream.

    ``then()`` only records the callback; the chain is materialized on
    ``wait()/synchronize()/block_current_stream()`` of the owning
    ``_ManagedWork``, inside the original stream context, with the first
    callback wrapped by ``manager.wrap_future`` for error swallowing.
    """

    def __init__(self, managed_work: "weakref.ReferenceType[_ManagedWork]") -> None:
        super().__init__()
        self._managed_work = managed_work
        self._fut: Optional[torch.futures.Future[T]] = None
        self._next: Optional["_ManagedFuture[object]"] = None
        self._callback: Optional[Callable[[torch.futures.Future[T]], object]] = None

    def then(
        self, callback: Callable[[torch.futures.Future[T]], S]
    ) -> torch.futures.Future[S]:
        managed_work = self._managed_work()
        assert managed_work is not None, "got garbage collected"
        self._callback = callback
        self._next = _ManagedFuture[object](self._managed_work)
        managed_work._managed_fut_tail = self._next
        return cast(torch.futures.Future[S], self._next)

    def wait(self) -> object:
        assert self._fut
        return self._fut.wait()

    def value(self) -> object:
        raise NotImplementedError("used to create callback chains only")

    def done(self) -> bool:
        raise NotImplementedError("used to create callback chains only")

    def add_done_callback(self, callback: Callable) -> None:
        raise NotImplementedError("used to create callback chains only")

    def set_result(self, result: object) -> None:
        raise NotImplementedError("used to create callback chains only")

    def set_exception(self, result: object) -> None:
        raise NotImplementedError("used to create callback chains only")


class _ManagedWork(dist._Work):
    """Work whose future-callback chain is created lazily on wait/synchronize,
    always inside the stream the collective was issued on."""

    def __init__(self, manager: Manager, work: dist._Work, value: object) -> None:
        super().__init__()
        self._work = work
        self._manager = manager
        self._value = value
        self._managed_fut_head = _ManagedFuture[object](weakref.ref(self))
        self._managed_fut_tail: _ManagedFuture[object] = self._managed_fut_head
        self._stream: Optional[torch.cuda.Stream] = (
            torch.cuda.current_stream() if torch.cuda.is_available() else None
        )
        self._is_set_future_callback_called = False

    def _set_future_callback(self) -> None:
        if self._is_set_future_callback_called:
            return

        managed_fut: _ManagedFuture[object] = self._managed_fut_head
        managed_fut._fut = self._work.get_future()
        value = self._value

        is_future_wrapped = False
        while managed_fut._next:

            def callback(fut: torch.futures.Future[object]) -> object:
                nonlocal managed_fut, value
                # keep the chain on the issuing stream, not the PG stream
                with get_stream_context(self._stream):
                    fut.wait()  # stream dependency
                    assert managed_fut._callback
                    value = managed_fut._callback(_SimpleFuture(value))
                    return value

            assert managed_fut._fut
            fut = managed_fut._fut.then(callback)
            assert managed_fut._next
            managed_fut = managed_fut._next
            managed_fut._fut = fut

            if is_future_wrapped:
                continue
            managed_fut._fut = self._manager.wrap_future(managed_fut._fut, value)
            is_future_wrapped = True

        self._value = value
        self._is_set_future_callback_called = True

    def _assert_same_stream(self) -> None:
        if self._stream is not None:
            assert self._stream == torch.cuda.current_stream()

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        self._assert_same_stream()
        try:
            with get_stream_context(self._stream):
                self._work.wait()
                self._set_future_callback()
            with get_stream_context(self._stream):
                self._managed_fut_tail.wait()
            return True
        except Exception as e:  # noqa: BLE001
            self._manager._logger.exception(f"got exception waiting for work {e}")
            self._manager.report_error(e)
            return False

    def block_current_stream(self, timeout: Optional[timedelta] = None) -> None:
        self._assert_same_stream()
        with get_stream_context(self._stream):
            self._work.block_current_stream()
        self._set_future_callback()

    def synchronize(self) -> None:
        self._assert_same_stream()
        if torch.cuda.is_available():
            self.block_current_stream()
        else:
            self._set_future_callback()

    def get_future(self) -> torch.futures.Future[object]:
        return self._managed_fut_tail
