"""Per-rank fault-tolerant training control plane for MI355X.

The ``Manager`` runs an explicit per-step protocol:

    start_quorum()  -> negotiate      (quorum RPC against the C++ services,
                                       on a background worker thread)
                    -> participate    (pure function: quorum -> who trains)
                    -> rebuild comm   (RCCL abort + re-init when membership
                                       changed; per-quorum store prefix)
                    -> recover        (serve or fetch a live checkpoint on a
                                       dedicated HIP recovery stream)
    allreduce()     -> fault-tolerant gradient averaging, continuations run
                       on the issuing HIP stream (``_ManagedWork``)
    should_commit() -> drain recovery + compute streams, apply staged heal,
                       all-ranks barrier; only then may the optimizer step.

Architecture notes (deliberately different from the reference,
torchft/manager.py, whose semantics we match — ctor :162-364, allreduce
:410-493, quorum :560-813, should_commit :855-943, managed work :1080-1363):

* each ``start_quorum`` produces a ``_StepAttempt`` record; participation is
  computed by the pure function ``_participation_from`` (unit-testable
  without any I/O);
* the quorum cycle is a pipeline of small named methods instead of one
  monolithic thread function;
* ``_ManagedWork`` keeps a flat list of continuations and materializes them
  with a single real ``Future.then`` at wait time, bound to the issuing
  HIP stream — not a linked list of lazy futures.
"""

from __future__ import annotations

import logging
import os
import queue
import socket
import threading
import traceback
import uuid
from dataclasses import dataclass
from datetime import timedelta
from enum import Enum
from typing import TYPE_CHECKING, Callable, Dict, List, Optional, Tuple, TypeVar, cast

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
    """Env-var seconds override for a timedelta default."""
    return timedelta(seconds=int(timeout_sec_env)) if timeout_sec_env else default


def extract_trailing_digits(s: str) -> int:
    """``"replica_57"`` -> 57; no trailing digits -> 0."""
    digits = ""
    for ch in reversed(s):
        if not ch.isdigit():
            break
        digits = ch + digits
    return int(digits) if digits else 0


class WorldSizeMode(Enum):
    """Numeric handling of a changing replica count.

    DYNAMIC: effective batch size follows membership.
    FIXED_WITH_SPARES: at most ``min_replica_size`` replicas contribute
        gradients; the rest train but are numerically inert spares.
    """

    DYNAMIC = 0
    FIXED_WITH_SPARES = 1


class ExceptionWithTraceback(Exception):
    def __init__(self, e: Exception) -> None:
        self.original_exception = e
        self.stack_trace: str = traceback.format_exc()
        super().__init__(f"{e}\n{self.stack_trace}")


# ---------------------------------------------------------------------------
# background worker
# ---------------------------------------------------------------------------


class _Pending:
    """Completion handle for a task submitted to ``_SerialExecutor``."""

    def __init__(self) -> None:
        self._done = threading.Event()
        self._exc: Optional[BaseException] = None

    def _finish(self, exc: Optional[BaseException]) -> None:
        self._exc = exc
        self._done.set()

    def result(self, timeout: Optional[float] = None) -> None:
        if not self._done.wait(timeout):
            raise TimeoutError("quorum task did not complete in time")
        if self._exc is not None:
            raise self._exc


class _SerialExecutor:
    """One daemon thread executing submitted thunks strictly in order.

    The quorum negotiation must never run concurrently with itself and must
    not block the training thread; a dedicated serial worker makes both
    properties structural instead of incidental.
    """

    def __init__(self, name: str) -> None:
        self._q: "queue.Queue[Optional[Tuple[Callable[[], None], _Pending]]]" = (
            queue.Queue()
        )
        self._thread = threading.Thread(target=self._run, name=name, daemon=True)
        self._started = False
        self._start_mu = threading.Lock()

    def submit(self, fn: Callable[[], None]) -> _Pending:
        with self._start_mu:
            if not self._started:
                self._thread.start()
                self._started = True
        pending = _Pending()
        self._q.put((fn, pending))
        return pending

    def _run(self) -> None:
        while True:
            item = self._q.get()
            if item is None:
                return
            fn, pending = item
            try:
                fn()
                pending._finish(None)
            except BaseException as e:  # noqa: BLE001
                pending._finish(e)

    def shutdown(self, wait: bool = True) -> None:
        if not self._started:
            return
        self._q.put(None)
        if wait:
            self._thread.join(timeout=60)


# ---------------------------------------------------------------------------
# per-step state
# ---------------------------------------------------------------------------


@dataclass
class _StepAttempt:
    """Everything the current step learned from its quorum."""

    pending: Optional[_Pending] = None
    # filled in by the quorum pipeline:
    participant_rank: Optional[int] = None
    participant_count: int = 0
    healing: bool = False
    staged_state: Optional[Dict[str, object]] = None
    recovery_event: Optional[torch.cuda.Event] = None
    error: Optional[ExceptionWithTraceback] = None

    def ready(self) -> None:
        assert self.pending is not None, "start_quorum must be called first"
        self.pending.result()


def _participation_from(
    *,
    replica_rank: int,
    replica_world_size: int,
    max_replica_rank: Optional[int],
    max_world_size: int,
    defer_healing: bool,
    world_size_mode: WorldSizeMode,
    min_replica_size: int,
) -> Tuple[Optional[int], int]:
    """Pure function: quorum result -> (participating rank | None, count).

    With deferred healing (async quorum) only the replicas already at
    max_step train this step; otherwise everyone in the quorum does.
    FIXED_WITH_SPARES then caps the numeric world at ``min_replica_size``
    and benches the overflow ranks.
    """
    if defer_healing:
        rank, count = max_replica_rank, max_world_size
    else:
        rank, count = replica_rank, replica_world_size

    if world_size_mode == WorldSizeMode.FIXED_WITH_SPARES:
        count = min(count, min_replica_size)
        if rank is not None and rank >= min_replica_size:
            rank = None
    return rank, count


# ---------------------------------------------------------------------------
# the Manager
# ---------------------------------------------------------------------------


class Manager:
    """Fault-tolerant training loop manager (one instance per rank).

    Expects the replica group's TCPStore to exist already (torchrun creates
    it; MASTER_ADDR/PORT or store_addr/store_port point at it). When saving
    periodic checkpoints, persist ``Manager.state_dict()`` alongside the
    model so a restore resumes the step/batch counters consistently.
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
        # telemetry channels (structured records; see telemetry.py)
        self.quorum_logger: logging.Logger = logging.getLogger("torchft_quorums")
        self.commits_logger: logging.Logger = logging.getLogger("torchft_commits")
        self.errors_logger: logging.Logger = logging.getLogger("torchft_errors")

        self._pg = pg
        self._min_replica_size = min_replica_size
        self._use_async_quorum = use_async_quorum
        self._world_size_mode = world_size_mode
        self._init_sync = init_sync
        self._max_retries = max_retries
        self._default_should_quantize = should_quantize
        self._replica_id = replica_id

        self._timeout = get_timeout(os.environ.get(TIMEOUT_SEC_ENV), timeout)
        self._quorum_timeout = get_timeout(
            os.environ.get(QUORUM_TIMEOUT_SEC_ENV), quorum_timeout
        )
        self._connect_timeout = get_timeout(
            os.environ.get(CONNECT_TIMEOUT_SEC_ENV), connect_timeout
        )
        self._quorum_retries: int = int(
            os.environ.get(QUORUM_RETRIES_ENV, str(quorum_retries))
        )

        self._group_rank: int = rank if rank is not None else int(os.environ["RANK"])
        self._group_world_size: int = world_size or int(os.environ["WORLD_SIZE"])

        # ---- state-dict registry + serving lock --------------------------
        self._load_state_dict_fns: Dict[str, Callable[[object], None]] = {}
        self._user_state_dicts: Dict[str, Callable[[], object]] = {}
        self._state_dict_lock = RWLock(timeout=timeout.total_seconds())
        self._is_state_dict_read_allowed = True
        if load_state_dict and state_dict:
            self.register_state_dict_fn("default", load_state_dict, state_dict)

        # ---- durable protocol counters ------------------------------------
        self._step = 0
        self._batches_committed = 0
        self._commit_failures = 0
        self._quorum_id = -1

        # ---- per-step state ----------------------------------------------
        self._attempt = _StepAttempt()
        self._attempt.pending = None

        # ---- transports ---------------------------------------------------
        if checkpoint_transport is None:
            checkpoint_transport = HTTPTransport[Dict[str, T]](
                timeout=timeout, num_chunks=0
            )
        self._checkpoint_transport: CheckpointTransport[Dict[str, T]] = (
            checkpoint_transport
        )

        # Healing copies run on their own HIP stream so a multi-GB fetch
        # overlaps compute; should_commit waits on the recorded event.
        self._recovery_stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream() if torch.cuda.is_available() else None
        )

        self._worker = _SerialExecutor("torchft-quorum")

        # ---- coordination services ---------------------------------------
        store_addr = store_addr or os.environ["MASTER_ADDR"]
        store_port = store_port or int(os.environ["MASTER_PORT"])
        self._store = TCPStore(
            host_name=store_addr,
            port=store_port,
            is_master=False,
            wait_for_workers=False,
        )

        if hostname is None:
            hostname = socket.gethostname()
            try:
                socket.getaddrinfo(hostname, None)
            except socket.gaierror:
                hostname = "127.0.0.1"

        self._manager: Optional[ManagerServer] = None
        if self._group_rank == 0:
            # group rank 0 hosts the per-replica aggregation service and
            # publishes its address through the group store
            if port is None:
                port = int(os.environ.get(MANAGER_PORT_ENV, 0))
            effective_id = str(uuid.uuid4())
            if replica_id:
                # unique suffix: a fast restart of the same replica name must
                # register as a distinct member at the lighthouse
                effective_id = f"{replica_id}:{effective_id}"
            self._manager = ManagerServer(
                replica_id=effective_id,
                lighthouse_addr=lighthouse_addr or os.environ[LIGHTHOUSE_ADDR_ENV],
                hostname=hostname,
                bind=f"0.0.0.0:{port}",
                store_addr=f"{store_addr}:{store_port}",
                world_size=self._group_world_size,
                heartbeat_interval=heartbeat_interval,
                connect_timeout=connect_timeout,
                quorum_retries=self._quorum_retries,
            )
            self._store.set(MANAGER_ADDR_KEY, self._manager.address())
            self._store.set(REPLICA_ID_KEY, effective_id)

        manager_addr = self._store.get(MANAGER_ADDR_KEY).decode("utf-8")
        self._client = ManagerClient(manager_addr, connect_timeout=connect_timeout)

        published_id = self._store.get(REPLICA_ID_KEY).decode("utf-8")
        self._logger = _ManagerLogger(
            manager=self, replica_id=published_id or "", group_rank=self._group_rank
        )

        self._global_rank: int = (
            self._group_rank
            if self._replica_id is None
            else extract_trailing_digits(self._replica_id) * self._group_world_size
            + self._group_rank
        )

    # -- state-dict registry --------------------------------------------------

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

    def allow_state_dict_read(self) -> None:
        if not self._is_state_dict_read_allowed:
            self._is_state_dict_read_allowed = True
            self._state_dict_lock.w_release()

    def disallow_state_dict_read(self) -> None:
        if self._is_state_dict_read_allowed:
            self._is_state_dict_read_allowed = False
            self._state_dict_lock.w_acquire()

    def shutdown(self, wait: bool = True) -> None:
        self._checkpoint_transport.shutdown(wait=wait)
        if self._manager is not None:
            self._manager.shutdown()
        self._worker.shutdown(wait=wait)

    # -- hot path: fault-tolerant allreduce -----------------------------------

    @torch.profiler.record_function("torchft_amd::manager::allreduce")
    def allreduce(
        self,
        tensor: torch.Tensor,
        should_quantize: Optional[bool] = None,
        reduce_op: ReduceOp = ReduceOp.AVG,
    ) -> Work:
        """Fault-tolerant allreduce of ``tensor``.

        AVG divides by the live participant count as a continuation on the
        issuing stream. On any error the returned work still "succeeds"
        (tensor contents undefined — zero before reuse); the error is
        remembered and fails the step at ``should_commit``.

        ``should_quantize=True`` takes the fp8 path: CDNA4 quantize kernels
        plus an alltoall/allgather decomposition that engages all 7 xGMI
        links.
        """
        if should_quantize is None:
            should_quantize = self._default_should_quantize

        if self.errored():
            return _DummyWork(tensor)

        self.wait_quorum()
        num_participants = self.num_participants()

        if not self.is_participating():
            tensor.zero_()

        if reduce_op == ReduceOp.AVG and not torch.is_floating_point(tensor):
            raise ValueError(
                "average reduce op is only supported for floating point tensors"
            )
        wire_op = ReduceOp.SUM if reduce_op == ReduceOp.AVG else reduce_op

        try:
            if should_quantize and torch.cuda.is_available():
                from torchft_amd.collectives import allreduce_quantized

                work = allreduce_quantized(
                    [tensor], wire_op, self._pg, torch.cuda.current_stream()
                )
            else:
                opts = AllreduceOptions()
                opts.reduceOp = wire_op
                work = self._pg.allreduce([tensor], opts)
        except Exception as e:  # noqa: BLE001
            self._logger.exception(f"allreduce failed to launch: {e}")
            self.report_error(e)
            return _DummyWork(tensor)

        managed = _ManagedWork(self, work, tensor)
        if reduce_op == ReduceOp.AVG:

            def normalize(value: torch.Tensor) -> torch.Tensor:
                value /= num_participants
                return value

            managed._append(normalize)
        return managed

    def report_error(self, e: Exception) -> None:
        """Mark the in-flight step failed; its gradients must be discarded."""
        self._attempt.error = ExceptionWithTraceback(e)

    def errored(self) -> Optional[ExceptionWithTraceback]:
        return self._attempt.error

    def wrap_future(
        self,
        fut: torch.futures.Future[T],
        default: T,
        timeout: Optional[timedelta] = None,
    ) -> torch.futures.Future[T]:
        """Timeout + error-swallow a future: failures are reported to the
        manager and replaced by ``default``."""
        timed = future_timeout(fut, timeout or self._timeout)
        stream = torch.cuda.current_stream() if torch.cuda.is_available() else None

        def absorb(f: torch.futures.Future[T]) -> T:
            with get_stream_context(stream):
                try:
                    return f.value()
                except Exception as e:  # noqa: BLE001
                    self._logger.exception(f"future failed; using default: {e}")
                    self.report_error(e)
                    return default

        return timed.then(absorb)

    # -- quorum pipeline -------------------------------------------------------

    def start_quorum(
        self,
        allow_heal: bool = True,
        shrink_only: bool = False,
        timeout: Optional[timedelta] = None,
    ) -> None:
        """Begin a new step: negotiate a quorum (async by default, so it
        overlaps the forward pass) and arm the per-step state."""
        if self._attempt.pending is not None:
            # a previous step's negotiation must fully settle first
            self._attempt.pending.result()

        attempt = _StepAttempt()
        self._attempt = attempt
        device = torch.cuda.current_device() if torch.cuda.is_available() else -1
        attempt.pending = self._worker.submit(
            lambda: self._quorum_cycle(
                attempt,
                allow_heal=allow_heal,
                shrink_only=shrink_only,
                quorum_timeout=timeout or self._quorum_timeout,
                device=device,
            )
        )

        if not self._use_async_quorum:
            attempt.pending.result()
            if attempt.healing:
                # apply eagerly so the forward pass sees healed weights
                self._apply_pending_state_dict()
                attempt.healing = False

    @torch.profiler.record_function("torchft_amd::manager::wait_quorum")
    def wait_quorum(self) -> None:
        self._attempt.ready()

    @torch.profiler.record_function("torchft_amd::manager::quorum_cycle")
    def _quorum_cycle(
        self,
        attempt: _StepAttempt,
        allow_heal: bool,
        shrink_only: bool,
        quorum_timeout: timedelta,
        device: int,
    ) -> None:
        """Worker-thread body: negotiate -> participate -> rebuild -> recover."""
        if device >= 0 and torch.cuda.is_available():
            torch.cuda.set_device(device)

        q = self._client._quorum(
            group_rank=self._group_rank,
            step=self._step,
            checkpoint_metadata=self._checkpoint_transport.metadata(),
            shrink_only=shrink_only,
            timeout=quorum_timeout,
            init_sync=self._init_sync,
            commit_failures=self._commit_failures,
        )

        attempt.participant_rank, attempt.participant_count = _participation_from(
            replica_rank=q.replica_rank,
            replica_world_size=q.replica_world_size,
            max_replica_rank=q.max_replica_rank,
            max_world_size=q.max_world_size,
            defer_healing=self._use_async_quorum or not allow_heal,
            world_size_mode=self._world_size_mode,
            min_replica_size=self._min_replica_size,
        )

        if q.quorum_id != self._quorum_id:
            if not self._rebuild_comm(attempt, q):
                return

        if allow_heal:
            self._recovery_phase(attempt, q)

    def _rebuild_comm(self, attempt: _StepAttempt, q: object) -> bool:
        """Membership changed: abort and re-init the RCCL communicator against
        the per-quorum store prefix. Returns False on failure (step aborts)."""
        quorum_id = q.quorum_id
        self.quorum_logger.info(
            "",
            extra={
                "job_id": os.environ.get("JOB_ID", "unknown"),
                "replica_id": self._replica_id,
                "rank": self._group_rank,
                "quorum_id": quorum_id,
                "step": q.max_step,
            },
        )
        store_prefixed_addr = (
            f"{q.store_address}/torchft/{quorum_id}/{self._group_rank}"
        )
        self._logger.info(f"reconfiguring for {quorum_id=} {store_prefixed_addr=}")

        # replica_ids -> global ranks for this rank's position in each group
        global_ranks = [
            extract_trailing_digits(rid.split(":")[0]) * self._group_world_size
            + self._group_rank
            for rid in q.replica_ids
        ]

        try:
            self._quorum_id = quorum_id
            if torch.cuda.is_available():
                # the abort must not race in-flight collectives
                torch.cuda.synchronize()
            with torch.profiler.record_function("torchft_amd::manager::pg::configure"):
                self._pg.configure(
                    store_prefixed_addr,
                    self._replica_id if self._replica_id is not None else "0",
                    q.replica_rank,
                    q.replica_world_size,
                    quorum_id,
                    self._group_rank,
                    self._group_world_size,
                    global_ranks,
                )
            return True
        except Exception as e:  # noqa: BLE001
            self._logger.exception(f"pg configure failed: {e}")
            self.report_error(e)
            return False

    def _recovery_phase(self, attempt: _StepAttempt, q: object) -> None:
        """Serve checkpoints to recovering peers and/or fetch our own heal,
        on the dedicated recovery stream so the copies overlap compute."""
        with get_stream_context(self._recovery_stream):
            try:
                if q.recover_dst_replica_ranks:
                    self._serve_checkpoint(q)
                if q.heal:
                    attempt.healing = True
                    self._fetch_checkpoint(attempt, q)
            except Exception as e:  # noqa: BLE001
                self._logger.exception(f"recovery failed: {e}")
                self.report_error(e)

            attempt.recovery_event = (
                torch.cuda.current_stream().record_event()
                if self._recovery_stream is not None
                else None
            )

    @torch.profiler.record_function("torchft_amd::manager::send_checkpoint")
    def _serve_checkpoint(self, q: object) -> None:
        self._logger.info(f"serving checkpoint to {q.recover_dst_replica_ranks}")
        self._checkpoint_transport.send_checkpoint(
            dst_ranks=q.recover_dst_replica_ranks,
            step=q.max_step,
            state_dict=self._manager_state_dict(),
            timeout=self._timeout,
        )

    def _fetch_checkpoint(self, attempt: _StepAttempt, q: object) -> None:
        """We are behind: pull the live checkpoint from the recovery source
        and stage it; the user portion is applied on the main thread only."""
        src_addr = q.recover_src_manager_address
        self._logger.info(f"healing from {src_addr} at step {q.max_step}")
        src_client = ManagerClient(src_addr, connect_timeout=self._connect_timeout)
        metadata = src_client._checkpoint_metadata(
            self._group_rank, timeout=self._timeout
        )
        src_replica_rank = q.recover_src_replica_rank
        assert src_replica_rank is not None, "healing requires a recovery source"
        self._logger.info(f"fetching checkpoint from {src_replica_rank=} {metadata=}")
        staged = self._checkpoint_transport.recv_checkpoint(
            src_rank=src_replica_rank,
            metadata=metadata,
            step=q.max_step,
            timeout=self._timeout,
        )
        attempt.staged_state = staged
        # manager counters can load immediately; they are plain ints
        self.load_state_dict(cast(Dict[str, int], staged["torchft"]))
        self._step = q.max_step

    def _apply_pending_state_dict(self) -> None:
        """Main-thread application of a staged heal."""
        assert self._attempt.healing, "no heal staged"
        self._attempt.ready()

        staged = self._attempt.staged_state
        if staged is None:
            assert self.errored(), "no staged checkpoint and no error"
            return

        self._logger.info("applying staged state dict")
        assert self._load_state_dict_fns, "no load_state_dict fns registered"
        user_state = cast(Dict[str, object], staged["user"])
        for key, load_fn in self._load_state_dict_fns.items():
            load_fn(user_state[key])
        self._attempt.staged_state = None
        self._logger.info("staged state dict applied")

    # -- commit barrier --------------------------------------------------------

    @torch.profiler.record_function("torchft_amd::manager::should_commit")
    def should_commit(self, timeout: Optional[timedelta] = None) -> bool:
        """All-ranks barrier gating the optimizer step.

        Drains the recovery and compute streams, applies any staged heal,
        then votes: the step commits iff every rank in the group is
        error-free and enough replicas participated.
        """
        attempt = self._attempt

        if attempt.recovery_event is not None:
            attempt.recovery_event.synchronize()
            attempt.recovery_event = None
        if torch.cuda.is_available():
            synchronize()

        if (pg_err := self._pg.errored()) is not None:
            self.report_error(pg_err)

        if attempt.healing:
            self._apply_pending_state_dict()

        enough = self.num_participants() >= self._min_replica_size
        my_vote = enough and attempt.error is None
        decision = self._client.should_commit(
            self._group_rank,
            self._step,
            my_vote,
            timeout=timeout or self._timeout,
        )
        self._logger.info(
            f"should_commit={decision} enough_replicas={enough} errored={attempt.error}"
        )
        self.commits_logger.info(
            "",
            extra={
                "job_id": os.environ.get("JOB_ID", "unknown"),
                "replica_id": self._replica_id,
                "rank": self._group_rank,
                "quorum_id": self._quorum_id,
                "step": self._step,
                "commit_result": decision,
            },
        )

        # no checkpoint may be served across the commit boundary
        self._checkpoint_transport.disallow_checkpoint()

        if decision:
            self._step += 1
            self._batches_committed += self.num_participants()
            self._commit_failures = 0
        else:
            self._commit_failures += 1
            if (
                self._max_retries is not None
                and self._commit_failures > self._max_retries
            ):
                msg = (
                    f"should_commit failed {self._commit_failures} times "
                    f"consecutively, exceeding max_retries={self._max_retries}"
                )
                self._logger.exception(msg)
                raise RuntimeError(msg)
        return decision

    # -- counters / introspection ---------------------------------------------

    def load_state_dict(self, state_dict: Dict[str, int]) -> None:
        self._step = state_dict["step"]
        self._batches_committed = state_dict["batches_committed"]

    def state_dict(self) -> Dict[str, int]:
        return {"step": self._step, "batches_committed": self._batches_committed}

    def _manager_state_dict(self) -> Dict[str, object]:
        with self._state_dict_lock.r_lock():
            assert self._user_state_dicts, "no user state_dict registered"
            return {
                "user": {key: fn() for key, fn in self._user_state_dicts.items()},
                "torchft": self.state_dict(),
            }

    def current_step(self) -> int:
        return self._step

    def batches_committed(self) -> int:
        return self._batches_committed

    def participating_rank(self) -> Optional[int]:
        if self._attempt.pending is None:
            return None
        self.wait_quorum()
        return self._attempt.participant_rank

    def num_participants(self) -> int:
        if self._attempt.pending is None:
            return 0
        self.wait_quorum()
        assert self._attempt.participant_count >= 0
        return self._attempt.participant_count

    def is_participating(self) -> bool:
        if self._attempt.participant_rank is None:
            return False
        if self._attempt.healing:
            assert self._use_async_quorum
            return False
        return True

    # kept as properties so algorithm wrappers / tests can read them
    @property
    def _healing(self) -> bool:
        return self._attempt.healing

    @_healing.setter
    def _healing(self, value: bool) -> None:
        self._attempt.healing = value

    @property
    def _errored(self) -> Optional[ExceptionWithTraceback]:
        return self._attempt.error

    @_errored.setter
    def _errored(self, value: Optional[ExceptionWithTraceback]) -> None:
        self._attempt.error = value

    @property
    def _pending_state_dict(self) -> Optional[Dict[str, object]]:
        return self._attempt.staged_state

    @_pending_state_dict.setter
    def _pending_state_dict(self, value: Optional[Dict[str, object]]) -> None:
        self._attempt.staged_state = value


class _ManagerLogger:
    def __init__(self, manager: Manager, replica_id: str, group_rank: int) -> None:
        self._logger: logging.Logger = logging.getLogger(__name__)
        self._replica_id = replica_id
        self._group_rank = group_rank
        self._manager = manager

    def prefix(self) -> str:
        return (
            f"[{self._replica_id}/{self._group_rank} "
            f"- step {self._manager.current_step()}]"
        )

    def info(self, msg: str) -> None:
        self._logger.info(f"{self.prefix()} {msg}")

    def warn(self, msg: str) -> None:
        self._logger.warning(f"{self.prefix()} {msg}")

    def exception(self, msg: str) -> None:
        self._logger.exception(f"{self.prefix()} {msg}")


# ---------------------------------------------------------------------------
# stream-bound deferred continuations
# ---------------------------------------------------------------------------


class _ValueFuture(torch.futures.Future):
    """Minimal already-completed future handed to user callbacks so they can
    call ``.value()`` without touching a real (possibly device-bound) future."""

    def __init__(self, value: object) -> None:
        super().__init__()
        self._v = value

    def value(self) -> object:
        return self._v

    def wait(self) -> object:
        return self._v


class _ManagedFuture(torch.futures.Future):
    """Future view of a ``_ManagedWork``.

    ``then()`` only records the callback on the owning work's continuation
    ledger — nothing runs until the work is waited on, and everything then
    runs inside the stream the collective was issued on.
    """

    def __init__(self, owner: "_ManagedWork") -> None:
        super().__init__()
        self._owner = owner
        # the realized torch future once materialized (for callers that need
        # a genuine torch.futures.Future, e.g. the DDP reducer)
        self._fut: Optional[torch.futures.Future] = None

    def then(self, callback: Callable) -> "torch.futures.Future":
        self._owner._append(lambda v: callback(_ValueFuture(v)))
        return self

    def wait(self) -> object:
        fut = self._owner._realized()
        self._fut = fut
        return fut.wait()

    def value(self) -> object:
        return self._owner._value


class _ManagedWork(dist._Work):
    """Work with a flat ledger of deferred continuations.

    The continuations (error-normalization, gradient scaling, bucket
    scatter-back, ...) are recorded eagerly but executed lazily: the first
    ``wait()/synchronize()/block_current_stream()`` attaches ONE real
    ``Future.then`` to the underlying collective, whose callback replays the
    ledger in order inside the issuing HIP stream. Any continuation error is
    reported to the manager and the last good value is returned, so the
    training loop never sees an exception mid-backward.
    """

    def __init__(self, manager: Manager, work: dist._Work, value: object) -> None:
        super().__init__()
        self._manager = manager
        self._work = work
        self._value = value
        self._ledger: List[Callable[[object], object]] = []
        self._stream: Optional[torch.cuda.Stream] = (
            torch.cuda.current_stream() if torch.cuda.is_available() else None
        )
        self._done_fut: Optional[torch.futures.Future] = None
        self._proxy = _ManagedFuture(self)

    def _append(self, step: Callable[[object], object]) -> None:
        assert self._done_fut is None, "cannot add continuations after wait()"
        self._ledger.append(step)

    def _realized(self) -> torch.futures.Future:
        assert self._done_fut is not None, "work must be waited on first"
        return self._done_fut

    def _materialize(self) -> None:
        if self._done_fut is not None:
            return

        inner = self._work.get_future()

        def replay(f: torch.futures.Future) -> object:
            with get_stream_context(self._stream):
                value = self._value
                try:
                    f.wait()  # establishes the stream dependency
                    for step in self._ledger:
                        value = step(value)
                except Exception as e:  # noqa: BLE001
                    self._manager._logger.exception(f"continuation failed: {e}")
                    self._manager.report_error(e)
                self._value = value
                return value

        self._done_fut = future_timeout(inner, self._manager._timeout).then(replay)

    def _on_issuing_stream(self) -> None:
        if self._stream is not None:
            assert self._stream == torch.cuda.current_stream(), (
                "managed work must be completed on the stream it was issued on"
            )

    def wait(self, timeout: Optional[timedelta] = None) -> bool:
        self._on_issuing_stream()
        try:
            with get_stream_context(self._stream):
                self._work.wait()
                self._materialize()
                self._realized().wait()
            return True
        except Exception as e:  # noqa: BLE001
            self._manager._logger.exception(f"wait failed: {e}")
            self._manager.report_error(e)
            return False

    def block_current_stream(self, timeout: Optional[timedelta] = None) -> None:
        self._on_issuing_stream()
        with get_stream_context(self._stream):
            self._work.block_current_stream()
        self._materialize()

    def synchronize(self) -> None:
        self._on_issuing_stream()
        if torch.cuda.is_available():
            self.block_current_stream()
        else:
            self._materialize()

    def get_future(self) -> torch.futures.Future:
        return self._proxy
