"""Epoch-aligned BSP worker loop.

Role parity: the reference's ``worker_main`` / ``step_or_park`` run loop
and core-operator implementations (reference src/worker.rs:68-155,
src/operators.rs, src/inputs.rs, src/outputs.rs).  Design difference,
deliberate: instead of Timely's per-operator capability/frontier
machinery, the cluster advances through epochs in lockstep — one shared
integer frontier — with exchange-delivery rounds inside each epoch.
This collapses the EagerNotificator/InBuffer machinery (items are
always at the open epoch) and is the contract the GPU scheduler
shares: per-epoch batched kernel launches + RCCL exchanges.

Epoch semantics preserved from the reference:

- sources are polled cooperatively honoring ``next_awake``;
- the epoch advances every ``epoch_interval`` of wall-clock (or
  immediately at EOF) and *only* once all downstream work and recovery
  writes of the epoch completed (backpressure is implicit in the BSP
  barrier);
- stateful logics get ``on_batch`` eagerly within the open epoch,
  ``on_notify`` when timers come due, ``on_eof`` when the upstream is
  exhausted, and are snapshotted at every epoch close (awoken keys
  only);
- ``AbortExecution`` from any source aborts the whole execution with
  no snapshot for the open epoch.
"""

import random
import threading
import time
import zlib
from collections import deque
from datetime import datetime, timedelta, timezone
from typing import Any, Callable, Dict, Iterable, List, Optional, Tuple

from ..dataflow import Dataflow
from ..inputs import (
    AbortExecution,
    DynamicSource,
    FixedPartitionedSource,
)
from ..outputs import DynamicSink, FixedPartitionedSink
from ..recovery import RecoveryConfig, RecoveryStore, de_state, ser_state
from .compile import CoreStep, ExecGraph, compile_graph

EPOCH_INTERVAL_DEFAULT = timedelta(seconds=10)
_EMPTY_COOLDOWN = timedelta(milliseconds=1)
_MAX_IDLE_SLEEP = 0.005  # seconds

__all__ = ["run_main", "cluster_main"]


def _now() -> datetime:
    return datetime.now(timezone.utc)


def _route_key(key: str, worker_count: int) -> int:
    # "shard-<i>" is a reserved namespace used by the GPU columnar
    # operators to pin rank i's device batches to worker i: routing it
    # by rank (not hash) guarantees the batch never crosses the data
    # plane regardless of worker count (adler32 only lines up for
    # power-of-two worlds by arithmetic accident).
    if key.startswith("shard-"):
        suffix = key[6:]
        if suffix.isdigit():
            return int(suffix) % worker_count
    return zlib.adler32(key.encode()) % worker_count


class _Interrupted(BaseException):
    """Internal unwind signal when the cluster is shutting down."""


class _LocalCluster:
    """Shared state for thread-based workers in one process."""

    def __init__(self, worker_count: int):
        self.worker_count = worker_count
        self.barrier = threading.Barrier(worker_count)
        self.lock = threading.Lock()
        self.inboxes: List[List[Tuple[int, int, List[Any]]]] = [
            [] for _ in range(worker_count)
        ]
        self.eof_votes = [False] * worker_count
        self.close = False
        self.all_eof = False
        self.abort_vote = False
        self.abort: Optional[BaseException] = None
        self.epoch_deadline = 0.0

    def fail(self, ex: BaseException) -> None:
        with self.lock:
            if self.abort is None:
                self.abort = ex
        self.barrier.abort()


class _WorkerCtx:
    """Thread-transport worker context.

    The transport interface (`barrier`, `exchange_round`, `vote_close`)
    is shared with the multi-process ``torch.distributed`` transport in
    :mod:`bytewax_amd._engine.dist` (gloo for CPU processes, RCCL for
    GPU workers).
    """

    def __init__(self, cluster: _LocalCluster, worker_index: int):
        self.cluster = cluster
        self.worker_index = worker_index
        self.worker_count = cluster.worker_count

    def barrier(self) -> None:
        if self.worker_count == 1:
            if self.cluster.abort is not None:
                raise _Interrupted()
            return
        try:
            self.cluster.barrier.wait()
        except threading.BrokenBarrierError:
            raise _Interrupted() from None
        if self.cluster.abort is not None:
            raise _Interrupted()

    def fail(self, ex: BaseException) -> None:
        self.cluster.fail(ex)

    def exchange_round(
        self, outbox: Dict[int, List[Tuple[int, int, List[Any]]]]
    ) -> List[Tuple[int, int, List[Any]]]:
        """Deliver per-destination buffered messages; returns messages
        addressed to this worker.  Collective: all workers call this
        the same number of times per epoch."""
        c = self.cluster
        with c.lock:
            for dst, msgs in outbox.items():
                c.inboxes[dst].extend(msgs)
        self.barrier()
        with c.lock:
            msgs = c.inboxes[self.worker_index]
            c.inboxes[self.worker_index] = []
        return msgs

    def vote_close(
        self, local_eof: bool, deadline: float, aborting: bool
    ) -> Tuple[bool, bool, bool]:
        """Returns (close, all_eof, abort); identical on every worker."""
        c = self.cluster
        if self.worker_count == 1:
            all_eof = local_eof
            return (
                all_eof or aborting or time.monotonic() >= deadline,
                all_eof,
                aborting,
            )
        with c.lock:
            c.eof_votes[self.worker_index] = local_eof
            if aborting:
                c.abort_vote = True
        self.barrier()
        if self.worker_index == 0:
            c.all_eof = all(c.eof_votes)
            c.close = (
                c.all_eof or c.abort_vote or time.monotonic() >= deadline
            )
        self.barrier()
        return (c.close, c.all_eof, c.abort_vote)


# ---------------------------------------------------------------------------
# Executors: one per core step per worker.
# ---------------------------------------------------------------------------


class _Exec:
    step: CoreStep

    def process(
        self, input_idx: int, items: List[Any], epoch: int
    ) -> List[Tuple[str, List[Any]]]:
        return []

    def snapshot_rows(self, epoch: int) -> List[Tuple[str, str, int, Optional[bytes]]]:
        return []

    def close(self) -> None:
        return


class _InputExec(_Exec):
    def __init__(
        self,
        step: CoreStep,
        worker_index: int,
        worker_count: int,
        resume_states: Dict[str, Any],
    ):
        self.step = step
        self.out = step.out_streams[0]
        source = step.payload["source"]
        self.stateful = isinstance(source, FixedPartitionedSource)
        self.parts: Dict[str, Any] = {}
        if self.stateful:
            all_parts = sorted(source.list_parts())
            for i, p in enumerate(all_parts):
                if i % worker_count == worker_index:
                    self.parts[p] = source.build_part(
                        step.step_id, p, resume_states.get(p)
                    )
        elif isinstance(source, DynamicSource):
            part = source.build(step.step_id, worker_index, worker_count)
            self.parts[f"dyn-{worker_index}"] = part
        else:  # pragma: no cover - validated at graph build
            msg = f"unknown source type {type(source)!r}"
            raise TypeError(msg)
        self.eof: set = set()
        self.awake: Dict[str, Optional[datetime]] = {p: None for p in self.parts}

    def poll(
        self, now: datetime, emit: Callable[[str, List[Any]], None]
    ) -> bool:
        """Poll all live partitions; returns True if any items came in."""
        progressed = False
        for name, part in self.parts.items():
            if name in self.eof:
                continue
            awake = self.awake.get(name)
            if awake is not None and awake > now:
                continue
            try:
                batch = list(part.next_batch())
            except StopIteration:
                self.eof.add(name)
                continue
            if batch:
                progressed = True
                emit(self.out, batch)
            nxt = part.next_awake()
            if nxt is None and not batch:
                nxt = now + _EMPTY_COOLDOWN
            self.awake[name] = nxt
        return progressed

    def eof_all(self) -> bool:
        return len(self.eof) == len(self.parts)

    def next_awake(self) -> Optional[datetime]:
        soonest = None
        for name in self.parts:
            if name in self.eof:
                continue
            a = self.awake.get(name)
            if a is None:
                return None  # poll ASAP
            if soonest is None or a < soonest:
                soonest = a
        return soonest

    def snapshot_rows(self, epoch: int):
        if not self.stateful:
            return []
        return [
            (self.step.step_id, name, epoch, ser_state(part.snapshot()))
            for name, part in self.parts.items()
        ]

    def close(self) -> None:
        for part in self.parts.values():
            part.close()


class _FlatMapBatchExec(_Exec):
    def __init__(self, step: CoreStep):
        self.step = step
        self.mapper = step.payload["mapper"]
        self.out = step.out_streams[0]

    def process(self, input_idx, items, epoch):
        return [(self.out, list(self.mapper(items)))]


class _BranchExec(_Exec):
    def __init__(self, step: CoreStep):
        self.step = step
        self.predicate = step.payload["predicate"]
        self.trues, self.falses = step.out_streams

    def process(self, input_idx, items, epoch):
        ts: List[Any] = []
        fs: List[Any] = []
        for x in items:
            keep = self.predicate(x)
            if keep.__class__ is not bool:
                from ..errors import BytewaxTypeError

                msg = (
                    "return value of `predicate` in step "
                    f"{self.step.step_id!r} must be a `bool`; "
                    f"got a {type(keep)!r} instead"
                )
                raise BytewaxTypeError(msg)
            (ts if keep else fs).append(x)
        out = []
        if ts:
            out.append((self.trues, ts))
        if fs:
            out.append((self.falses, fs))
        return out


class _InspectDebugExec(_Exec):
    def __init__(self, step: CoreStep, worker_index: int):
        self.step = step
        self.inspector = step.payload["inspector"]
        self.worker_index = worker_index
        self.out = step.out_streams[0]

    def process(self, input_idx, items, epoch):
        for x in items:
            self.inspector(self.step.step_id, x, epoch, self.worker_index)
        return [(self.out, items)]


class _PassthroughExec(_Exec):
    """merge / redistribute / _noop: exchange (if any) happened on the
    inbound edge; the executor itself forwards."""

    def __init__(self, step: CoreStep):
        self.step = step
        self.out = step.out_streams[0]

    def process(self, input_idx, items, epoch):
        return [(self.out, items)]


class _OutputExec(_Exec):
    def __init__(
        self,
        step: CoreStep,
        worker_index: int,
        worker_count: int,
        resume_states: Dict[str, Any],
    ):
        self.step = step
        sink = step.payload["sink"]
        self.sink = sink
        self.stateful = isinstance(sink, FixedPartitionedSink)
        self.parts: Dict[str, Any] = {}
        self.all_parts: List[str] = []
        if self.stateful:
            self.all_parts = sorted(sink.list_parts())
            for i, p in enumerate(self.all_parts):
                if i % worker_count == worker_index:
                    self.parts[p] = sink.build_part(
                        step.step_id, p, resume_states.get(p)
                    )
        elif isinstance(sink, DynamicSink):
            self.parts[f"dyn-{worker_index}"] = sink.build(
                step.step_id, worker_index, worker_count
            )
        else:  # pragma: no cover
            msg = f"unknown sink type {type(sink)!r}"
            raise TypeError(msg)

    def part_for_key(self, key: str) -> int:
        """Global partition index for an item key."""
        return self.sink.part_fn(key) % len(self.all_parts)

    def process(self, input_idx, items, epoch):
        if not self.stateful:
            part = next(iter(self.parts.values()))
            part.write_batch(items)
            return []
        by_part: Dict[str, List[Any]] = {}
        for kv in items:
            try:
                k, v = kv
            except (TypeError, ValueError) as ex:
                msg = (
                    f"step {self.step.step_id!r} requires `(key, value)` "
                    f"2-tuples from upstream; got a {type(kv)!r} instead"
                )
                raise TypeError(msg) from ex
            name = self.all_parts[self.part_for_key(k)]
            by_part.setdefault(name, []).append(v)
        for name, values in by_part.items():
            part = self.parts.get(name)
            if part is None:  # pragma: no cover - mis-routed
                msg = (
                    f"step {self.step.step_id!r}: partition {name!r} is "
                    "not local to this worker"
                )
                raise AssertionError(msg)
            part.write_batch(values)
        return []

    def snapshot_rows(self, epoch: int):
        if not self.stateful:
            return []
        return [
            (self.step.step_id, name, epoch, ser_state(part.snapshot()))
            for name, part in self.parts.items()
        ]

    def close(self) -> None:
        for part in self.parts.values():
            part.close()


class _StatefulBatchExec(_Exec):
    def __init__(self, step: CoreStep, resume_states: Dict[str, Any]):
        self.step = step
        self.builder = step.payload["builder"]
        self.out = step.out_streams[0]
        self.logics: Dict[str, Any] = {}
        self.sched: Dict[str, datetime] = {}
        self.awoken: set = set()
        # Build logics for resumed keys immediately so EOF-only keys
        # still participate (reference operators.rs:976-1006).
        for key, state in resume_states.items():
            logic = self.builder(state)
            self.logics[key] = logic
            self._refresh_notify(key)

    def _refresh_notify(self, key: str) -> None:
        logic = self.logics.get(key)
        if logic is None:
            self.sched.pop(key, None)
            return
        at = logic.notify_at()
        if at is not None:
            self.sched[key] = at
        else:
            self.sched.pop(key, None)

    def _handle(self, key: str, ws: Iterable[Any], discard: bool, out: List):
        for w in ws:
            out.append((key, w))
        self.awoken.add(key)
        if discard:
            self.logics.pop(key, None)
            self.sched.pop(key, None)
        else:
            self._refresh_notify(key)

    def process(self, input_idx, items, epoch):
        grouped: Dict[str, List[Any]] = {}
        get = grouped.get
        for kv in items:
            # Hot loop: exact-type checks first (`type(x) is` is much
            # cheaper than isinstance), subclass fallbacks second.
            if (
                type(kv) is not tuple and not isinstance(kv, tuple)
            ) or len(kv) != 2:
                from ..errors import BytewaxTypeError

                msg = (
                    f"step {self.step.step_id!r} requires `(key, value)` "
                    f"2-tuples from upstream; got a {type(kv)!r} instead"
                )
                raise BytewaxTypeError(msg)
            k, v = kv
            if type(k) is not str and not isinstance(k, str):
                from ..errors import BytewaxTypeError

                msg = (
                    f"step {self.step.step_id!r} requires keys to be `str`; "
                    f"got a {type(k)!r} instead"
                )
                raise BytewaxTypeError(msg)
            lst = get(k)
            if lst is None:
                grouped[k] = [v]
            else:
                lst.append(v)
        out: List[Tuple[str, Any]] = []
        # Hot loop: `_handle`/`_refresh_notify` inlined (two Python
        # frames per (key, batch) otherwise), and the sched pop is
        # skipped while no key schedules notifications at all.
        out_append = out.append
        logics = self.logics
        sched = self.sched
        awoken_add = self.awoken.add
        builder = self.builder
        for k in sorted(grouped):
            logic = logics.get(k)
            if logic is None:
                logic = builder(None)
                logics[k] = logic
            ws, discard = logic.on_batch(grouped[k])
            for w in ws:
                out_append((k, w))
            awoken_add(k)
            if discard:
                logics.pop(k, None)
                sched.pop(k, None)
            else:
                at = logic.notify_at()
                if at is not None:
                    sched[k] = at
                elif sched:
                    sched.pop(k, None)
        if out:
            return [(self.out, out)]
        return []

    def fire_timers(self, now: datetime):
        due = [k for k, at in self.sched.items() if at <= now]
        out: List[Tuple[str, Any]] = []
        for k in sorted(due):
            logic = self.logics.get(k)
            if logic is None:
                self.sched.pop(k, None)
                continue
            ws, discard = logic.on_notify()
            self._handle(k, ws, discard, out)
        if out:
            return [(self.out, out)]
        return []

    def next_awake(self) -> Optional[datetime]:
        if not self.sched:
            return None
        return min(self.sched.values())

    def on_eof(self):
        out: List[Tuple[str, Any]] = []
        for k in sorted(self.logics):
            logic = self.logics[k]
            ws, discard = logic.on_eof()
            self._handle(k, ws, discard, out)
        if out:
            return [(self.out, out)]
        return []

    def snapshot_rows(self, epoch: int):
        rows = []
        for key in sorted(self.awoken):
            logic = self.logics.get(key)
            if logic is None:
                rows.append((self.step.step_id, key, epoch, None))  # Discard
            else:
                rows.append(
                    (self.step.step_id, key, epoch, ser_state(logic.snapshot()))
                )
        self.awoken.clear()
        return rows


# ---------------------------------------------------------------------------
# The worker
# ---------------------------------------------------------------------------


class _Worker:
    def __init__(
        self,
        graph: ExecGraph,
        ctx: _WorkerCtx,
        store: Optional[RecoveryStore],
        ex_num: int,
        resume_epoch: int,
        epoch_interval: timedelta,
        resume_snaps: Dict[Tuple[str, str], Any],
    ):
        self.graph = graph
        self.ctx = ctx
        self.store = store
        self.ex_num = ex_num
        self.epoch = resume_epoch
        self.epoch_interval = epoch_interval
        self.fifo: deque = deque()
        self.outbox: Dict[int, List[Tuple[int, int, List[Any]]]] = {}
        self.execs: List[_Exec] = []
        # A user exception caught mid-drain; resolved to an abort vote
        # at the next epoch-close round.
        self._poison: Optional[BaseException] = None
        w, n = ctx.worker_index, ctx.worker_count

        for step in graph.steps:
            if step.op_name == "input":
                states = self._states_for_parts(
                    resume_snaps, step.step_id, w, n,
                    self._source_parts(step),
                )
                self.execs.append(_InputExec(step, w, n, states))
            elif step.op_name == "flat_map_batch":
                self.execs.append(_FlatMapBatchExec(step))
            elif step.op_name == "branch":
                self.execs.append(_BranchExec(step))
            elif step.op_name == "inspect_debug":
                self.execs.append(_InspectDebugExec(step, w))
            elif step.op_name in ("merge", "redistribute", "_noop"):
                self.execs.append(_PassthroughExec(step))
            elif step.op_name == "output":
                sink = step.payload["sink"]
                parts = (
                    sorted(sink.list_parts())
                    if isinstance(sink, FixedPartitionedSink)
                    else []
                )
                states = self._states_for_parts(
                    resume_snaps, step.step_id, w, n, parts
                )
                self.execs.append(_OutputExec(step, w, n, states))
            elif step.op_name == "stateful_batch":
                states = {
                    key: state
                    for (sid, key), state in resume_snaps.items()
                    if sid == step.step_id and _route_key(key, n) == w
                }
                self.execs.append(_StatefulBatchExec(step, states))
            else:  # pragma: no cover
                msg = f"unknown core operator {step.op_name!r}"
                raise AssertionError(msg)

        self.stateful_execs = [
            e for e in self.execs if isinstance(e, _StatefulBatchExec)
        ]
        self.input_execs = [e for e in self.execs if isinstance(e, _InputExec)]

    @staticmethod
    def _source_parts(step: CoreStep) -> List[str]:
        source = step.payload["source"]
        if isinstance(source, FixedPartitionedSource):
            return sorted(source.list_parts())
        return []

    @staticmethod
    def _states_for_parts(
        resume_snaps: Dict[Tuple[str, str], Any],
        step_id: str,
        worker_index: int,
        worker_count: int,
        all_parts: List[str],
    ) -> Dict[str, Any]:
        """Resume states for the partitions this worker is primary for."""
        idx = {p: i for i, p in enumerate(all_parts)}
        out = {}
        for (sid, key), state in resume_snaps.items():
            if sid != step_id:
                continue
            i = idx.get(key)
            if i is not None and i % worker_count == worker_index:
                out[key] = state
        return out

    # -- item propagation --

    def _emit(self, stream_id: str, items: List[Any]) -> None:
        consumers = self.graph.consumers.get(stream_id, ())
        n_cons = len(consumers)
        for ci, (step_idx, input_idx) in enumerate(consumers):
            payload = items if ci == n_cons - 1 else list(items)
            self._route(step_idx, input_idx, payload)

    def _route(self, step_idx: int, input_idx: int, items: List[Any]) -> None:
        step = self.graph.steps[step_idx]
        n = self.ctx.worker_count
        if step.exchange == "local" or n == 1:
            self.fifo.append((step_idx, input_idx, items))
            return
        buckets: Dict[int, List[Any]] = {}
        if step.exchange == "key":
            for kv in items:
                try:
                    k, _v = kv
                except (TypeError, ValueError) as ex:
                    from ..errors import BytewaxTypeError

                    msg = (
                        f"step {step.step_id!r} requires `(key, value)` "
                        f"2-tuples from upstream; got a {type(kv)!r} instead"
                    )
                    raise BytewaxTypeError(msg) from ex
                if not isinstance(k, str):
                    from ..errors import BytewaxTypeError

                    msg = (
                        f"step {step.step_id!r} requires keys to be `str`; "
                        f"got a {type(k)!r} instead"
                    )
                    raise BytewaxTypeError(msg)
                buckets.setdefault(_route_key(k, n), []).append(kv)
        elif step.exchange == "random":
            for kv in items:
                buckets.setdefault(random.randrange(n), []).append(kv)
        elif step.exchange == "part":
            ex = self.execs[step_idx]
            for kv in items:
                try:
                    k, _v = kv
                except (TypeError, ValueError) as exc:
                    msg = (
                        f"step {step.step_id!r} requires `(key, value)` "
                        f"2-tuples from upstream; got a {type(kv)!r} instead"
                    )
                    raise TypeError(msg) from exc
                part_idx = ex.part_for_key(k)
                buckets.setdefault(part_idx % n, []).append(kv)
        for dst, chunk in buckets.items():
            if dst == self.ctx.worker_index:
                self.fifo.append((step_idx, input_idx, chunk))
            else:
                self.outbox.setdefault(dst, []).append(
                    (step_idx, input_idx, chunk)
                )

    def _drain(self) -> None:
        from .._metrics import metrics_enabled, observe_batch
        from ..tracing import tracing_active

        self._metrics_on = metrics_enabled()
        self._tracing_on = tracing_active()
        instrumented = self._metrics_on or self._tracing_on
        if self._poison is not None:
            # Aborting: discard queued work but keep the collective
            # schedule aligned (exchange rounds still run with empty
            # outboxes so peers don't block on a missing collective).
            self.fifo.clear()
            self.outbox.clear()
            return
        try:
            self._drain_inner(instrumented, observe_batch)
        except Exception as ex:  # noqa: BLE001
            # A user function raised mid-epoch: poison this worker so
            # the next epoch-close vote aborts the whole cluster
            # (reference run.rs:273-304 panic hook + shutdown flag);
            # this rank re-raises the error after the vote.
            self._poison = ex
            self.fifo.clear()
            self.outbox.clear()

    def _drain_inner(self, instrumented, observe_batch) -> None:
        while self.fifo:
            step_idx, input_idx, items = self.fifo.popleft()
            ex = self.execs[step_idx]
            if instrumented:
                t0 = time.perf_counter()
                outs = ex.process(input_idx, items, self.epoch)
                n_out = 0
                for stream_id, out_items in outs:
                    n_out += len(out_items)
                    if out_items:
                        self._emit(stream_id, out_items)
                t1 = time.perf_counter()
                if self._metrics_on:
                    observe_batch(
                        ex.step.step_id,
                        self.ctx.worker_index,
                        len(items),
                        n_out,
                        t1 - t0,
                    )
                if self._tracing_on:
                    from ..tracing import record_operator_span

                    record_operator_span(
                        ex.step.step_id,
                        self.ctx.worker_index,
                        len(items),
                        n_out,
                        t0,
                        t1,
                    )
            else:
                for stream_id, out_items in ex.process(
                    input_idx, items, self.epoch
                ):
                    if out_items:
                        self._emit(stream_id, out_items)

    def _exchange_rounds(self) -> None:
        self._drain()
        if self.ctx.worker_count == 1:
            return
        for _ in range(self.graph.n_exchange_rounds):
            outbox, self.outbox = self.outbox, {}
            for msg in self.ctx.exchange_round(outbox):
                self.fifo.append(msg)
            self._drain()

    # -- the run loop --

    def run(self) -> None:
        interval_s = self.epoch_interval.total_seconds()
        while True:
            self.ctx.barrier()  # align epoch start
            deadline = time.monotonic() + interval_s
            all_eof = False
            abort_exc: Optional[BaseException] = None
            while True:
                now = _now()
                progressed = False
                if abort_exc is None:
                    try:
                        for src in self.input_execs:
                            progressed |= src.poll(now, self._emit)
                        for sf in self.stateful_execs:
                            for stream_id, out_items in sf.fire_timers(
                                now
                            ):
                                progressed = True
                                self._emit(stream_id, out_items)
                    except AbortExecution as ex:
                        abort_exc = ex
                    except Exception as ex:  # noqa: BLE001
                        # A user function raised: vote the whole
                        # cluster into an abort so peers exit cleanly
                        # (reference run.rs:273-304 panic hook +
                        # shutdown flag) instead of dying on a broken
                        # collective; this rank re-raises the error.
                        abort_exc = ex
                self._exchange_rounds()
                local_eof = all(e.eof_all() for e in self.input_execs)
                failing = abort_exc or self._poison
                close, all_eof, abort = self.ctx.vote_close(
                    local_eof, deadline, failing is not None
                )
                if abort:
                    # Abort the whole execution: no snapshot for the
                    # open epoch; resume replays it.
                    self.ctx.fail(failing or AbortExecution())
                    raise _Interrupted()
                if close:
                    break
                if not progressed:
                    self._idle_sleep(deadline)

            if all_eof:
                self._eof_pass()

            self._close_epoch()
            self.epoch += 1
            if all_eof:
                break
        for e in self.execs:
            e.close()

    def _idle_sleep(self, deadline: float) -> None:
        now = _now()
        waits = [_MAX_IDLE_SLEEP, max(0.0, deadline - time.monotonic())]
        for src in self.input_execs:
            a = src.next_awake()
            if a is not None:
                waits.append(max(0.0, (a - now).total_seconds()))
            else:
                waits.append(0.0)
        for sf in self.stateful_execs:
            a = sf.next_awake()
            if a is not None:
                waits.append(max(0.0, (a - now).total_seconds()))
        dur = min(waits)
        if dur > 0:
            time.sleep(min(dur, _MAX_IDLE_SLEEP))

    def _eof_pass(self) -> None:
        """Call `on_eof` through the graph in topological order."""
        for idx in self.graph.topo:
            step = self.graph.steps[idx]
            if step.op_name == "stateful_batch":
                ex = self.execs[idx]
                for stream_id, out_items in ex.on_eof():
                    self._emit(stream_id, out_items)
            self._exchange_rounds()

    def _close_epoch(self) -> None:
        if self.store is not None:
            rows = []
            for e in self.execs:
                rows.extend(e.snapshot_rows(self.epoch))
            self.store.write_snaps(rows)
        else:
            # Still reset awoken-key tracking.
            for e in self.stateful_execs:
                e.awoken.clear()
        self.ctx.barrier()
        if self.store is not None:
            self.store.write_frontier(
                self.ex_num, self.ctx.worker_index, self.epoch + 1
            )
        self.ctx.barrier()
        if self.store is not None and self.ctx.worker_index == 0:
            self.store.commit_and_gc(self.epoch + 1)
        self.ctx.barrier()


# ---------------------------------------------------------------------------
# Entry points
# ---------------------------------------------------------------------------


def _load_resume_snaps(
    store: Optional[RecoveryStore], resume_epoch: int
) -> Dict[Tuple[str, str], Any]:
    snaps: Dict[Tuple[str, str], Any] = {}
    if store is None:
        return snaps
    for step_id, state_key, _epoch, ser_change in store.load_resume_snaps(
        resume_epoch
    ):
        if ser_change is None:
            continue  # Discard: no state
        snaps[(step_id, state_key)] = de_state(ser_change)
    return snaps


def _run_cluster(
    flow: Dataflow,
    worker_count: int,
    epoch_interval: Optional[timedelta],
    recovery_config: Optional[RecoveryConfig],
) -> None:
    if epoch_interval is None:
        epoch_interval = EPOCH_INTERVAL_DEFAULT
    if epoch_interval < timedelta(0):
        msg = "epoch interval must be non-negative"
        raise ValueError(msg)
    graph = compile_graph(flow)

    from .._webserver import maybe_start_webserver

    maybe_start_webserver(flow)

    store = None
    ex_num, resume_epoch = 0, 1
    if recovery_config is not None:
        interval = epoch_interval if epoch_interval > timedelta(0) else timedelta(
            microseconds=1
        )
        store = RecoveryStore(recovery_config, interval)
        ex_num, resume_epoch = store.resume_from()
        store.write_ex(ex_num, worker_count, resume_epoch)
    resume_snaps = _load_resume_snaps(store, resume_epoch)

    cluster = _LocalCluster(worker_count)
    workers = [
        _Worker(
            graph,
            _WorkerCtx(cluster, w),
            store,
            ex_num,
            resume_epoch,
            epoch_interval,
            resume_snaps,
        )
        for w in range(worker_count)
    ]

    def run_worker(wk: _Worker) -> None:
        try:
            wk.run()
        except _Interrupted:
            pass
        except BaseException as ex:  # noqa: BLE001
            cluster.fail(ex)

    threads = [
        threading.Thread(
            target=run_worker, args=(wk,), name=f"bytewax-amd-worker-{w}",
            daemon=True,
        )
        for w, wk in enumerate(workers[1:], start=1)
    ]
    for t in threads:
        t.start()
    run_worker(workers[0])
    for t in threads:
        t.join()

    if store is not None:
        store.close()
    if cluster.abort is not None:
        if isinstance(cluster.abort, AbortExecution):
            return  # aborted execution ends cleanly; resume replays
        if isinstance(cluster.abort, KeyboardInterrupt):
            raise cluster.abort
        # User/step errors surface WRAPPED as BytewaxRuntimeError
        # (the original chained as __cause__), matching the
        # reference's panic bridging: its pytests assert the custom
        # exception does NOT escape raw (test_execution.py
        # test_reraises_custom_exception's nested raises).
        from ..errors import BytewaxRuntimeError, BytewaxTypeError

        if isinstance(cluster.abort, (BytewaxRuntimeError, BytewaxTypeError)):
            # Engine-origin contract errors are already RuntimeError
            # (or the dual TypeError/RuntimeError) — no double wrap.
            raise cluster.abort
        msg = "error while executing dataflow; see the cause above"
        raise BytewaxRuntimeError(msg) from cluster.abort


def run_main(
    flow: Dataflow,
    *,
    epoch_interval: Optional[timedelta] = None,
    recovery_config: Optional[RecoveryConfig] = None,
) -> None:
    """Execute a dataflow in the current process with a single worker.

    Blocks until execution is complete.  Use for testing and
    prototyping; see :func:`cluster_main` for multi-worker execution.
    """
    _run_cluster(flow, 1, epoch_interval, recovery_config)


def cluster_main(
    flow: Dataflow,
    addresses: List[str],
    proc_id: int,
    *,
    epoch_interval: Optional[timedelta] = None,
    recovery_config: Optional[RecoveryConfig] = None,
    worker_count_per_proc: int = 1,
) -> None:
    """Execute a dataflow in the current process as part of a cluster.

    Blocks until execution is complete.

    :arg addresses: Addresses of all the processes in the cluster,
        including this one.  An empty list means a single-process
        cluster.  Multi-process CPU clusters use the
        ``torch.distributed`` gloo transport (see
        :mod:`bytewax_amd._engine.dist`); GPU clusters use RCCL.
    :arg proc_id: Index of this process within ``addresses``.
    :arg worker_count_per_proc: Number of worker threads in this
        process.
    """
    if addresses and len(addresses) > 1:
        from .dist import dist_cluster_main

        return dist_cluster_main(
            flow,
            addresses,
            proc_id,
            epoch_interval=epoch_interval,
            recovery_config=recovery_config,
            worker_count_per_proc=worker_count_per_proc,
        )
    _run_cluster(flow, worker_count_per_proc, epoch_interval, recovery_config)
