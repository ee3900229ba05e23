"""Dataflow operators for the GPU columnar path.

These integrate :mod:`bytewax_amd.gpu` device state into the engine's
epoch machinery: `keyed_window_agg` is a `stateful_batch` under the
hood, so EOF flushes and recovery snapshots behave like any other
stateful operator; the snapshot payload is the pinned-host spill of
the HBM table.

Constraint of the collective path: with ``torch.distributed``
initialized, every rank must feed this operator exactly one
RecordBatch per scheduling step (the sources below do), because the
keyed exchange is an RCCL collective.
"""

from dataclasses import dataclass
from datetime import datetime, timedelta
from typing import Any, Dict, List, Optional

import bytewax_amd.operators as op
from ..dataflow import Stream, operator
from ..inputs import DynamicSource, StatelessSourcePartition
from ..outputs import DynamicSink, StatelessSinkPartition
from ..operators import StatefulBatchLogic
from . import (
    AGG_COUNT,
    AGG_SUM,
    RecordBatch,
    WindowAggState,
    _ms,
    exchange_by_key,
)

__all__ = [
    "CollectCountsSink",
    "SyntheticEventSource",
    "filter_batch",
    "map_batch",
    "keyed_session_agg",
    "keyed_stats_agg",
    "keyed_window_agg",
    "keyed_window_agg_str",
    "stream_join",
]


class _SyntheticPartition(StatelessSourcePartition[RecordBatch]):
    """Generates keyed event batches on-device.

    A pool of pre-generated key batches is reused round-robin (RNG off
    the hot path); timestamps advance at a fixed simulated rate so
    tumbling windows close at a steady cadence.
    """

    def __init__(
        self,
        device,
        events_per_batch: int,
        n_batches: Optional[int],
        vocab: int,
        sim_ms_per_batch: int,
        align_ms: int,
        seed: int,
        pool: int = 8,
        vals: bool = False,
        per_poll: int = 1,
        ts32: bool = True,
    ):
        import torch

        self.device = device
        self.n_batches = n_batches
        self.emitted = 0
        self.per_poll = per_poll
        self.sim_ms_per_batch = sim_ms_per_batch
        self.align_ms = align_ms
        # Generate pools directly on the target device (CPU-side
        # generation + H2D copies would dominate short runs).
        g = torch.Generator(device=device).manual_seed(seed)
        self.key_pool = [
            torch.randint(
                0,
                vocab,
                (events_per_batch,),
                dtype=torch.int32,
                generator=g,
                device=device,
            )
            for _ in range(pool)
        ]
        # Zero-based timestamp template; int32 by default (spans <
        # sim_ms_per_batch ms << 2^31) — halves the hot path's
        # timestamp read traffic.  int64 for consumers that need
        # absolute columns (hipGraph capture path).
        base = torch.arange(events_per_batch, dtype=torch.int64, device=device)
        self.ts_template = (
            base * sim_ms_per_batch
        ) // max(events_per_batch, 1)
        if ts32:
            self.ts_template = self.ts_template.to(torch.int32)
        self.val_pool = None
        if vals:
            self.val_pool = [
                torch.randint(
                    0,
                    100,
                    (events_per_batch,),
                    dtype=torch.int64,
                    generator=g,
                    device=device,
                )
                for _ in range(pool)
            ]

    def _mk_batch(self, i: int) -> RecordBatch:
        start = self.align_ms + i * self.sim_ms_per_batch
        keys = self.key_pool[i % len(self.key_pool)]
        vals = (
            self.val_pool[i % len(self.val_pool)]
            if self.val_pool is not None
            else None
        )
        # Zero-based template + scalar base: consumers (insert kernels
        # and the exchange wire format) apply the base without ever
        # materializing an absolute int64 timestamp column.
        return RecordBatch(
            keys,
            self.ts_template,
            vals,
            max_ts=start + self.sim_ms_per_batch - 1,
            ts_base=start,
        )

    def next_batch(self) -> List[RecordBatch]:
        if self.n_batches is not None and self.emitted >= self.n_batches:
            raise StopIteration()
        out = []
        for _ in range(self.per_poll):
            if (
                self.n_batches is not None
                and self.emitted >= self.n_batches
            ):
                break
            out.append(self._mk_batch(self.emitted))
            self.emitted += 1
        return out


@dataclass
class SyntheticEventSource(DynamicSource):
    """Per-worker synthetic keyed event stream (columnar, on-device).

    :arg events_per_batch: Events per RecordBatch per worker.
    :arg n_batches: Batches per worker before EOF (None = endless).
    :arg vocab: Key cardinality.
    :arg align_to: Window alignment instant; timestamps start here.
    :arg sim_ms_per_batch: How much simulated time one batch spans.
    """

    events_per_batch: int
    n_batches: Optional[int]
    vocab: int
    align_to: datetime
    sim_ms_per_batch: int = 1000
    device: str = "cuda"
    seed: int = 42
    with_vals: bool = False
    per_poll: int = 1
    ts32: bool = True

    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> _SyntheticPartition:
        import torch

        dev = torch.device(self.device)
        return _SyntheticPartition(
            dev,
            self.events_per_batch,
            self.n_batches,
            self.vocab,
            self.sim_ms_per_batch,
            _ms(self.align_to),
            self.seed + worker_index * 7919,
            vals=self.with_vals,
            per_poll=self.per_poll,
            ts32=self.ts32,
        )


class _DeviceWindowLogic(StatefulBatchLogic):
    """Holds the HBM window table; exchanges, inserts, closes.

    With the exchange enabled, the RCCL all-to-allv of step N is
    issued asynchronously and overlaps the insert of step N-1's
    received batch on the compute stream (one-step pipeline; the
    pending batch is flushed at EOF and before snapshots).
    """

    def __init__(
        self,
        state: WindowAggState,
        wait_ms: int,
        exchange: bool,
        resume: Optional[Dict[str, Any]],
        shard: Optional[str] = None,
        world: int = 1,
        registry: Optional[Any] = None,
    ):
        self.state = state
        self.wait_ms = wait_ms
        self.exchange = exchange
        self.pending: Optional[tuple] = None  # (works, RecordBatch)
        # Rescale support: `registry` collects spilled rows from
        # donor logics (shards restored at a different world size);
        # applied via one aligned exchange on the first activation.
        self._shard = shard
        self._world = world
        self._registry = registry
        self._applied_rescale = registry is None
        # Closed rows resolved during snapshot() (which cannot emit):
        # re-emitted on the next activation; persisted in the snapshot
        # so a resume replays them exactly once.
        self._carry: Optional[RecordBatch] = None
        # Bounded in-flight window: per-step events let the host stay
        # at most PIPELINE steps ahead of the device, so the close
        # readback (and the step latency) never waits on a deep queue.
        self._evq: List[Any] = []
        if resume is not None:
            import torch

            resume = dict(resume)
            resume.pop("__world__", None)
            resume.pop("__shard__", None)
            carry = resume.pop("__carry__", None)
            if carry is not None:
                dev = self.state.device
                self._carry = RecordBatch(
                    torch.as_tensor(carry["keys"]).to(dev),
                    torch.as_tensor(carry["ts"]).to(dev),
                    torch.as_tensor(carry["vals"]).to(dev),
                )
            self.state.restore_from_host(resume)

    PIPELINE = 2

    def _take_carry(self, out: List[RecordBatch]) -> None:
        if self._carry is not None:
            out.append(self._carry)
            self._carry = None

    def _add_carry(self, batch: Optional[RecordBatch]) -> None:
        import torch

        if batch is None or len(batch) == 0:
            return
        if self._carry is None:
            self._carry = batch
        else:
            self._carry = RecordBatch(
                torch.cat([self._carry.keys, batch.keys]),
                torch.cat([self._carry.ts, batch.ts]),
                torch.cat([self._carry.vals, batch.vals]),
            )

    def _apply_rescale(self) -> None:
        """Consume donor-shard snapshots (worker count changed since
        the snapshot was taken): every restored accumulator row is
        re-exchanged by key hash so it lands on its new owning worker,
        then re-added with SUM semantics.  Runs exactly once, on the
        first activation; with the exchange enabled EVERY rank calls
        the collective exactly once (empty batches are fine), so the
        all-to-all stays aligned whether or not this rank held
        donors."""
        if self._applied_rescale:
            return
        self._applied_rescale = True
        import numpy as np
        import torch

        rt = self._registry
        dev = self.state.device
        keys_l, ts_l, vals_l = [], [], []
        max_ts = None
        for entry in rt.rescale_rows:
            if entry["consumed"]:
                continue
            entry["consumed"] = True
            snap = entry["snap"]
            if len(snap.get("keys", ())):
                keys_l.append(np.asarray(snap["keys"], dtype="int32"))
                wins = np.asarray(snap["wins"], dtype="int64")
                ts_l.append(
                    wins * self.state.len_ms + self.state.align_ms
                )
                vals_l.append(np.asarray(snap["vals"], dtype="int64"))
            c = snap.get("__carry__")
            if c is not None and len(c["keys"]):
                self._add_carry(
                    RecordBatch(
                        torch.as_tensor(c["keys"]).to(dev),
                        torch.as_tensor(c["ts"]).to(dev),
                        torch.as_tensor(c["vals"]).to(dev),
                    )
                )
            mts = snap.get("max_ts")
            if mts is not None:
                max_ts = mts if max_ts is None else max(max_ts, mts)
        if not self.exchange and not keys_l:
            return
        if keys_l:
            keys_np = np.concatenate(keys_l)
            ts_np = np.concatenate(ts_l)
            vals_np = np.concatenate(vals_l)
            batch = RecordBatch(
                torch.as_tensor(keys_np).to(dev),
                torch.as_tensor(ts_np).to(dev),
                torch.as_tensor(vals_np).to(dev),
                max_ts=int(ts_np.max()),
            )
        else:
            batch = RecordBatch(
                torch.zeros(0, dtype=torch.int32, device=dev),
                torch.zeros(0, dtype=torch.int64, device=dev),
                torch.zeros(0, dtype=torch.int64, device=dev),
                max_ts=0,
            )
        if self.exchange:
            batch = exchange_by_key(batch)
        self.state.reinsert_labeled(batch)
        if max_ts is not None and max_ts > self.state.max_ts_host:
            self.state.max_ts_host = max_ts

    def _use_pipe(self) -> bool:
        import os

        st = self.state
        return (
            not st.cpu
            and st.radix
            and not getattr(st, "radix_v2", False)
            and os.environ.get("BYTEWAX_PY_PIPELINE", "1") != "0"
        )

    def _flush_pending(self) -> None:
        if self.pending is None:
            return
        works, batch = self.pending
        self.pending = None
        if self._use_pipe():
            # Two-stream insert: the exchange completion and this
            # batch's scatter land on the side stream, overlapping the
            # previous batch's aggregation on the compute stream.
            if hasattr(batch, "materialize"):
                self.state.insert_pipelined(
                    batch.keys, batch.ts32, batch.vals, 0,
                    batch.seg_counts, batch.seg_bases, batch.max_ts,
                    works,
                )
            else:
                self.state.insert_pipelined(
                    batch.keys, batch.ts, batch.vals, batch.ts_base,
                    [], [], batch.max_ts, works,
                )
            return
        for w in works:
            w.wait()
        if hasattr(batch, "materialize"):
            if hasattr(self.state, "insert_lazy"):
                # Wire-format insert: no int64 timestamp rebuild.
                self.state.insert_lazy(batch)
                return
            batch = batch.materialize()
        self.state.insert(batch)

    def on_batch(self, batches: List[RecordBatch]):
        import torch

        self._apply_rescale()
        out: List[RecordBatch] = []
        self._take_carry(out)
        # Close cadence is per BATCH (not per poll): sources may hand
        # the engine many batches per scheduling step, and the window
        # table is sized for the close-every-window working set.  The
        # close of batch i resolves while batch i+1 inserts (one-step
        # deferred readback).
        for batch in batches:
            closed = self.state.close_resolve()
            if closed is not None:
                out.append(closed)
            if self.exchange:
                works, exchanged = exchange_by_key(batch, async_op=True)
                self._flush_pending()
                self.pending = (works, exchanged)
            elif self._use_pipe():
                self.state.insert_pipelined(
                    batch.keys, batch.ts, batch.vals, batch.ts_base,
                    [], [], batch.max_ts, None,
                )
            else:
                self.state.insert(batch)
            self.state.close_launch(self.wait_ms)
            if not self.state.cpu:
                ev = torch.cuda.Event()
                ev.record()
                self._evq.append(ev)
                if len(self._evq) > self.PIPELINE:
                    self._evq.pop(0).synchronize()
        return (out, StatefulBatchLogic.RETAIN)

    def on_eof(self):
        self._apply_rescale()
        out: List[RecordBatch] = []
        self._take_carry(out)
        self._flush_pending()
        closed = self.state.close_resolve()
        if closed is not None:
            out.append(closed)
        final = self.state.close_all()
        if final is not None:
            out.append(final)
        return (out, StatefulBatchLogic.RETAIN)

    def snapshot(self) -> Dict[str, Any]:
        import torch

        self._flush_pending()
        resolved = self.state.close_resolve()
        if resolved is not None:
            if self._carry is not None:
                self._carry = RecordBatch(
                    torch.cat([self._carry.keys, resolved.keys]),
                    torch.cat([self._carry.ts, resolved.ts]),
                    torch.cat([self._carry.vals, resolved.vals]),
                )
            else:
                self._carry = resolved
        snap = dict(self.state.snapshot_to_host())
        if self._carry is not None:
            snap["__carry__"] = {
                "keys": self._carry.keys.cpu().numpy().copy(),
                "ts": self._carry.ts.cpu().numpy().copy(),
                "vals": self._carry.vals.cpu().numpy().copy(),
            }
        if self._shard is not None:
            snap["__shard__"] = self._shard
            snap["__world__"] = self._world
        return snap


class _DonorLogic(StatefulBatchLogic):
    """A shard restored under a different cluster shape.

    Registers its spilled rows with the worker's rescale registry (the
    active shard re-exchanges and re-adds them on its first
    activation) and then lingers, re-snapshotting the rows each epoch,
    until consumption is durable: it only DISCARDs after the active
    shard has consumed the rows, so the donor rows and the merged
    state always change hands inside one atomic epoch commit
    (exactly-once across crashes at any point)."""

    def __init__(self, snap: Dict[str, Any], rt: Any):
        self._entry = {"snap": snap, "consumed": False}
        rt.rescale_rows.append(self._entry)

    def on_batch(self, batches):
        # Donors hold foreign shard names; live batches always carry
        # the worker's own shard key.  Receiving one means the shard
        # keying is broken — dropping data silently is worse.
        msg = "donor shard received live batches; shard keying broken"
        raise RuntimeError(msg)

    def on_notify(self):
        if self._entry["consumed"]:
            return ([], StatefulBatchLogic.DISCARD)
        return ([], StatefulBatchLogic.RETAIN)

    def notify_at(self):
        from datetime import datetime, timezone

        return datetime.now(timezone.utc)

    def snapshot(self) -> Dict[str, Any]:
        if self._entry["consumed"]:
            return {"__consumed__": True}
        return self._entry["snap"]


class _ConsumedDonor(StatefulBatchLogic):
    """Tombstone for an already-consumed donor snapshot: discards on
    the first timer pass (writing the discard into recovery)."""

    def on_batch(self, batches):  # pragma: no cover — never routed
        return ([], StatefulBatchLogic.RETAIN)

    def on_notify(self):
        return ([], StatefulBatchLogic.DISCARD)

    def notify_at(self):
        from datetime import datetime, timezone

        return datetime.now(timezone.utc)

    def snapshot(self) -> Dict[str, Any]:
        return {"__consumed__": True}


@operator
def keyed_window_agg(
    step_id: str,
    up: Stream[RecordBatch],
    align_to: datetime,
    length: timedelta,
    mode: str = "count",
    wait: timedelta = timedelta(0),
    slots_pow: int = 20,
    dedup: bool = False,
    out_cap: int = 1 << 20,
    device: str = "cuda",
    exchange: Optional[bool] = None,
    radix: bool = False,
    region_bits: int = 11,
    offset: Optional[timedelta] = None,
) -> Stream[RecordBatch]:
    """Keyed tumbling-window aggregation over columnar batches on GPU.

    The engine-side equivalent of ``key_on |> fold_window(count/sum)``
    for device-resolvable folds: events are routed across
    workers-as-GPUs by key hash (RCCL all-to-allv over xGMI), folded
    into HBM-resident open-address keyed window state by a fused HIP
    kernel, and emitted as `(key, window_start_ts, value)` record
    batches when the watermark closes each window.

    :arg mode: "count" or "sum" (sum folds the batch `vals` column).
    :arg wait: Watermark lateness allowance.
    :arg dedup: Enable wave-level duplicate aggregation (use for
        low-cardinality keys).
    :arg offset: Window stride for sliding windows (< length puts each
        event in multiple windows; single-pass insert path).

    Path selection guidance (measured on MI355X, 32M-event steps):
    high-cardinality keys (>= thousands) -> ``radix=True`` (32e9+
    events/s); low-cardinality keys (tens or fewer, e.g. the
    reference's 2-key benchmark_windowing) -> ``dedup=True`` (5.3e9 —
    32x over contended plain atomics); radix degrades gracefully but
    slowly when most keys hash into few table regions.
    :arg exchange: Force the RCCL exchange on/off; default: on iff
        torch.distributed is initialized with world > 1.
    """
    import threading

    import torch

    agg_mode = {"count": AGG_COUNT, "sum": AGG_SUM}[mode]
    align_ms = _ms(align_to)
    len_ms = int(length.total_seconds() * 1000)
    off_ms = int(offset.total_seconds() * 1000) if offset else len_ms
    wait_ms = int(wait.total_seconds() * 1000)
    sliding = off_ms < len_ms

    # Per-worker-thread runtime: the shard key is resolved LAZILY (the
    # process group may not exist yet while the user builds the flow),
    # and the rescale registry collects donor-shard rows (see
    # `_DeviceWindowLogic._apply_rescale`).
    _rt = threading.local()

    def _runtime():
        if not hasattr(_rt, "shard"):
            import torch.distributed as dist

            if dist.is_available() and dist.is_initialized():
                _rt.world = dist.get_world_size()
                _rt.shard = f"shard-{dist.get_rank()}"
            else:
                _rt.world = 1
                _rt.shard = "shard-0"
            _rt.rescale_rows = []
        return _rt

    def shim_builder(resume_state):
        rt = _runtime()
        ex = exchange if exchange is not None else rt.world > 1
        if resume_state is not None:
            if resume_state.get("__consumed__"):
                return _ConsumedDonor()
            snap_world = resume_state.get("__world__", rt.world)
            snap_shard = resume_state.get("__shard__", rt.shard)
            if snap_world != rt.world:
                # State written under a different cluster shape: its
                # rows must re-exchange (key ownership changed).
                if snap_shard != rt.shard:
                    # Foreign shard name: a pure donor.
                    return _DonorLogic(resume_state, rt)
                # Same shard name as this worker's active key: the
                # engine will route live batches to THIS logic, so it
                # must be a real one — start it empty and donate the
                # old rows to the registry like any other donor.
                rt.rescale_rows.append(
                    {"snap": resume_state, "consumed": False}
                )
                resume_state = None
            elif snap_shard != rt.shard:
                return _DonorLogic(resume_state, rt)
        dev = torch.device(device)
        state = WindowAggState(
            dev,
            align_ms,
            len_ms,
            agg_mode,
            slots_pow=slots_pow,
            dedup=dedup and not sliding,
            out_cap=out_cap,
            radix=radix and dev.type != "cpu",
            region_bits=region_bits,
            off_ms=off_ms,
        )
        return _DeviceWindowLogic(
            state,
            wait_ms,
            ex,
            resume_state,
            shard=rt.shard,
            world=rt.world,
            registry=rt,
        )

    keyed = op.map("wrap", up, lambda b: (_runtime().shard, b))
    agg = op.stateful_batch("agg", keyed, shim_builder)
    return op.map("unwrap", agg, lambda kv: kv[1])


class _StrWindowLogic(StatefulBatchLogic):
    """Dictionary-encode str-keyed host batches into the HBM window
    table; decode ids back to strings at window close.

    Items are ``(strings, ts[, vals])`` host batches (strings: list of
    str or a packed ``(uint8 bytes, int64 offsets)`` pair; ts:
    epoch-ms ints).  Emissions are ``(key_str, win_start_ms, value)``
    tuples — the same rows the host windowing path produces for a
    keyed count/sum fold (reference windowing.py count_window /
    fold_window semantics under watermark close).
    """

    def __init__(
        self, sdict, state, wait_ms: int, resume, exchange: bool = False
    ):
        self.sdict = sdict
        self.state = state
        self.wait_ms = wait_ms
        self.exchange = exchange
        if resume is not None:
            self.sdict.restore(resume["dict"])
            self.state.restore_from_host(resume["win"])

    def _emit(self, closed) -> List[tuple]:
        if closed is None:
            return []
        keys = self.sdict.decode(closed.keys)
        ts = closed.ts.cpu().tolist()
        vals = closed.vals.cpu().tolist()
        return list(zip(keys, ts, vals))

    def on_batch(self, batches):
        import torch

        out: List[tuple] = []
        for item in batches:
            strings, ts = item[0], item[1]
            vals = item[2] if len(item) > 2 else None
            dev = self.state.device
            ts_t = torch.as_tensor(ts, dtype=torch.int64).to(dev)
            vals_t = (
                torch.as_tensor(vals, dtype=torch.int64).to(dev)
                if vals is not None
                else None
            )
            if self.exchange:
                # Multi-GPU: route raw bytes by CONTENT hash so every
                # string is encoded at its owning rank (per-rank
                # dictionary ids never cross ranks).  Collective:
                # every rank feeds one batch per scheduling step.
                import numpy as np

                from .strings import exchange_str_by_key, pack_strings

                if isinstance(strings, tuple):
                    data, offs = strings
                else:
                    data, offs = pack_strings(strings)
                if not isinstance(data, torch.Tensor):
                    data = torch.from_numpy(
                        np.asarray(data, dtype=np.uint8)
                    )
                    offs = torch.from_numpy(
                        np.asarray(offs, dtype=np.int64)
                    )
                if dev.type != "cpu":
                    data, offs = data.to(dev), offs.to(dev)
                data, offs, ts_t, vals_t = exchange_str_by_key(
                    data, offs, ts_t, vals_t
                )
                ids = self.sdict.encode(
                    (data, offs)
                    if dev.type != "cpu"
                    else (data.numpy(), offs.numpy())
                )
            else:
                ids = self.sdict.encode(strings)
            self.state.insert(
                RecordBatch(ids.to(dev), ts_t, vals_t)
            )
        closed = self.state.close_due(self.wait_ms)
        out.extend(self._emit(closed))
        return (out, StatefulBatchLogic.RETAIN)

    def on_eof(self):
        return (
            self._emit(self.state.close_all()),
            StatefulBatchLogic.RETAIN,
        )

    def snapshot(self):
        return {
            "dict": self.sdict.snapshot(),
            "win": self.state.snapshot_to_host(),
        }


@operator
def keyed_window_agg_str(
    step_id: str,
    up: Stream,
    align_to: datetime,
    length: timedelta,
    mode: str = "count",
    wait: timedelta = timedelta(0),
    slots_pow: int = 20,
    dict_slots_pow: int = 21,
    out_cap: int = 1 << 20,
    device: str = "cuda",
    exchange: bool = False,
) -> Stream:
    """Str-keyed tumbling-window aggregation on GPU.

    The string-keyed twin of :func:`keyed_window_agg`: the reference's
    key contract is `str` (reference src/operators.rs:363-439), so
    this lowers str-keyed streams onto the columnar kernels through a
    device string dictionary (:class:`bytewax_amd.gpu.strings.
    StringDict`) — hash/dedupe on device, id->str decode on close.

    Upstream items: ``(strings, ts[, vals])`` host batches.  Output:
    ``(key_str, win_start_ms, value)`` tuples at watermark close,
    matching the host windowing path's rows exactly (gpu test
    `tests/test_gpu_strings.py`).

    With ``exchange=True`` (multi-GPU) raw string bytes are routed to
    their owning rank by content hash before encoding
    (:func:`bytewax_amd.gpu.strings.exchange_str_by_key`), so each
    rank's dictionary only ever holds the strings it owns.
    Collective: every rank must feed one batch per scheduling step.
    Snapshots restore at the SAME world size (ownership is
    ``hash % world``; resuming at a different world would strand
    state at its old owner — rescale through the host path).
    """
    import torch

    from .strings import StringDict

    agg_mode = {"count": AGG_COUNT, "sum": AGG_SUM}[mode]
    align_ms = _ms(align_to)
    len_ms = int(length.total_seconds() * 1000)
    wait_ms = int(wait.total_seconds() * 1000)

    def shim_builder(resume_state):
        dev = torch.device(device)
        sdict = StringDict(dev, slots_pow=dict_slots_pow)
        state = WindowAggState(
            dev, align_ms, len_ms, agg_mode,
            slots_pow=slots_pow, out_cap=out_cap,
            radix=dev.type != "cpu",
        )
        return _StrWindowLogic(
            sdict, state, wait_ms, resume_state, exchange=exchange
        )

    keyed = op.map("wrap", up, lambda b: ("shard-0", b))
    agg = op.stateful_batch("agg", keyed, shim_builder)
    return op.map("unwrap", agg, lambda kv: kv[1])


_STATS_COLS = ("keys", "wins", "cnt", "sum", "min", "max")


class _DeviceStatsLogic(StatefulBatchLogic):
    """Holds the HBM stats table (1BRC-style count/sum/min/max)."""

    def __init__(
        self,
        state,
        wait_ms: int,
        exchange: bool,
        resume,
        shard: Optional[str] = None,
        world: int = 1,
        registry: Optional[Any] = None,
    ):
        self.state = state
        self.wait_ms = wait_ms
        self.exchange = exchange
        self._shard = shard
        self._world = world
        self._registry = registry
        self._applied_rescale = registry is None
        if resume is not None:
            resume = dict(resume)
            resume.pop("__world__", None)
            resume.pop("__shard__", None)
            self.state.restore_from_host(resume)

    def _apply_rescale(self) -> None:
        """Merge donor-shard stats rows after a worker-count change.

        Stats cells carry four accumulators, so instead of the
        columnar wire exchange this startup-only path all-gathers the
        (small) spills as host objects; each rank keeps the rows whose
        key-hash it now owns and merges them (count/sum add, min/max
        observe — duplicate-safe).  Collective: with the exchange on,
        EVERY rank calls the all_gather exactly once."""
        if self._applied_rescale:
            return
        self._applied_rescale = True
        import numpy as np
        import torch

        from . import _mix64_torch

        rows = []
        max_ts = None
        for entry in self._registry.rescale_rows:
            if entry["consumed"]:
                continue
            entry["consumed"] = True
            snap = entry["snap"]
            if len(snap.get("keys", ())):
                rows.append(
                    {c: np.asarray(snap[c]) for c in _STATS_COLS}
                )
            mts = snap.get("max_ts")
            if mts is not None:
                max_ts = mts if max_ts is None else max(max_ts, mts)
        local = (
            {
                c: np.concatenate([r[c] for r in rows])
                for c in _STATS_COLS
            }
            if rows
            else None
        )
        if self.exchange:
            import torch.distributed as dist

            world = dist.get_world_size()
            rank = dist.get_rank()
            gathered = [None] * world
            dist.all_gather_object(gathered, local)
            parts = []
            for g in gathered:
                if g is None:
                    continue
                keys_t = torch.as_tensor(g["keys"]).to(torch.int64)
                mine = (
                    torch.remainder(_mix64_torch(keys_t), world) == rank
                ).numpy()
                if mine.any():
                    parts.append({c: g[c][mine] for c in _STATS_COLS})
            merged = (
                {
                    c: np.concatenate([p[c] for p in parts])
                    for c in _STATS_COLS
                }
                if parts
                else None
            )
        else:
            merged = local
        if merged is not None:
            self.state.merge_rows(merged)
        if (
            max_ts is not None
            and max_ts > self.state.max_ts_host
        ):
            self.state.max_ts_host = max_ts

    def on_batch(self, batches):
        self._apply_rescale()
        out = []
        for batch in batches:
            if self.exchange:
                batch = exchange_by_key(batch)
            self.state.insert(batch)
        wm = self.state.watermark_ms() if hasattr(self.state, "watermark_ms") else self.state.max_ts_host
        horizon = (wm - self.wait_ms - self.state.align_ms) // self.state.len_ms
        if horizon > self.state.closed_horizon:
            closed = self.state.extract(horizon, clear=True)
            self.state.closed_horizon = horizon
            if closed is not None:
                out.append(closed)
        return (out, StatefulBatchLogic.RETAIN)

    def on_eof(self):
        self._apply_rescale()
        closed = self.state.extract(None, clear=True)
        return (
            [closed] if closed is not None else [],
            StatefulBatchLogic.RETAIN,
        )

    def snapshot(self):
        snap = dict(self.state.snapshot_to_host())
        if self._shard is not None:
            snap["__shard__"] = self._shard
            snap["__world__"] = self._world
        return snap


@operator
def keyed_stats_agg(
    step_id: str,
    up: Stream[RecordBatch],
    align_to: datetime,
    length: timedelta,
    wait: timedelta = timedelta(0),
    slots_pow: int = 20,
    out_cap: int = 1 << 20,
    device: str = "cuda",
    exchange: Optional[bool] = None,
) -> Stream[Dict[str, Any]]:
    """Keyed windowed count/sum/min/max (1BRC-style) over columnar
    batches on GPU; one fused pass per batch.

    Use a very long `length` for an unwindowed whole-stream
    aggregation.  Emits dicts of columnar device tensors
    {keys, wins, cnt, sum, min, max} at window close / EOF.
    """
    import threading

    import torch

    from .state import StatsAggState
    from . import _ms as to_ms

    align_ms = to_ms(align_to)
    len_ms = int(length.total_seconds() * 1000)
    wait_ms = int(wait.total_seconds() * 1000)

    _rt = threading.local()

    def _runtime():
        if not hasattr(_rt, "shard"):
            import torch.distributed as dist

            if dist.is_available() and dist.is_initialized():
                _rt.world = dist.get_world_size()
                _rt.shard = f"shard-{dist.get_rank()}"
            else:
                _rt.world = 1
                _rt.shard = "shard-0"
            _rt.rescale_rows = []
        return _rt

    def shim_builder(resume_state):
        rt = _runtime()
        ex = exchange if exchange is not None else rt.world > 1
        if resume_state is not None:
            if resume_state.get("__consumed__"):
                return _ConsumedDonor()
            snap_world = resume_state.get("__world__", rt.world)
            snap_shard = resume_state.get("__shard__", rt.shard)
            if snap_world != rt.world:
                if snap_shard != rt.shard:
                    return _DonorLogic(resume_state, rt)
                rt.rescale_rows.append(
                    {"snap": resume_state, "consumed": False}
                )
                resume_state = None
            elif snap_shard != rt.shard:
                return _DonorLogic(resume_state, rt)
        state = StatsAggState(
            torch.device(device), align_ms, len_ms,
            slots_pow=slots_pow, out_cap=out_cap,
        )
        return _DeviceStatsLogic(
            state,
            wait_ms,
            ex,
            resume_state,
            shard=rt.shard,
            world=rt.world,
            registry=rt,
        )

    keyed = op.map("wrap", up, lambda b: (_runtime().shard, b))
    agg = op.stateful_batch("agg", keyed, shim_builder)
    return op.map("unwrap", agg, lambda kv: kv[1])


class _DeviceJoinLogic(StatefulBatchLogic):
    """Holds the HBM join table; values are (side, RecordBatch)."""

    def __init__(self, state, exchange: bool, resume=None):
        self.state = state
        self.exchange = exchange
        if resume is not None:
            self.state.restore_from_host(resume)

    def on_batch(self, side_batches):
        for side, batch in side_batches:
            if self.exchange:
                batch = exchange_by_key(batch)
            self.state.insert(side, batch.keys, batch.vals)
        joined = self.state.take_joined()
        if joined is None:
            return ([], StatefulBatchLogic.RETAIN)
        keys, v0, v1 = joined
        return (
            [RecordBatch(keys, v0, v1)],
            StatefulBatchLogic.RETAIN,
        )

    def snapshot(self):
        # Live single-side cells spill to host; completed pairs were
        # already emitted and reset, so nothing re-emits on resume.
        return self.state.snapshot_to_host()

    def on_eof(self):
        return ([], StatefulBatchLogic.RETAIN)


@operator
def stream_join(
    step_id: str,
    left: Stream[RecordBatch],
    right: Stream[RecordBatch],
    slots_pow: int = 20,
    out_cap: int = 1 << 20,
    device: str = "cuda",
    exchange: Optional[bool] = None,
) -> Stream[RecordBatch]:
    """Stream-stream hash join over columnar batches on GPU
    ("last" insert / "complete" emit).

    Emits RecordBatches whose `ts` column holds the left value and
    `vals` the right value for each completed key pair.

    Concurrency semantic: within one batch, duplicate keys on the same
    side race — which duplicate's value lands (and whether a duplicate
    arriving during a completion's reset survives as a fresh presence)
    is unspecified, mirroring the reference's unspecified cross-worker
    arrival order.  Pre-deduplicate per batch if you need the serial
    "last" semantics exactly.

    Engine: by default each side's batch is radix-partitioned into
    table-region segments, collapsed to distinct keys in LDS, and
    merged once per key (20.5e9 events/s at 1M keys; see profiles/).
    ``BYTEWAX_JOIN_RADIX=0`` restores the direct-atomic path.
    """
    import torch

    from .state import HashJoinState

    def shim_builder(resume_state):
        import torch.distributed as dist

        ex = exchange
        if ex is None:
            ex = (
                dist.is_available()
                and dist.is_initialized()
                and dist.get_world_size() > 1
            )
        return _DeviceJoinLogic(
            HashJoinState(
                torch.device(device), slots_pow=slots_pow, out_cap=out_cap
            ),
            ex,
            resume_state,
        )

    import torch.distributed as dist

    shard = (
        f"shard-{dist.get_rank()}"
        if dist.is_available() and dist.is_initialized()
        else "shard-0"
    )
    l_labeled = op.map("wrap_l", left, lambda b: (shard, (0, b)))
    r_labeled = op.map("wrap_r", right, lambda b: (shard, (1, b)))
    merged = op.merge("merge", l_labeled, r_labeled)
    joined = op.stateful_batch("join", merged, shim_builder)
    return op.map("unwrap", joined, lambda kv: kv[1])


@operator
def map_batch(
    step_id: str,
    up: Stream[RecordBatch],
    mapper,
) -> Stream[RecordBatch]:
    """Transform RecordBatches one-to-one with device tensor ops.

    ``mapper(batch) -> RecordBatch`` runs on the GPU via torch ops (or
    custom kernels); this is the columnar twin of ``op.map``.
    """
    return op.flat_map_batch(
        "flat_map_batch", up, lambda bs: [mapper(b) for b in bs]
    )


@operator
def filter_batch(
    step_id: str,
    up: Stream[RecordBatch],
    predicate,
) -> Stream[RecordBatch]:
    """Keep only events where ``predicate(batch)`` (a device bool mask
    over the batch) is true; compaction runs as a wave-ballot HIP
    kernel — the columnar twin of ``op.filter``.
    """

    def shim(batches):
        import torch

        from ._ext import ext

        out = []
        for b in batches:
            m = predicate(b)
            if b.keys.device.type == "cpu":
                idx = m.nonzero(as_tuple=True)[0]
                out.append(
                    RecordBatch(
                        b.keys[idx],
                        b.ts[idx],
                        b.vals[idx] if b.vals is not None else None,
                        max_ts=b.max_ts,
                        ts_base=b.ts_base,
                    )
                )
                continue
            k = ext()
            if b.ts.dtype != torch.int64:
                b = RecordBatch(
                    b.keys, b.ts.to(torch.int64), b.vals,
                    max_ts=b.max_ts, ts_base=b.ts_base,
                )
            n = len(b)
            dev = b.keys.device
            out_keys = torch.empty(n, dtype=torch.int32, device=dev)
            out_ts = torch.empty(n, dtype=torch.int64, device=dev)
            out_vals = torch.empty(
                n if b.vals is not None else 0,
                dtype=torch.int64,
                device=dev,
            )
            out_n = torch.zeros(1, dtype=torch.int32, device=dev)
            k.filter_compact(
                b.keys, b.ts, b.vals, m.to(torch.uint8), out_keys, out_ts,
                out_vals, out_n,
            )
            kept = int(out_n.item())
            out.append(
                RecordBatch(
                    out_keys[:kept],
                    out_ts[:kept],
                    out_vals[:kept] if b.vals is not None else None,
                    max_ts=b.max_ts,
                    ts_base=b.ts_base,
                )
            )
        return out

    return op.flat_map_batch("flat_map_batch", up, shim)


class _CollectCountsPartition(StatelessSinkPartition[RecordBatch]):
    def __init__(self, ls: List):
        self._ls = ls

    def write_batch(self, items: List[RecordBatch]) -> None:
        for b in items:
            self._ls.append(b)


class CollectCountsSink(DynamicSink[RecordBatch]):
    """Collect closed-window RecordBatches into a list (testing/bench).

    Batches stay on device; callers decide when (and whether) to copy
    to host.
    """

    def __init__(self, ls: List):
        self._ls = ls

    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> _CollectCountsPartition:
        return _CollectCountsPartition(self._ls)


class _DeviceSessionLogic(StatefulBatchLogic):
    """Holds the HBM session table (gap-based windows)."""

    def __init__(self, state, wait_ms: int, exchange: bool, resume):
        self.state = state
        self.wait_ms = wait_ms
        self.exchange = exchange
        if resume is not None:
            self.state.restore_from_host(resume)

    def on_batch(self, batches):
        out = []
        for batch in batches:
            if self.exchange:
                batch = exchange_by_key(batch)
            self.state.insert(batch)
        closed = self.state.close_due(self.wait_ms)
        if closed is not None:
            out.append(closed)
        return (out, StatefulBatchLogic.RETAIN)

    def on_eof(self):
        closed = self.state.close_all()
        return (
            [closed] if closed is not None else [],
            StatefulBatchLogic.RETAIN,
        )

    def snapshot(self):
        return self.state.snapshot_to_host()


@operator
def keyed_session_agg(
    step_id: str,
    up: Stream[RecordBatch],
    gap: timedelta,
    mode: str = "count",
    wait: timedelta = timedelta(0),
    slots_pow: int = 20,
    out_cap: int = 1 << 20,
    device: str = "cuda",
    exchange: Optional[bool] = None,
) -> Stream[Dict[str, Any]]:
    """Keyed session windows over columnar batches on GPU.

    Sessions close when a key sees no events for ``gap`` (reference
    `SessionWindower`, windowing.py).  Batches are sorted by
    (key, ts) on device and one thread walks each key's run in event
    order, so parallelism equals the batch's distinct-key count —
    use the host path's `SessionWindower` for low-cardinality
    streams.  Ingestion must be watermark-ordered across batches
    (each batch's events at or after the previous watermark), which
    in-order sources provide; under that ordering the reference's
    session merges degenerate to extensions.

    Emits dicts of device columns {keys, start, end, vals} at session
    close / EOF.
    """
    import torch

    from .state import SessionAggState

    agg_mode = {"count": AGG_COUNT, "sum": AGG_SUM}[mode]
    gap_ms = int(gap.total_seconds() * 1000)
    wait_ms = int(wait.total_seconds() * 1000)

    def shim_builder(resume_state):
        import torch.distributed as dist

        ex = exchange
        if ex is None:
            ex = (
                dist.is_available()
                and dist.is_initialized()
                and dist.get_world_size() > 1
            )
        state = SessionAggState(
            torch.device(device), gap_ms, agg_mode,
            slots_pow=slots_pow, out_cap=out_cap,
        )
        return _DeviceSessionLogic(state, wait_ms, ex, resume_state)

    import torch.distributed as dist

    shard = (
        f"shard-{dist.get_rank()}"
        if dist.is_available() and dist.is_initialized()
        else "shard-0"
    )
    keyed = op.map("wrap", up, lambda b: (shard, b))
    agg = op.stateful_batch("agg", keyed, shim_builder)
    return op.map("unwrap", agg, lambda kv: kv[1])
