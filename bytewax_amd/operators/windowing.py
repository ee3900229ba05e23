"""Time-based windowing operators.

API parity target: ``bytewax.operators.windowing`` (reference
pysrc/bytewax/operators/windowing.py) — clocks (system / event-time
with watermark), windowers (tumbling / sliding / session with merges),
window logics, and the derived operators (`collect_window`,
`count_window`, `fold_window`, `join_window`, `max_window`,
`min_window`, `reduce_window`).

The composition is the reference's: a `window` operator is a
`stateful_batch` whose per-key logic composes a `ClockLogic` (watermark
tracking), a `WindowerLogic` (window assignment / merge / close) and
per-window `WindowLogic` instances, plus three unwrapping
`filter_map_value` steps producing the `down` / `late` / `meta`
streams of :class:`WindowOut`.

On the GPU engine the tumbling/sliding fold path for device-resolvable
folders is replaced by HIP segmented-hash aggregation kernels (see
:mod:`bytewax_amd.gpu`); this module is the semantic reference for
those kernels and the host path for arbitrary Python logic.
"""

import copy
import typing
from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from functools import partial
from typing import (
    Any,
    Callable,
    Dict,
    Generic,
    Iterable,
    List,
    Optional,
    Set,
    Tuple,
    TypeVar,
)

import bytewax_amd.operators as op
from ..dataflow import KeyedStream, Stream, operator
from . import (
    JoinEmitMode,
    JoinInsertMode,
    StatefulBatchLogic,
    _identity,
    _JoinState,
)

V = TypeVar("V")
W = TypeVar("W")
W_co = TypeVar("W_co", covariant=True)
X = TypeVar("X")
S = TypeVar("S")
SC = TypeVar("SC")
SW = TypeVar("SW")

ZERO_TD = timedelta(seconds=0)
UTC_MIN = datetime.min.replace(tzinfo=timezone.utc)
"""Minimum possible UTC date time."""
UTC_MAX = datetime.max.replace(tzinfo=timezone.utc)

_EPOCH_UTC = datetime(1970, 1, 1, tzinfo=timezone.utc)
"""Maximum possible UTC date time."""

LATE_SESSION_ID: int = -1
"""Sentinel window ID assigned to late items by session windowers."""

_EMPTY: Tuple = ()

__all__ = [
    "Clock",
    "ClockLogic",
    "EventClock",
    "LATE_SESSION_ID",
    "SessionWindower",
    "SlidingWindower",
    "SystemClock",
    "TumblingWindower",
    "UTC_MAX",
    "UTC_MIN",
    "WindowLogic",
    "WindowMetadata",
    "WindowOut",
    "Windower",
    "WindowerLogic",
    "COLUMNAR_WINDOW_ID",
    "COUNT_FOLD",
    "DeviceFoldable",
    "SUM_FOLD",
    "collect_window",
    "count_window",
    "device_count",
    "device_max",
    "device_mean",
    "device_min",
    "device_sum",
    "fold_window",
    "join_window",
    "max_window",
    "min_window",
    "reduce_window",
    "window",
]


def _get_system_utc() -> datetime:
    return datetime.now(timezone.utc)


# ---------------------------------------------------------------------------
# Clocks
# ---------------------------------------------------------------------------


class ClockLogic(ABC, Generic[V, S]):
    """Abstract class to define a sense of time for windowing.

    Instantiated for each key which is encountered.  See the concrete
    subclasses of {py:obj}`Clock` for built-in options.
    """

    @abstractmethod
    def before_batch(self) -> None:
        """Prepare to process items incoming simultaneously.

        Called once before a series of {py:obj}`on_item` calls; use it
        to cache a "current time".
        """
        ...

    @abstractmethod
    def on_item(self, value: V) -> Tuple[datetime, datetime]:
        """Called on each new upstream item.

        :returns: A 2-tuple of (item timestamp, current watermark).
        """
        ...

    @abstractmethod
    def on_notify(self) -> datetime:
        """Get the current watermark when there are no items."""
        ...

    @abstractmethod
    def on_eof(self) -> datetime:
        """Get the watermark once the upstream is EOF."""
        ...

    @abstractmethod
    def to_system_utc(self, timestamp: datetime) -> Optional[datetime]:
        """Convert a timestamp to a UTC system time for wake-ups."""
        ...

    @abstractmethod
    def snapshot(self) -> S:
        """Return an immutable copy of the state for recovery."""
        ...


class Clock(ABC, Generic[V, S]):
    """Abstract class defining a type of clock.

    Every subclass must have a matching {py:obj}`ClockLogic`.
    """

    @abstractmethod
    def build(self, resume_state: Optional[S]) -> ClockLogic[V, S]:
        """Construct a new clock logic instance."""
        ...


@dataclass
class _SystemClockLogic(ClockLogic[Any, None]):
    now_getter: Callable[[], datetime]

    def __post_init__(self) -> None:
        self._now = self.now_getter()

    def before_batch(self) -> None:
        self._now = self.now_getter()

    def on_item(self, value: Any) -> Tuple[datetime, datetime]:
        return (self._now, self._now)

    def on_notify(self) -> datetime:
        self._now = self.now_getter()
        return self._now

    def on_eof(self) -> datetime:
        return UTC_MAX

    def to_system_utc(self, timestamp: datetime) -> Optional[datetime]:
        return timestamp

    def snapshot(self) -> None:
        return None


@dataclass
class SystemClock(Clock[Any, None]):
    """Uses the current system time as the timestamp for each item.

    The watermark is the current system time; when the dataflow has no
    more input, all windows are closed.
    """

    def build(self, resume_state: None) -> _SystemClockLogic:
        return _SystemClockLogic(_get_system_utc)


@dataclass
class _EventClockState:
    system_time_of_max_event: datetime
    watermark_base: datetime


class _EventClockLogic(ClockLogic[V, _EventClockState]):
    """Watermark = max event ts seen − wait duration + system time
    elapsed since that max was seen; never regresses even if the
    system clock does."""

    def __init__(
        self,
        now_getter: Callable[[], datetime],
        ts_getter: Callable[[V], datetime],
        to_system: Callable[[datetime], Optional[datetime]],
        wait_for_system_duration: timedelta,
        state: Optional[_EventClockState] = None,
    ):
        self.now_getter = now_getter
        self.ts_getter = ts_getter
        self.to_system = to_system
        self.wait = wait_for_system_duration
        self._system_now = now_getter()
        if state is None:
            state = _EventClockState(
                system_time_of_max_event=self._system_now,
                watermark_base=UTC_MIN,
            )
        self.state = state

    def before_batch(self) -> None:
        system_now = self.now_getter()
        # Never let "now" go backwards (NTP adjustments etc.); the
        # watermark holds steady until the clock catches up.
        if system_now > self._system_now:
            self._system_now = system_now

    def _watermark(self) -> datetime:
        return self.state.watermark_base + (
            self._system_now - self.state.system_time_of_max_event
        )

    def on_item(self, value: V) -> Tuple[datetime, datetime]:
        ts = self.ts_getter(value)
        watermark = self._watermark()
        try:
            base = ts - self.wait
            if base > watermark:
                self.state.watermark_base = base
                self.state.system_time_of_max_event = self._system_now
                return (ts, base)
        except OverflowError:
            # New watermark unrepresentable; keep advancing from the
            # old base so the watermark does not regress.
            pass
        return (ts, watermark)

    def on_notify(self) -> datetime:
        self.before_batch()
        return self._watermark()

    def on_eof(self) -> datetime:
        return UTC_MAX

    def to_system_utc(self, timestamp: datetime) -> Optional[datetime]:
        return self.to_system(timestamp)

    def snapshot(self) -> _EventClockState:
        return copy.deepcopy(self.state)


@dataclass
class EventClock(Clock[V, _EventClockState]):
    """Use a timestamp embedded within each item.

    The watermark is the largest timestamp seen thus far, minus the
    waiting duration, plus the system time elapsed since it was seen.

    :arg ts_getter: Called once on each item to get its (aware, UTC)
        timestamp.
    :arg wait_for_system_duration: How much system time to wait after
        seeing a timestamp for the watermark to catch up to it.
    :arg now_getter: Return the current "system" timestamp; defaults
        to UTC system time.
    :arg to_system_utc: Map a window-close timestamp to the UTC system
        time to wake up at; `None` disables timed wake-ups.
    """

    ts_getter: Callable[[V], datetime]
    wait_for_system_duration: timedelta
    now_getter: Callable[[], datetime] = _get_system_utc
    to_system_utc: Callable[[datetime], Optional[datetime]] = _identity

    def build(self, resume_state: Optional[_EventClockState]) -> _EventClockLogic[V]:
        return _EventClockLogic(
            self.now_getter,
            self.ts_getter,
            self.to_system_utc,
            self.wait_for_system_duration,
            resume_state,
        )


# ---------------------------------------------------------------------------
# Windowers
# ---------------------------------------------------------------------------


@dataclass
class WindowMetadata:
    """Metadata about a window.

    Exact semantics depend on the windower: for
    {py:obj}`SessionWindower` `close_time` is inclusive; for
    {py:obj}`SlidingWindower` it is not.
    """

    open_time: datetime
    """The timestamp this window opened."""
    close_time: datetime
    """The timestamp this window closed."""
    merged_ids: Set[int] = field(default_factory=set)
    """Any original window IDs merged into this window before close."""


class WindowerLogic(ABC, Generic[S]):
    """Abstract class which defines a type of window.

    Instantiated for each key which is encountered.
    """

    @abstractmethod
    def open_for(self, timestamp: datetime) -> Iterable[int]:
        """Find which windows an item is in and mark them as open."""
        ...

    @abstractmethod
    def late_for(self, timestamp: datetime) -> Iterable[int]:
        """Find which windows an item would have been in, if on-time."""
        ...

    @abstractmethod
    def merged(self) -> Iterable[Tuple[int, int]]:
        """Report any `(original_window_id, target_window_id)` merges
        caused by the last batch of items."""
        ...

    @abstractmethod
    def close_for(self, watermark: datetime) -> Iterable[Tuple[int, WindowMetadata]]:
        """Report windows now closed by the watermark, with their
        final metadata."""
        ...

    @abstractmethod
    def notify_at(self) -> Optional[datetime]:
        """Next time at which a window might close."""
        ...

    @abstractmethod
    def is_empty(self) -> bool:
        """Whether no state needs to be maintained anymore."""
        ...

    @abstractmethod
    def snapshot(self) -> S:
        """Return an immutable copy of the state for recovery."""
        ...


class Windower(ABC, Generic[S]):
    """A type of window; every subclass must have a matching
    {py:obj}`WindowerLogic`."""

    @abstractmethod
    def build(self, resume_state: Optional[S]) -> WindowerLogic[S]:
        """Construct a new windower logic instance."""
        ...


@dataclass
class _SlidingWindowerState:
    opened: Dict[int, WindowMetadata] = field(default_factory=dict)


class _SlidingWindowerLogic(WindowerLogic[_SlidingWindowerState]):
    """Window IDs are integers counting `offset` strides from
    `align_to`; an item intersects every window whose `[open, open +
    length)` span contains its timestamp."""

    def __init__(
        self,
        length: timedelta,
        offset: timedelta,
        align_to: datetime,
        state: _SlidingWindowerState,
    ):
        self.length = length
        self.offset = offset
        self.align_to = align_to
        self.state = state

    def intersects(self, timestamp: datetime) -> List[int]:
        since = timestamp - self.align_to
        if self.length == self.offset:
            # Tumbling fast path: exactly one window, one floordiv.
            return [since // self.offset]
        lo = (since - self.length) // self.offset + 1
        hi = since // self.offset + 1
        return list(range(lo, hi))

    def _metadata_for(self, window_id: int) -> WindowMetadata:
        open_time = self.align_to + self.offset * window_id
        return WindowMetadata(open_time, open_time + self.length)

    def open_for(self, timestamp: datetime) -> List[int]:
        found = self.intersects(timestamp)
        opened = self.state.opened
        for window_id in found:
            # Metadata built lazily: `setdefault` would construct it
            # per ITEM even for an already-open window.
            if window_id not in opened:
                opened[window_id] = self._metadata_for(window_id)
        return found

    def late_for(self, timestamp: datetime) -> List[int]:
        return self.intersects(timestamp)

    def merged(self) -> Iterable[Tuple[int, int]]:
        return _EMPTY

    def close_for(self, watermark: datetime) -> Iterable[Tuple[int, WindowMetadata]]:
        closed = [
            (window_id, meta)
            for window_id, meta in self.state.opened.items()
            if meta.close_time <= watermark
        ]
        for window_id, _meta in closed:
            del self.state.opened[window_id]
        return closed

    def notify_at(self) -> Optional[datetime]:
        return min(
            (meta.close_time for meta in self.state.opened.values()),
            default=None,
        )

    def is_empty(self) -> bool:
        return len(self.state.opened) <= 0

    def snapshot(self) -> _SlidingWindowerState:
        return copy.deepcopy(self.state)


@dataclass
class SlidingWindower(Windower[_SlidingWindowerState]):
    """Sliding windows of fixed duration.

    If `offset == length` windows tumble (never overlap); `offset >
    length` is forbidden (items would fall in gaps).  Window open
    times are inclusive, close times exclusive.

    :arg length: Length of windows.
    :arg offset: Duration between start times of adjacent windows.
    :arg align_to: Align windows so this instant starts one.
    """

    length: timedelta
    offset: timedelta
    align_to: datetime

    def __post_init__(self):
        if self.offset > self.length:
            msg = (
                "sliding window `offset` can't be longer than `length`; "
                "there would be undefined gaps between windows"
            )
            raise ValueError(msg)

    def build(
        self, resume_state: Optional[_SlidingWindowerState]
    ) -> _SlidingWindowerLogic:
        state = resume_state if resume_state is not None else _SlidingWindowerState()
        return _SlidingWindowerLogic(self.length, self.offset, self.align_to, state)


@dataclass
class TumblingWindower(Windower[_SlidingWindowerState]):
    """Tumbling windows of fixed duration; each item falls in exactly
    one window.  Open times inclusive, close times exclusive.

    :arg length: Length of windows.
    :arg align_to: Align windows so this instant starts one.
    """

    length: timedelta
    align_to: datetime

    def build(
        self, resume_state: Optional[_SlidingWindowerState]
    ) -> _SlidingWindowerLogic:
        state = resume_state if resume_state is not None else _SlidingWindowerState()
        return _SlidingWindowerLogic(self.length, self.length, self.align_to, state)


@dataclass
class _SessionWindowerState:
    max_key: int = LATE_SESSION_ID
    sessions: Dict[int, WindowMetadata] = field(default_factory=dict)
    merge_queue: List[Tuple[int, int]] = field(default_factory=list)


def _session_find_merges(
    sessions: Dict[int, WindowMetadata], gap: timedelta
) -> List[Tuple[int, int]]:
    """Merge any sessions now within `gap` of each other; mutates
    `sessions`, returns `(original_id, target_id)` pairs."""
    merges: List[Tuple[int, int]] = []
    by_open = sorted(sessions.items(), key=lambda kv: kv[1].open_time)
    last_id, last_meta = by_open[0]
    for this_id, this_meta in by_open[1:]:
        if this_meta.open_time - last_meta.close_time <= gap:
            last_meta.close_time = max(last_meta.close_time, this_meta.close_time)
            last_meta.merged_ids.add(this_id)
            merges.append((this_id, last_id))
            del sessions[this_id]
        else:
            last_id, last_meta = this_id, this_meta
    return merges


class _SessionWindowerLogic(WindowerLogic[_SessionWindowerState]):
    def __init__(self, gap: timedelta, state: _SessionWindowerState):
        self.gap = gap
        self.state = state

    def _find_merges(self) -> None:
        if len(self.state.sessions) >= 2:
            self.state.merge_queue.extend(
                _session_find_merges(self.state.sessions, self.gap)
            )

    def open_for(self, timestamp: datetime) -> Iterable[int]:
        for window_id, meta in self.state.sessions.items():
            until_open = meta.open_time - timestamp
            since_close = timestamp - meta.close_time
            if until_open <= ZERO_TD and since_close <= ZERO_TD:
                # Perfectly within an existing session: no boundary
                # change, no merges possible.
                return (window_id,)
            elif ZERO_TD < until_open <= self.gap:
                meta.open_time = timestamp
                self._find_merges()
                return (window_id,)
            elif ZERO_TD < since_close <= self.gap:
                meta.close_time = timestamp
                self._find_merges()
                return (window_id,)
        self.state.max_key += 1
        window_id = self.state.max_key
        self.state.sessions[window_id] = WindowMetadata(timestamp, timestamp)
        return (window_id,)

    def late_for(self, timestamp: datetime) -> Iterable[int]:
        return (LATE_SESSION_ID,)

    def merged(self) -> Iterable[Tuple[int, int]]:
        merged = self.state.merge_queue
        self.state.merge_queue = []
        return merged

    def close_for(self, watermark: datetime) -> Iterable[Tuple[int, WindowMetadata]]:
        try:
            close_after = watermark - self.gap
        except OverflowError:
            close_after = UTC_MIN
        closed = [
            (window_id, meta)
            for window_id, meta in self.state.sessions.items()
            if meta.close_time < close_after
        ]
        for window_id, _meta in closed:
            del self.state.sessions[window_id]
        return closed

    def notify_at(self) -> Optional[datetime]:
        min_close = min(
            (meta.close_time for meta in self.state.sessions.values()),
            default=None,
        )
        return min_close + self.gap if min_close is not None else None

    def is_empty(self) -> bool:
        # Never discard: re-using a window ID could give a downstream
        # join incorrect window metadata.
        return False

    def snapshot(self) -> _SessionWindowerState:
        return copy.deepcopy(self.state)


@dataclass
class SessionWindower(Windower[_SessionWindowerState]):
    """Session windows with a fixed inactivity gap.

    :arg gap: Gap of inactivity before considering a session closed;
        must not be negative.
    """

    gap: timedelta

    def __post_init__(self):
        if self.gap < ZERO_TD:
            msg = "session window `gap` must not be negative"
            raise ValueError(msg)

    def build(
        self, resume_state: Optional[_SessionWindowerState]
    ) -> _SessionWindowerLogic:
        state = resume_state if resume_state is not None else _SessionWindowerState()
        return _SessionWindowerLogic(self.gap, state)


# ---------------------------------------------------------------------------
# Window logic composition
# ---------------------------------------------------------------------------


class WindowLogic(ABC, Generic[V, W, S]):
    """Abstract class to define a {py:obj}`window` operator.

    A unique instance is created for each window within each key.
    """

    @abstractmethod
    def on_value(self, value: V) -> Iterable[W]:
        """Called on each new upstream item within this window, in
        timestamp order (if `ordered`)."""
        ...

    @abstractmethod
    def on_merge(self, original: "WindowLogic[V, W, S]") -> Iterable[W]:
        """Called when two windows merge; consume the original's
        state."""
        ...

    @abstractmethod
    def on_close(self) -> Iterable[W]:
        """Called when this window closes."""
        ...

    @abstractmethod
    def snapshot(self) -> S:
        """Return an immutable copy of the state for recovery."""
        ...


@dataclass(frozen=True)
class _WindowSnapshot:
    clock_state: Any
    windower_state: Any
    logic_states: Dict[int, Any]
    queue: List[Tuple[Any, datetime]]
    last_watermark: datetime


class _WindowLogic(StatefulBatchLogic):
    """Composes clock + windower + per-window logics.

    Events emitted downstream are `(window_id, tag, payload)` where tag
    is "E" (emit), "L" (late item) or "M" (window metadata) — unwrapped
    by the `window` operator into the three output streams.
    """

    def __init__(
        self,
        clock: ClockLogic,
        windower: WindowerLogic,
        builder: Callable[[Optional[Any]], WindowLogic],
        ordered: bool,
        logics: Optional[Dict[int, WindowLogic]] = None,
        queue: Optional[List[Tuple[Any, datetime]]] = None,
        last_watermark: datetime = UTC_MIN,
    ):
        self.clock = clock
        self.windower = windower
        self.builder = builder
        self.ordered = ordered
        self.logics = logics if logics is not None else {}
        self.queue = queue if queue is not None else []
        self._last_watermark = last_watermark

    def _handle_inserts(self, due: List[Tuple[Any, datetime]]) -> Iterable:
        for value, timestamp in due:
            for window_id in self.windower.open_for(timestamp):
                logic = self.logics.get(window_id)
                if logic is None:
                    logic = self.builder(None)
                    self.logics[window_id] = logic
                for w in logic.on_value(value):
                    yield (window_id, "E", w)

    def _handle_merged(self) -> Iterable:
        for orig_id, targ_id in self.windower.merged():
            if targ_id != orig_id:
                orig_logic = self.logics.pop(orig_id)
                into_logic = self.logics[targ_id]
                for w in into_logic.on_merge(orig_logic):
                    yield (targ_id, "E", w)

    def _handle_closed(self, watermark: datetime) -> Iterable:
        for window_id, meta in self.windower.close_for(watermark):
            logic = self.logics.pop(window_id)
            for w in logic.on_close():
                yield (window_id, "E", w)
            yield (window_id, "M", meta)

    def _flush_queue(self, watermark: datetime) -> Iterable:
        if self.ordered:
            due = [e for e in self.queue if e[1] <= watermark]
            self.queue = [e for e in self.queue if e[1] > watermark]
            due.sort(key=lambda e: e[1])
        else:
            due = self.queue
            self.queue = []
        yield from self._handle_inserts(due)
        yield from self._handle_merged()
        yield from self._handle_closed(watermark)

    def _is_empty(self) -> bool:
        return (
            len(self.logics) <= 0
            and len(self.queue) <= 0
            and self.windower.is_empty()
        )

    def on_batch(self, values: List[Any]) -> Tuple[Iterable, bool]:
        self.clock.before_batch()
        events: List = []
        watermark = self._last_watermark
        for value in values:
            ts, watermark = self.clock.on_item(value)
            assert watermark >= self._last_watermark
            self._last_watermark = watermark
            if ts < watermark:
                events.extend(
                    (window_id, "L", value)
                    for window_id in self.windower.late_for(ts)
                )
            else:
                self.queue.append((value, ts))
        events.extend(self._flush_queue(watermark))
        return (events, self._is_empty())

    def on_notify(self) -> Tuple[Iterable, bool]:
        watermark = self.clock.on_notify()
        assert watermark >= self._last_watermark
        self._last_watermark = watermark
        events = list(self._flush_queue(watermark))
        return (events, self._is_empty())

    def on_eof(self) -> Tuple[Iterable, bool]:
        watermark = self.clock.on_eof()
        assert watermark >= self._last_watermark
        self._last_watermark = watermark
        events = list(self._flush_queue(watermark))
        return (events, self._is_empty())

    def notify_at(self) -> Optional[datetime]:
        at = self.windower.notify_at()
        if self.ordered and self.queue:
            queue_at = min(ts for _v, ts in self.queue)
            at = queue_at if at is None else min(at, queue_at)
        if at is not None:
            at = self.clock.to_system_utc(at)
        return at

    def snapshot(self) -> _WindowSnapshot:
        return _WindowSnapshot(
            self.clock.snapshot(),
            self.windower.snapshot(),
            {wid: logic.snapshot() for wid, logic in self.logics.items()},
            list(self.queue),
            self._last_watermark,
        )


@dataclass(frozen=True)
class WindowOut(Generic[V, W_co]):
    """Streams returned from a windowing operator."""

    down: KeyedStream
    """Items emitted from this operator, sub-keyed by window ID."""
    late: KeyedStream
    """Upstream items that were deemed late, sub-keyed by window ID."""
    meta: KeyedStream
    """Metadata about closed windows, sub-keyed by window ID."""


#: Window-id placeholder on emissions from the columnar (GPU) lowering
#: of `fold_window`/`count_window`: one emission carries a whole
#: RecordBatch of closed `(key_id, window_start_ms, value)` rows, so a
#: single scalar window id does not apply.
COLUMNAR_WINDOW_ID = -2


class DeviceFoldable:
    """A fold recognized by the columnar (GPU) lowering.

    Behaves as a plain host folder callable, so one flow definition
    runs on both engines: over Python-object streams the host
    `_FoldWindowLogic` calls it per item; over
    :class:`bytewax_amd.gpu.RecordBatch` streams `fold_window`
    dispatches to the fused HIP window kernels instead
    (`WindowAggState`), chosen by stream type at runtime.
    """

    def __init__(
        self,
        mode: str,
        host_fn: Callable,
        finish: Optional[Callable] = None,
    ):
        self.mode = mode
        self._host_fn = host_fn
        #: Optional host-path finisher applied to the accumulator at
        #: window close (e.g. mean = sum / count); the columnar path
        #: computes the finished value in the extract kernel instead.
        self.finish = finish

    def __call__(self, acc, value):
        return self._host_fn(acc, value)

    def __repr__(self) -> str:  # pragma: no cover - cosmetic
        return f"DeviceFoldable({self.mode!r})"


#: Sum fold: host = `acc + value`; columnar = fused kernel sum of the
#: batch `vals` column.
SUM_FOLD = DeviceFoldable("sum", lambda acc, value: acc + value)
#: Count fold: host = `acc + 1`; columnar = fused kernel count.
COUNT_FOLD = DeviceFoldable("count", lambda acc, _value: acc + 1)


def device_sum(value_getter: Callable = _identity) -> DeviceFoldable:
    """A `fold_window` sum fold that lowers to the HIP kernels over
    RecordBatch streams; `value_getter` extracts the number from each
    host-path item (columnar batches use their `vals` column)."""
    return DeviceFoldable(
        "sum", lambda acc, value: acc + value_getter(value)
    )


def device_count() -> DeviceFoldable:
    """A `fold_window` count fold that lowers to the HIP kernels over
    RecordBatch streams."""
    return COUNT_FOLD


def device_min(value_getter: Callable = _identity) -> DeviceFoldable:
    """A `fold_window` min fold that lowers to the fused stats
    kernels over RecordBatch streams (tumbling windows); pair with a
    builder like ``lambda: float("inf")`` for the host path."""
    return DeviceFoldable(
        "min", lambda acc, value: min(acc, value_getter(value))
    )


def device_max(value_getter: Callable = _identity) -> DeviceFoldable:
    """A `fold_window` max fold that lowers to the fused stats
    kernels over RecordBatch streams (tumbling windows); pair with a
    builder like ``lambda: -float("inf")`` for the host path."""
    return DeviceFoldable(
        "max", lambda acc, value: max(acc, value_getter(value))
    )


def device_mean(value_getter: Callable = _identity) -> DeviceFoldable:
    """A `fold_window` mean fold that lowers to the fused stats
    kernels over RecordBatch streams (tumbling windows): the kernels
    keep (count, sum) per cell and the close emits sum / count.

    Host path: the accumulator is a ``(sum, count)`` pair — pair with
    a builder of ``lambda: (0, 0)``; the finisher divides at close so
    both paths emit the mean itself.
    """
    return DeviceFoldable(
        "mean",
        lambda acc, value: (acc[0] + value_getter(value), acc[1] + 1),
        finish=lambda acc: acc[0] / acc[1],
    )


@dataclass
class _ColumnarSpec:
    """Device-lowering parameters resolved from clock + windower."""

    mode: str
    align_ms: int
    len_ms: int
    off_ms: int
    wait_ms: int

    @staticmethod
    def resolve(mode, clock, windower) -> Optional["_ColumnarSpec"]:
        """Lowerable iff the fold is a device fold, time comes from an
        EventClock (RecordBatch.ts IS the event timestamp), and the
        windower is tumbling/sliding (sessions keep the dedicated
        `gpu.operators.keyed_session_agg`)."""
        if mode is None or not isinstance(clock, EventClock):
            return None
        if isinstance(windower, TumblingWindower):
            len_ms = int(windower.length.total_seconds() * 1000)
            off_ms = len_ms
        elif isinstance(windower, SlidingWindower):
            len_ms = int(windower.length.total_seconds() * 1000)
            off_ms = int(windower.offset.total_seconds() * 1000)
        else:
            return None
        if mode in ("min", "max", "mean") and off_ms != len_ms:
            # The stats table is tumbling-only; sliding min/max/mean
            # stays on the host path.
            return None
        align_ms = int(
            (windower.align_to - _EPOCH_UTC).total_seconds() * 1000
        )
        wait_ms = int(
            clock.wait_for_system_duration.total_seconds() * 1000
        )
        return _ColumnarSpec(mode, align_ms, len_ms, off_ms, wait_ms)

    def build(self, resume) -> "_ColumnarWindowLogic":
        return _ColumnarWindowLogic(self, resume)


class _ColumnarWindowLogic(StatefulBatchLogic):
    """The columnar engine behind a lowered `fold_window`.

    Values are RecordBatches; emissions are
    `(COLUMNAR_WINDOW_ID, "E", closed_rows_RecordBatch)` — the same
    tagged-tuple protocol `_WindowLogic` uses, so the `window`
    operator's unwrappers split the streams identically.  Ingestion
    must be watermark-ordered (in-order columnar sources); the late
    stream is not populated on this path.
    """

    def __init__(self, spec: _ColumnarSpec, resume):
        self.spec = spec
        self.state = None  # built on first batch (device known then)
        self._resume = resume

    def _ensure(self, batch) -> None:
        if self.state is not None:
            return
        from ..gpu import AGG_COUNT, AGG_SUM, WindowAggState

        dev = batch.keys.device
        on_gpu = dev.type != "cpu"
        if self.spec.mode in ("min", "max", "mean"):
            from ..gpu.state import StatsAggState

            self.state = StatsAggState(
                dev,
                self.spec.align_ms,
                self.spec.len_ms,
                slots_pow=22 if on_gpu else 16,
                out_cap=1 << (22 if on_gpu else 16),
            )
            if self._resume is not None:
                self.state.restore_from_host(self._resume)
                self._resume = None
            return
        self.state = WindowAggState(
            dev,
            self.spec.align_ms,
            self.spec.len_ms,
            AGG_COUNT if self.spec.mode == "count" else AGG_SUM,
            slots_pow=22 if on_gpu else 16,
            out_cap=1 << (22 if on_gpu else 16),
            radix=on_gpu,
            off_ms=self.spec.off_ms,
        )
        if self._resume is not None:
            self.state.restore_from_host(self._resume)
            self._resume = None

    def _stats_close(self, horizon) -> Optional[Any]:
        """Extract closed (key, window) stats rows as a RecordBatch of
        the fold's column (min or max)."""
        from ..gpu import RecordBatch

        if horizon is not None and horizon <= self.state.closed_horizon:
            return None
        cols = self.state.extract(horizon, clear=True)
        if horizon is not None:
            self.state.closed_horizon = horizon
        if cols is None:
            return None
        wins = cols["wins"].to(dtype=cols["cnt"].dtype)
        if self.spec.mode == "mean":
            vals = cols["sum"].double() / cols["cnt"].double()
        else:
            vals = cols[self.spec.mode]
        return RecordBatch(
            cols["keys"],
            wins * self.spec.len_ms + self.spec.align_ms,
            vals,
        )

    def on_batch(self, values) -> Tuple[Iterable, bool]:
        events: List = []
        for batch in values:
            self._ensure(batch)
            self.state.insert(batch)
        if self.spec.mode in ("min", "max", "mean"):
            wm = self.state.max_ts_host
            horizon = (
                wm - self.spec.wait_ms - self.spec.align_ms
            ) // self.spec.len_ms
            closed = self._stats_close(horizon)
        else:
            closed = self.state.close_due(self.spec.wait_ms)
        if closed is not None:
            events.append((COLUMNAR_WINDOW_ID, "E", closed))
        return (events, False)

    def on_notify(self) -> Tuple[Iterable, bool]:
        return ([], False)

    def on_eof(self) -> Tuple[Iterable, bool]:
        if self.state is None:
            return ([], True)
        if self.spec.mode in ("min", "max", "mean"):
            final = self._stats_close(None)
        else:
            final = self.state.close_all()
        if final is None:
            return ([], True)
        return ([(COLUMNAR_WINDOW_ID, "E", final)], True)

    def notify_at(self) -> Optional[datetime]:
        return None

    def snapshot(self):
        if self.state is None:
            return {"__columnar__": self._resume}
        return {"__columnar__": self.state.snapshot_to_host()}


class _AutoWindowLogic(StatefulBatchLogic):
    """Engine dispatch for `window`: host `_WindowLogic` for
    Python-object values, the fused HIP kernels for RecordBatch values
    — decided on the first batch (streams are dynamically typed), and
    persisted through snapshots."""

    def __init__(self, host_builder, spec: Optional[_ColumnarSpec], resume):
        self._host_builder = host_builder
        self._spec = spec
        self._impl: Optional[StatefulBatchLogic] = None
        if resume is not None:
            if isinstance(resume, dict) and "__columnar__" in resume:
                self._impl = spec.build(resume["__columnar__"])
            else:
                self._impl = host_builder(resume)

    def _pick(self, values) -> None:
        if self._impl is not None:
            return
        v = values[0] if values else None
        if (
            self._spec is not None
            and hasattr(v, "keys")
            and hasattr(v, "ts")
            and hasattr(v, "ts_base")
        ):
            self._impl = self._spec.build(None)
        else:
            self._impl = self._host_builder(None)

    def on_batch(self, values):
        self._pick(values)
        return self._impl.on_batch(values)

    def on_notify(self):
        if self._impl is None:
            return ([], False)
        return self._impl.on_notify()

    def on_eof(self):
        if self._impl is None:
            return ([], True)
        return self._impl.on_eof()

    def notify_at(self):
        if self._impl is None:
            return None
        return self._impl.notify_at()

    def snapshot(self):
        if self._impl is None:
            self._impl = self._host_builder(None)
        return self._impl.snapshot()


def _unwrap_emit(id_typ_obj):
    window_id, typ, obj = id_typ_obj
    return (window_id, obj) if typ == "E" else None


def _unwrap_late(id_typ_obj):
    window_id, typ, obj = id_typ_obj
    return (window_id, obj) if typ == "L" else None


def _unwrap_meta(id_typ_obj):
    window_id, typ, obj = id_typ_obj
    return (window_id, obj) if typ == "M" else None


@operator
def window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    builder: Callable[[Optional[S]], WindowLogic[V, W, S]],
    ordered: bool = True,
    device_mode: Optional[str] = None,
) -> WindowOut[V, W]:
    """Advanced generic windowing operator.

    :arg clock: Time definition.
    :arg windower: Window definition.
    :arg builder: Called whenever a new window is opened with the
        resume state for that window, if any.
    :arg ordered: Whether to apply values to the logic in timestamp
        order (at a performance cost).  Defaults to `True`.
    :returns: Window result streams.
    """

    spec = _ColumnarSpec.resolve(device_mode, clock, windower)

    def host_builder(resume_state: Optional[_WindowSnapshot]) -> _WindowLogic:
        if resume_state is not None:
            clock_logic = clock.build(resume_state.clock_state)
            windower_logic = windower.build(resume_state.windower_state)
            logics = {
                wid: builder(state)
                for wid, state in resume_state.logic_states.items()
            }
            return _WindowLogic(
                clock_logic,
                windower_logic,
                builder,
                ordered,
                logics,
                list(resume_state.queue),
                resume_state.last_watermark,
            )
        return _WindowLogic(
            clock.build(None), windower.build(None), builder, ordered
        )

    def shim_builder(resume_state) -> StatefulBatchLogic:
        if spec is None:
            return host_builder(resume_state)
        return _AutoWindowLogic(host_builder, spec, resume_state)

    events = op.stateful_batch("stateful_batch", up, shim_builder)
    downs = op.filter_map_value("unwrap_down", events, _unwrap_emit)
    lates = op.filter_map_value("unwrap_late", events, _unwrap_late)
    metas = op.filter_map_value("unwrap_meta", events, _unwrap_meta)
    return WindowOut(downs, lates, metas)


# ---------------------------------------------------------------------------
# Derived window operators
# ---------------------------------------------------------------------------


class _FoldWindowLogic(WindowLogic[V, S, S]):
    def __init__(
        self,
        folder: Callable[[S, V], S],
        merger: Callable[[S, S], S],
        state: S,
    ):
        self.folder = folder
        self.merger = merger
        self.state = state

    def on_value(self, value: V) -> Iterable[S]:
        self.state = self.folder(self.state, value)
        return _EMPTY

    def on_merge(self, original: "_FoldWindowLogic[V, S]") -> Iterable[S]:
        self.state = self.merger(self.state, original.state)
        return _EMPTY

    def on_close(self) -> Iterable[S]:
        return (self.state,)

    def snapshot(self) -> S:
        return copy.deepcopy(self.state)


@operator
def fold_window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    builder: Callable[[], S],
    folder: Callable[[S, V], S],
    merger: Callable[[S, S], S],
    ordered: bool = True,
) -> WindowOut[V, S]:
    """Build an empty accumulator, then combine values into it per
    window; the accumulator is emitted downstream when the window
    closes.

    :arg builder: Called whenever a new window opens to create the
        empty accumulator.
    :arg folder: Combines a new value into the accumulator.
    :arg merger: Combines two accumulators when windows merge
        (session windows).

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [
    ...     (align + timedelta(seconds=1), "a"),
    ...     (align + timedelta(seconds=2), "b"),
    ...     (align + timedelta(seconds=61), "c"),
    ... ]
    >>> flow = Dataflow("fold_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> keyed = op.key_on("key", s, lambda x: "ALL")
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> wo = win.fold_window(
    ...     "fold",
    ...     keyed,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     list,
    ...     lambda acc, x: acc + [x[1]],
    ...     lambda a, b: a + b,
    ... )
    >>> op.output("out", wo.down, StdOutSink())
    >>> run_main(flow)
    ('ALL', (0, ['a', 'b']))
    ('ALL', (1, ['c']))
    """

    def shim_builder(resume_state: Optional[S]) -> _FoldWindowLogic[V, S]:
        state = resume_state if resume_state is not None else builder()
        return _FoldWindowLogic(folder, merger, state)

    mode = folder.mode if isinstance(folder, DeviceFoldable) else None
    out = window(
        "window", up, clock, windower, shim_builder, ordered=ordered,
        device_mode=mode,
    )
    fin = folder.finish if isinstance(folder, DeviceFoldable) else None
    if fin is not None:
        # Host-path accumulators need the finisher at close; columnar
        # closes (COLUMNAR_WINDOW_ID) already emit finished values.
        def _finish_value(wid_acc):
            wid, acc = wid_acc
            if wid == COLUMNAR_WINDOW_ID:
                return wid_acc
            return (wid, fin(acc))

        downs = op.map_value("finish", out.down, _finish_value)
        return WindowOut(downs, out.late, out.meta)
    return out


def _collect_list_folder(s: List[V], v: V) -> List[V]:
    s.append(v)
    return s


def _collect_set_folder(s: Set[V], v: V) -> Set[V]:
    s.add(v)
    return s


def _collect_dict_folder(s: Dict, k_v: Tuple) -> Dict:
    try:
        k, v = k_v
    except (TypeError, ValueError) as ex:
        msg = (
            "collect_window with `dict` requires (key, value) 2-tuples "
            f"as values; got a {type(k_v)!r} instead"
        )
        raise TypeError(msg) from ex
    s[k] = v
    return s


def _merge_dicts(a: Dict, b: Dict) -> Dict:
    a.update(b)
    return a


def _merge_sets(a: Set, b: Set) -> Set:
    return a | b


@operator
def collect_window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    into=list,
    ordered: bool = True,
) -> WindowOut[V, Any]:
    """Collect items in a window into a container (`list`, `set` or
    `dict`; for `dict` the values must be `(key, value)` 2-tuples).

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [
    ...     (align + timedelta(seconds=s), v)
    ...     for s, v in [(1, 1), (2, 2), (61, 3)]
    ... ]
    >>> flow = Dataflow("collect_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> keyed = op.key_on("key", s, lambda x: "ALL")
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> wo = win.collect_window(
    ...     "collect",
    ...     keyed,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ... )
    >>> out = []
    >>> op.output("out", wo.down, TestingSink(out))
    >>> run_main(flow)
    >>> [(k, (wid, [v for _ts, v in items])) for k, (wid, items) in out]
    [('ALL', (0, [1, 2])), ('ALL', (1, [3]))]
    """
    if into is list:
        folder, merger = _collect_list_folder, (lambda a, b: a + b)
    elif into is set:
        folder, merger = _collect_set_folder, _merge_sets
    elif into is dict:
        folder, merger = _collect_dict_folder, _merge_dicts
    else:
        msg = f"`into` must be `list`, `set` or `dict`; got {into!r}"
        raise TypeError(msg)
    return fold_window(
        "fold_window", up, clock, windower, into, folder, merger,
        ordered=ordered,
    )


@operator
def count_window(
    step_id: str,
    up: Stream[X],
    clock: Clock[X, Any],
    windower: Windower[Any],
    key: Callable[[X], str],
) -> WindowOut[X, int]:
    """Count the number of occurrences of items in a window.

    :arg key: Called on each item to route the counts.
    :returns: Window result streams; downstream contains `(key,
        (window_id, count))` once the window closes.

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [align, align + timedelta(seconds=30), align + timedelta(seconds=65)]
    >>> flow = Dataflow("count_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x, wait_for_system_duration=timedelta(0)
    ... )
    >>> wo = win.count_window(
    ...     "count",
    ...     s,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     key=lambda x: "ALL",
    ... )
    >>> op.output("out", wo.down, StdOutSink())
    >>> run_main(flow)
    ('ALL', (0, 2))
    ('ALL', (1, 1))
    """
    keyed = op.key_on("key", up, key)
    return fold_window(
        "fold_window",
        keyed,
        clock,
        windower,
        int,
        COUNT_FOLD,
        lambda a, b: a + b,
        ordered=False,
    )


@operator
def max_window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    by=_identity,
) -> WindowOut[V, V]:
    """Find the maximum value for each key per window.

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [
    ...     (align + timedelta(seconds=s), v)
    ...     for s, v in [(1, 5), (2, 9), (61, 3)]
    ... ]
    >>> flow = Dataflow("max_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> keyed = op.key_on("key", s, lambda x: "ALL")
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> wo = win.max_window(
    ...     "max",
    ...     keyed,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     by=lambda x: x[1],
    ... )
    >>> out = []
    >>> op.output("out", wo.down, TestingSink(out))
    >>> run_main(flow)
    >>> [(k, (wid, v[1])) for k, (wid, v) in out]
    [('ALL', (0, 9)), ('ALL', (1, 3))]

    Over :class:`bytewax_amd.gpu.RecordBatch` streams this lowers
    onto the fused stats kernels (the batch's `vals` column is the
    compared value; `by` applies to host items only).
    """
    return reduce_window(
        "reduce_window",
        up,
        clock,
        windower,
        DeviceFoldable("max", partial(max, key=by)),
    )


@operator
def min_window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    by=_identity,
) -> WindowOut[V, V]:
    """Find the minimum value for each key per window.

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [
    ...     (align + timedelta(seconds=s), v)
    ...     for s, v in [(1, 5), (2, 9), (61, 3)]
    ... ]
    >>> flow = Dataflow("min_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> keyed = op.key_on("key", s, lambda x: "ALL")
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> wo = win.min_window(
    ...     "min",
    ...     keyed,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     by=lambda x: x[1],
    ... )
    >>> out = []
    >>> op.output("out", wo.down, TestingSink(out))
    >>> run_main(flow)
    >>> [(k, (wid, v[1])) for k, (wid, v) in out]
    [('ALL', (0, 5)), ('ALL', (1, 3))]

    Over :class:`bytewax_amd.gpu.RecordBatch` streams this lowers
    onto the fused stats kernels (the batch's `vals` column is the
    compared value; `by` applies to host items only).
    """
    return reduce_window(
        "reduce_window",
        up,
        clock,
        windower,
        DeviceFoldable("min", partial(min, key=by)),
    )


@operator
def reduce_window(
    step_id: str,
    up: KeyedStream[V],
    clock: Clock[V, Any],
    windower: Windower[Any],
    reducer: Callable[[V, V], V],
) -> WindowOut[V, V]:
    """Distill all values for a key in a window down into a single
    value; like {py:obj}`fold_window` but the first value is the
    initial accumulator.

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> inp = [
    ...     (align + timedelta(seconds=s), v)
    ...     for s, v in [(1, 5), (2, 9), (61, 3)]
    ... ]
    >>> flow = Dataflow("reduce_window_eg")
    >>> s = op.input("inp", flow, TestingSource(inp))
    >>> keyed = op.key_on("key", s, lambda x: "ALL")
    >>> clock = win.EventClock(
    ...     ts_getter=lambda x: x[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> wo = win.reduce_window(
    ...     "sum",
    ...     keyed,
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     lambda a, b: (a[0], a[1] + b[1]),
    ... )
    >>> out = []
    >>> op.output("out", wo.down, TestingSink(out))
    >>> run_main(flow)
    >>> [(k, (wid, v[1])) for k, (wid, v) in out]
    [('ALL', (0, 14)), ('ALL', (1, 3))]
    """

    def shim_folder(s, v):
        return v if s is None else reducer(s, v)

    if isinstance(reducer, DeviceFoldable):
        # Keep the device-lowering mode through the reduce shim
        # (max_window/min_window pass device folds so RecordBatch
        # streams hit the stats kernels; host items still reduce
        # with the first value as the initial accumulator).
        shim_folder = DeviceFoldable(
            reducer.mode, shim_folder, finish=reducer.finish
        )

    def none_builder():
        return None

    return fold_window(
        "fold_window",
        up,
        clock,
        windower,
        none_builder,
        shim_folder,
        reducer,
        ordered=False,
    )


class _JoinWindowLogic(WindowLogic):
    def __init__(
        self,
        insert_mode: str,
        emit_mode: str,
        state: _JoinState,
    ):
        self.insert_mode = insert_mode
        self.emit_mode = emit_mode
        self.state = state

    def on_value(self, value: Tuple[int, Any]) -> Iterable[Tuple]:
        side, v = value
        if self.insert_mode == "first":
            if not self.state.is_set(side):
                self.state.set_val(side, v)
        elif self.insert_mode == "last":
            self.state.set_val(side, v)
        else:
            self.state.add_val(side, v)
        if self.emit_mode == "running":
            return self.state.astuples()
        if self.emit_mode == "complete" and self.state.all_set():
            out = self.state.astuples()
            self.state.clear()
            return out
        return _EMPTY

    def on_merge(self, original: "_JoinWindowLogic") -> Iterable[Tuple]:
        for side, vals in enumerate(original.state.seen):
            for v in vals:
                if self.insert_mode == "first":
                    if not self.state.is_set(side):
                        self.state.set_val(side, v)
                elif self.insert_mode == "last":
                    self.state.set_val(side, v)
                else:
                    self.state.add_val(side, v)
        return _EMPTY

    def on_close(self) -> Iterable[Tuple]:
        if self.emit_mode == "final":
            return self.state.astuples()
        return _EMPTY

    def snapshot(self) -> _JoinState:
        return copy.deepcopy(self.state)


@operator
def join_window(
    step_id: str,
    clock: Clock[Any, Any],
    windower: Windower[Any],
    *sides: KeyedStream[Any],
    insert_mode: JoinInsertMode = "last",
    emit_mode: JoinEmitMode = "final",
    ordered: bool = True,
) -> WindowOut[Any, Tuple]:
    """Gather together the value for a key on multiple streams within
    a window.

    :returns: Window result streams; downstream contains tuples with
        the value from each side in argument order.

    Example:

    >>> align = datetime(2024, 1, 1, tzinfo=timezone.utc)
    >>> names = [(align + timedelta(seconds=1), ("1", "alice"))]
    >>> mails = [(align + timedelta(seconds=2), ("1", "a@x.io"))]
    >>> flow = Dataflow("join_window_eg")
    >>> n = op.input("n", flow, TestingSource(names))
    >>> m = op.input("m", flow, TestingSource(mails))
    >>> clock = win.EventClock(
    ...     ts_getter=lambda ts_kv: ts_kv[0],
    ...     wait_for_system_duration=timedelta(0),
    ... )
    >>> n_keyed = op.map("nk", n, lambda ts_kv: (ts_kv[1][0], ts_kv))
    >>> m_keyed = op.map("mk", m, lambda ts_kv: (ts_kv[1][0], ts_kv))
    >>> wo = win.join_window(
    ...     "join",
    ...     clock,
    ...     win.TumblingWindower(align_to=align, length=timedelta(minutes=1)),
    ...     n_keyed,
    ...     m_keyed,
    ... )
    >>> out = []
    >>> op.output("out", wo.down, TestingSink(out))
    >>> run_main(flow)
    >>> [(k, (wid, tuple(v[1][1] for v in vals))) for k, (wid, vals) in out]
    [('1', (0, ('alice', 'a@x.io')))]
    """
    if insert_mode not in typing.get_args(JoinInsertMode):
        msg = f"unknown join insert mode {insert_mode!r}"
        raise ValueError(msg)
    if emit_mode not in typing.get_args(JoinEmitMode):
        msg = f"unknown join emit mode {emit_mode!r}"
        raise ValueError(msg)

    side_count = len(sides)
    merged = op._join_label_merge("add_names", *sides)

    # The merged stream's values are (side, value); un-shim the
    # timestamp getter so users can keep clocks in terms of their own
    # values (reference windowing.py:2110-2124).
    if isinstance(clock, EventClock):
        value_ts_getter = clock.ts_getter

        def shim_getter(i_v):
            _i, v = i_v
            return value_ts_getter(v)

        clock = EventClock(
            ts_getter=shim_getter,
            wait_for_system_duration=clock.wait_for_system_duration,
            now_getter=clock.now_getter,
            to_system_utc=clock.to_system_utc,
        )

    def shim_builder(resume_state: Optional[_JoinState]) -> _JoinWindowLogic:
        state = (
            resume_state
            if resume_state is not None
            else _JoinState.for_side_count(side_count)
        )
        return _JoinWindowLogic(insert_mode, emit_mode, state)

    return window(
        "window", merged, clock, windower, shim_builder, ordered=ordered
    )
