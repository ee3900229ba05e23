"""Built-in operators.

Only nine *core* operators exist in the engine (`_noop`, `branch`,
`flat_map_batch`, `input`, `inspect_debug`, `merge`, `output`,
`redistribute`, `stateful_batch`); everything else here is Python
composition over them, mirroring the reference's design
(``bytewax/operators/__init__.py``, reference operators.rs:34-1041 for
the engine side).

See the {py:obj}`bytewax_amd.dataflow` module docstring for how graphs
are built.
"""

import copy
import typing
from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import (
    Any,
    Callable,
    Dict,
    Generic,
    Iterable,
    List,
    Optional,
    Tuple,
    TypeVar,
)

from ..errors import BytewaxTypeError
from ..dataflow import (
    Dataflow,
    KeyedStream,
    Stream,
    operator,
)

V = TypeVar("V")
W = TypeVar("W")
X = TypeVar("X")
Y = TypeVar("Y")
S = TypeVar("S")
DK = TypeVar("DK")
DV = TypeVar("DV")

__all__ = [
    "BranchOut",
    "JoinEmitMode",
    "JoinInsertMode",
    "StatefulBatchLogic",
    "StatefulLogic",
    "TTLCache",
    "branch",
    "collect",
    "count_final",
    "enrich_cached",
    "filter",
    "filter_map",
    "filter_map_value",
    "filter_value",
    "flat_map",
    "flat_map_batch",
    "flat_map_value",
    "flatten",
    "fold_final",
    "input",
    "inspect",
    "inspect_debug",
    "join",
    "key_on",
    "key_rm",
    "map",
    "map_value",
    "max_final",
    "merge",
    "min_final",
    "output",
    "raises",
    "redistribute",
    "reduce_final",
    "stateful",
    "stateful_batch",
    "stateful_flat_map",
    "stateful_map",
]


def f_repr(f: Callable) -> str:
    """Nicer repr for callables in error messages."""
    name = getattr(f, "__qualname__", None) or getattr(f, "__name__", None)
    if name is not None:
        mod = getattr(f, "__module__", None)
        return f"`{mod}.{name}`" if mod else f"`{name}`"
    return repr(f)


def _identity(x: X) -> X:
    return x


def _get_system_utc() -> datetime:
    return datetime.now(timezone.utc)


# ---------------------------------------------------------------------------
# Core operators.  The builder bodies only *declare* output port names;
# the engine interprets the recorded payloads (see _engine/compile.py).
# ---------------------------------------------------------------------------


@operator(_core=True)
def _noop(step_id: str, up: Stream[X]) -> Stream[X]:
    """No-op; passes the stream through.

    Used for graph-shape bookkeeping.
    """
    return "down"  # type: ignore[return-value]


@dataclass(frozen=True)
class BranchOut(Generic[X, Y]):
    """Streams returned from the {py:obj}`branch` operator."""

    trues: Stream[X]
    falses: Stream[Y]


@operator(_core=True)
def branch(
    step_id: str,
    up: Stream[X],
    predicate: Callable[[X], bool],
) -> BranchOut[X, Any]:
    """Divide items into two streams with a predicate.

    :arg step_id: Unique ID.
    :arg up: Stream to divide.
    :arg predicate: Called on each item; which output stream the item
        is sent to depends on the truthiness of the return value.
    :returns: A {py:obj}`BranchOut` with `trues` and `falses` streams.

    Example:

    >>> flow = Dataflow("branch_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2, 3, 4]))
    >>> b = op.branch("is_even", s, lambda x: x % 2 == 0)
    >>> evens, odds = [], []
    >>> op.output("e", b.trues, TestingSink(evens))
    >>> op.output("o", b.falses, TestingSink(odds))
    >>> run_main(flow)
    >>> (evens, odds)
    ([2, 4], [1, 3])
    """
    if not callable(predicate):
        msg = "`predicate` must be callable"
        raise TypeError(msg)
    return BranchOut(trues="trues", falses="falses")  # type: ignore[arg-type]


@operator(_core=True)
def flat_map_batch(
    step_id: str,
    up: Stream[X],
    mapper: Callable[[List[X]], Iterable[Y]],
) -> Stream[Y]:
    """Transform an entire batch of items 1-to-many.

    The mapper sees batches of items in the order the engine happens to
    batch them — use this to amortize per-call overhead of expensive
    vectorized transformations.

    Example:

    >>> flow = Dataflow("flat_map_batch_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2]))
    >>> s = op.flat_map_batch(
    ...     "expand", s, lambda xs: [x * 10 for x in xs]
    ... )
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    10
    20
    """
    if not callable(mapper):
        msg = "`mapper` must be callable"
        raise TypeError(msg)
    return "down"  # type: ignore[return-value]


@operator(_core=True)
def input(  # noqa: A001
    step_id: str,
    flow: Dataflow,
    source: Any,
) -> Stream[Any]:
    """Introduce items into a dataflow from a source.

    :arg source: A {py:obj}`bytewax_amd.inputs.Source`.
    """
    from ..inputs import Source

    if not isinstance(source, Source):
        msg = f"`source` must be a `Source`; got {type(source)!r}"
        raise TypeError(msg)
    return "down"  # type: ignore[return-value]


def _default_debug_inspector(step_id: str, item: Any, epoch: int, worker: int) -> None:
    print(f"{step_id} W{worker} @{epoch}: {item!r}", flush=True)


@operator(_core=True)
def inspect_debug(
    step_id: str,
    up: Stream[X],
    inspector: Callable[[str, X, int, int], None] = _default_debug_inspector,
) -> Stream[X]:
    """Observe items, their epoch, and worker.

    :arg inspector: Called with the step ID, item, epoch, and worker
        index for each item.

    Example:

    >>> flow = Dataflow("dbg")
    >>> s = op.input("inp", flow, TestingSource([1]))
    >>> s = op.inspect_debug("d", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    dbg.d W0 @1: 1
    1
    """
    return "down"  # type: ignore[return-value]


@operator(_core=True)
def merge(step_id: str, *ups: Stream[Any]) -> Stream[Any]:
    """Combine multiple streams together into one.

    Example:

    >>> flow = Dataflow("merge_eg")
    >>> a = op.input("a", flow, TestingSource([1, 2]))
    >>> b = op.input("b", flow, TestingSource([10, 20]))
    >>> out = []
    >>> op.output("out", op.merge("merge", a, b), TestingSink(out))
    >>> run_main(flow)
    >>> sorted(out)
    [1, 2, 10, 20]
    """
    if len(ups) < 1:
        msg = "`merge` needs at least one upstream"
        raise TypeError(msg)
    return "down"  # type: ignore[return-value]


@operator(_core=True)
def output(step_id: str, up: Stream[X], sink: Any) -> None:
    """Write items out of a dataflow to a sink.

    :arg sink: A {py:obj}`bytewax_amd.outputs.Sink`.
    """
    from ..outputs import Sink

    if not isinstance(sink, Sink):
        msg = f"`sink` must be a `Sink`; got {type(sink)!r}"
        raise TypeError(msg)
    return None


@operator(_core=True)
def redistribute(step_id: str, up: Stream[X]) -> Stream[X]:
    """Redistribute items randomly across all workers.

    Use to rebalance skewed load after a filter or a skewed-key
    section.  On the GPU engine this is an RCCL all-to-all with random
    bucket assignment (reference operators.rs:345-361).

    Example:

    >>> flow = Dataflow("redistribute_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2, 3]))
    >>> s = op.redistribute("spread", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    1
    2
    3
    """
    return "down"  # type: ignore[return-value]


@operator(_core=True)
def stateful_batch(
    step_id: str,
    up: KeyedStream[V],
    builder: Callable[[Optional[S]], "StatefulBatchLogic[V, W, S]"],
) -> KeyedStream[W]:
    """Advanced generic stateful operator.

    Subclass {py:obj}`StatefulBatchLogic` to define its behavior.

    :arg builder: Called whenever a new key is encountered with the
        resume state returned from snapshotting, if any.

    Example:

    >>> class BatchSum(op.StatefulBatchLogic):
    ...     def __init__(self, resume):
    ...         self.total = resume or 0
    ...     def on_batch(self, values):
    ...         self.total += sum(values)
    ...         return ([self.total], op.StatefulBatchLogic.RETAIN)
    ...     def snapshot(self):
    ...         return self.total
    >>> flow = Dataflow("sb")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("a", 2)]))
    >>> s = op.stateful_batch("sum", s, BatchSum)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 1)
    ('a', 3)
    """
    if not callable(builder):
        msg = "`builder` must be callable"
        raise TypeError(msg)
    return "down"  # type: ignore[return-value]


# ---------------------------------------------------------------------------
# Stateful logic ABCs
# ---------------------------------------------------------------------------


class StatefulBatchLogic(ABC, Generic[V, W, S]):
    """Abstract class to define a {py:obj}`stateful_batch` operator.

    The operator will call these methods in order: {py:obj}`on_batch`
    once with all items queued, then {py:obj}`on_notify` if the
    notification time has passed, then {py:obj}`on_eof` if the upstream
    is EOF and no new items will be received this execution.  If the
    logic is retained after all the above calls then
    {py:obj}`notify_at` will be called.  {py:obj}`snapshot` is
    periodically called.
    """

    #: This logic should be retained after this returns.
    RETAIN: bool = False
    #: This logic should be discarded immediately after this returns.
    DISCARD: bool = True

    @abstractmethod
    def on_batch(self, values: List[V]) -> Tuple[Iterable[W], bool]:
        """Called on each new batch of values for this key.

        :returns: 2-tuple of (emitted items, whether to discard this
            logic).
        """
        ...

    def on_notify(self) -> Tuple[Iterable[W], bool]:
        """Called when the scheduled notification time has passed."""
        return ([], StatefulBatchLogic.RETAIN)

    def on_eof(self) -> Tuple[Iterable[W], bool]:
        """Called once the upstream is EOF (this execution only)."""
        return ([], StatefulBatchLogic.RETAIN)

    def notify_at(self) -> Optional[datetime]:
        """Next system time this logic should be awoken, if any."""
        return None

    @abstractmethod
    def snapshot(self) -> S:
        """Return the immutable state to pickle for recovery.

        The state must be `copy.deepcopy`-able or otherwise isolated
        from later mutation.
        """
        ...


class StatefulLogic(ABC, Generic[V, W, S]):
    """Abstract class to define a {py:obj}`stateful` operator.

    Like {py:obj}`StatefulBatchLogic` but called per-item.
    """

    RETAIN: bool = False
    DISCARD: bool = True

    @abstractmethod
    def on_item(self, value: V) -> Tuple[Iterable[W], bool]:
        """Called on each new upstream item for this key."""
        ...

    def on_notify(self) -> Tuple[Iterable[W], bool]:
        """Called when the scheduled notification time has passed."""
        return ([], StatefulLogic.RETAIN)

    def on_eof(self) -> Tuple[Iterable[W], bool]:
        """Called once the upstream is EOF (this execution only)."""
        return ([], StatefulLogic.RETAIN)

    def notify_at(self) -> Optional[datetime]:
        """Next system time this logic should be awoken, if any."""
        return None

    @abstractmethod
    def snapshot(self) -> S:
        """Return the immutable state to pickle for recovery."""
        ...


class _StatefulLogicShim(StatefulBatchLogic[V, W, S]):
    def __init__(
        self,
        inner: StatefulLogic[V, W, S],
        builder: Optional[Callable[[Optional[S]], StatefulLogic[V, W, S]]] = None,
    ):
        self.inner: Optional[StatefulLogic[V, W, S]] = inner
        self.builder = builder

    def on_batch(self, values: List[V]) -> Tuple[Iterable[W], bool]:
        # A mid-batch complete DISCARDS the per-item logic but must
        # NOT drop the remaining values: a fresh logic is built for
        # them, like the reference's shim (which rebuilds and
        # continues the loop; pysrc _StatefulLogic.on_batch).
        out: List[W] = []
        inner = self.inner
        for v in values:
            if inner is None:
                inner = self.builder(None) if self.builder else None
                if inner is None:
                    break
            ws, is_complete = inner.on_item(v)
            out.extend(ws)
            if is_complete:
                inner = None
        self.inner = inner
        if inner is None:
            return (out, StatefulBatchLogic.DISCARD)
        return (out, StatefulBatchLogic.RETAIN)

    def on_notify(self) -> Tuple[Iterable[W], bool]:
        return self.inner.on_notify()

    def on_eof(self) -> Tuple[Iterable[W], bool]:
        return self.inner.on_eof()

    def notify_at(self) -> Optional[datetime]:
        return self.inner.notify_at()

    def snapshot(self) -> S:
        return self.inner.snapshot()


@operator
def stateful(
    step_id: str,
    up: KeyedStream[V],
    builder: Callable[[Optional[S]], StatefulLogic[V, W, S]],
) -> KeyedStream[W]:
    """Advanced generic stateful operator (per-item variant).

    Subclass {py:obj}`StatefulLogic` to define its behavior.

    Example:

    >>> class RunningMax(op.StatefulLogic):
    ...     def __init__(self, resume):
    ...         self.mx = resume if resume is not None else 0
    ...     def on_item(self, v):
    ...         self.mx = max(self.mx, v)
    ...         return ((self.mx,), op.StatefulLogic.RETAIN)
    ...     def on_notify(self):
    ...         return ((), op.StatefulLogic.RETAIN)
    ...     def on_eof(self):
    ...         return ((), op.StatefulLogic.DISCARD)
    ...     def notify_at(self):
    ...         return None
    ...     def snapshot(self):
    ...         return self.mx
    >>> flow = Dataflow("stateful_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 2), ("a", 1)]))
    >>> s = op.stateful("max", s, RunningMax)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 2)
    ('a', 2)
    
    """

    def shim_builder(resume_state: Optional[S]) -> _StatefulLogicShim[V, W, S]:
        return _StatefulLogicShim(builder(resume_state), builder)

    return stateful_batch("stateful_batch", up, shim_builder)


# ---------------------------------------------------------------------------
# Derived stateless operators
# ---------------------------------------------------------------------------


@operator
def flat_map(
    step_id: str,
    up: Stream[X],
    mapper: Callable[[X], Iterable[Y]],
) -> Stream[Y]:
    """Transform items one-to-many.

    Example:

    >>> flow = Dataflow("flat_map_eg")
    >>> s = op.input("inp", flow, TestingSource(["hello world", "hi"]))
    >>> s = op.flat_map("split", s, str.split)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    hello
    world
    hi
    """

    def shim_mapper(xs: List[X]) -> Iterable[Y]:
        for x in xs:
            for y in mapper(x):
                yield y

    return flat_map_batch("flat_map_batch", up, shim_mapper)


@operator
def flat_map_value(
    step_id: str,
    up: KeyedStream[V],
    mapper: Callable[[V], Iterable[W]],
) -> KeyedStream[W]:
    """Transform values one-to-many.

    Example:

    >>> flow = Dataflow("flat_map_value_eg")
    >>> s = op.input("inp", flow, TestingSource([("k", "a b")]))
    >>> s = op.flat_map_value("split", s, str.split)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('k', 'a')
    ('k', 'b')
    """

    def shim_mapper(k_v: Tuple[str, V]) -> Iterable[Tuple[str, W]]:
        try:
            k, v = k_v
        except TypeError as ex:
            msg = (
                f"step {step_id!r} requires `(key, value)` 2-tuples "
                f"as upstream items; got a {type(k_v)!r} instead"
            )
            raise TypeError(msg) from ex
        for w in mapper(v):
            yield (k, w)

    return flat_map("flat_map", up, shim_mapper)


@operator
def flatten(
    step_id: str,
    up: Stream[Iterable[X]],
) -> Stream[X]:
    """Move all sub-items up a level.

    Example:

    >>> flow = Dataflow("flatten_eg")
    >>> s = op.input("inp", flow, TestingSource([[1, 2], [3]]))
    >>> s = op.flatten("flatten", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    1
    2
    3
    """

    def shim_mapper(x: Iterable[X]) -> Iterable[X]:
        if not isinstance(x, Iterable):
            msg = (
                f"step {step_id!r} requires iterable items; "
                f"got a {type(x)!r} instead"
            )
            raise BytewaxTypeError(msg)
        return x

    return flat_map("flat_map", up, shim_mapper)


@operator
def filter(  # noqa: A001
    step_id: str,
    up: Stream[X],
    predicate: Callable[[X], bool],
) -> Stream[X]:
    """Keep only the items where `predicate` returns `True`.

    ```python
    s = op.filter("odds", s, lambda x: x % 2 == 1)
    ```

    The predicate must return a `bool` (not merely truthy).

    Example:

    >>> flow = Dataflow("filter_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2, 3, 4]))
    >>> s = op.filter("keep_even", s, lambda x: x % 2 == 0)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    2
    4
    """

    def shim_mapper(x: X) -> Iterable[X]:
        keep = predicate(x)
        if not isinstance(keep, bool):
            msg = (
                f"return value of `predicate` {f_repr(predicate)} "
                f"in step {step_id!r} must be a `bool`; "
                f"got a {type(keep)!r} instead"
            )
            raise BytewaxTypeError(msg)
        if keep:
            return (x,)
        return ()

    return flat_map("flat_map", up, shim_mapper)


@operator
def filter_value(
    step_id: str,
    up: KeyedStream[V],
    predicate: Callable[[V], bool],
) -> KeyedStream[V]:
    """Keep only some values.

    Example:

    >>> flow = Dataflow("filter_value_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("a", 2)]))
    >>> s = op.filter_value("keep_big", s, lambda v: v > 1)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 2)
    """

    def shim_mapper(v: V) -> Iterable[V]:
        keep = predicate(v)
        if not isinstance(keep, bool):
            msg = (
                f"return value of `predicate` {f_repr(predicate)} "
                f"in step {step_id!r} must be a `bool`; "
                f"got a {type(keep)!r} instead"
            )
            raise BytewaxTypeError(msg)
        if keep:
            return (v,)
        return ()

    return flat_map_value("filter_value", up, shim_mapper)


@operator
def filter_map(
    step_id: str,
    up: Stream[X],
    mapper: Callable[[X], Optional[Y]],
) -> Stream[Y]:
    """A one-to-maybe-one transformation; `None` is discarded.

    Example:

    >>> def parse(x):
    ...     try:
    ...         return int(x)
    ...     except ValueError:
    ...         return None
    >>> flow = Dataflow("filter_map_eg")
    >>> s = op.input("inp", flow, TestingSource(["1", "x", "3"]))
    >>> s = op.filter_map("parse", s, parse)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    1
    3
    """

    def shim_mapper(x: X) -> Iterable[Y]:
        y = mapper(x)
        if y is not None:
            return (y,)
        return ()

    return flat_map("flat_map", up, shim_mapper)


@operator
def filter_map_value(
    step_id: str,
    up: KeyedStream[V],
    mapper: Callable[[V], Optional[W]],
) -> KeyedStream[W]:
    """Transform values one-to-maybe-one; `None` is discarded.

    Example:

    >>> flow = Dataflow("filter_map_value_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", "1"), ("a", "x")]))
    >>> s = op.filter_map_value(
    ...     "parse", s, lambda v: int(v) if v.isdigit() else None
    ... )
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 1)
    """

    def shim_mapper(v: V) -> Iterable[W]:
        w = mapper(v)
        if w is not None:
            return (w,)
        return ()

    return flat_map_value("filter_map_value", up, shim_mapper)


@operator
def inspect(
    step_id: str,
    up: Stream[X],
    inspector: Optional[Callable[[str, X], None]] = None,
) -> Stream[X]:
    """Observe items for debugging.

    The default inspector prints ``{step_id}: {item!r}``.

    Example:

    >>> flow = Dataflow("inspect_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2]))
    >>> s = op.inspect("check", s)
    >>> op.output("out", s, TestingSink([]))
    >>> run_main(flow)
    inspect_eg.check: 1
    inspect_eg.check: 2
    """
    if inspector is None:

        def inspector(s_id: str, item: X) -> None:
            print(f"{s_id}: {item!r}", flush=True)

    def shim_inspector(
        _fq_step_id: str, item: X, _epoch: int, _worker_idx: int
    ) -> None:
        # `step_id` here is already fully qualified (builders receive
        # fq ids), matching the reference's printed format.
        inspector(step_id, item)

    return inspect_debug("inspect_debug", up, shim_inspector)


@operator
def map(  # noqa: A001
    step_id: str,
    up: Stream[X],
    mapper: Callable[[X], Y],
) -> Stream[Y]:
    """Transform items one-by-one.

    ```python
    s = op.input("inp", flow, TestingSource([1, 2, 3]))
    s = op.map("add_one", s, lambda x: x + 1)  # 2, 3, 4
    ```

    Example:

    >>> flow = Dataflow("map_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2, 3]))
    >>> s = op.map("add_one", s, lambda x: x + 1)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    2
    3
    4
    """

    def shim_mapper(xs: List[X]) -> Iterable[Y]:
        return [mapper(x) for x in xs]

    return flat_map_batch("flat_map_batch", up, shim_mapper)


@operator
def map_value(
    step_id: str,
    up: KeyedStream[V],
    mapper: Callable[[V], W],
) -> KeyedStream[W]:
    """Transform values one-by-one.

    Example:

    >>> flow = Dataflow("map_value_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("b", 2)]))
    >>> s = op.map_value("add_one", s, lambda v: v + 1)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 2)
    ('b', 3)
    """

    def shim_mapper(k_v: Tuple[str, V]) -> Tuple[str, W]:
        try:
            k, v = k_v
        except TypeError as ex:
            msg = (
                f"step {step_id!r} requires `(key, value)` 2-tuples "
                f"as upstream items; got a {type(k_v)!r} instead"
            )
            raise TypeError(msg) from ex
        return (k, mapper(v))

    return map("map", up, shim_mapper)


@operator
def key_on(step_id: str, up: Stream[X], key: Callable[[X], str]) -> KeyedStream[X]:
    """Add a key for each item, making a {py:obj}`KeyedStream`.

    ```python
    keyed = op.key_on("k", s, lambda x: x["user"])  # -> (key, item)
    ```

    The key function must return a `str` (keys route state across
    workers).

    Example:

    >>> flow = Dataflow("key_on_eg")
    >>> s = op.input("inp", flow, TestingSource([1, 2]))
    >>> s = op.key_on("key", s, str)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('1', 1)
    ('2', 2)
    """

    def shim_mapper(x: X) -> Tuple[str, X]:
        k = key(x)
        if not isinstance(k, str):
            msg = (
                f"return value of `key` {f_repr(key)} "
                f"in step {step_id!r} must be a `str`; "
                f"got a {type(k)!r} instead"
            )
            raise BytewaxTypeError(msg)
        return (k, x)

    return map("map", up, shim_mapper)


@operator
def key_rm(step_id: str, up: KeyedStream[X]) -> Stream[X]:
    """Discard keys from a keyed stream.

    Example:

    >>> flow = Dataflow("key_rm_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("b", 2)]))
    >>> s = op.key_rm("unkey", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    1
    2
    """

    def shim_mapper(k_v: Tuple[str, X]) -> X:
        _k, v = k_v
        return v

    return map("map", up, shim_mapper)


@operator
def raises(step_id: str, up: Stream[Any]) -> None:
    """Raise an exception and crash the dataflow on any item.

    Example:

    Mark a branch that must stay empty; any item that reaches it
    crashes the execution:

    >>> flow = Dataflow("raises_eg")
    >>> s = op.input("inp", flow, TestingSource([1, -2]))
    >>> b = op.branch("valid", s, lambda x: x > 0)
    >>> op.raises("no_negatives", b.falses)
    >>> op.output("out", b.trues, StdOutSink())
    >>> run_main(flow)
    Traceback (most recent call last):
        ...
    bytewax_amd.errors.BytewaxRuntimeError: `raises` step 'raises_eg.no_negatives' got an item: -2
    """
    from ..errors import BytewaxRuntimeError

    def shim_mapper(x: Any) -> Iterable[Any]:
        msg = f"`raises` step {step_id!r} got an item: {x!r}"
        raise BytewaxRuntimeError(msg)

    from ..connectors.stdio import StdOutSink

    errs = flat_map("flat_map", up, shim_mapper)
    return output("output", errs, StdOutSink())


# ---------------------------------------------------------------------------
# Derived stateful operators
# ---------------------------------------------------------------------------


@dataclass
class _CollectState(Generic[V]):
    acc: List[V] = field(default_factory=list)
    timeout_at: Optional[datetime] = None


class _CollectLogic(StatefulLogic[V, List[V], _CollectState[V]]):
    def __init__(
        self,
        step_id: str,
        now_getter: Callable[[], datetime],
        timeout: timedelta,
        max_size: int,
        state: _CollectState[V],
    ):
        self.step_id = step_id
        self.now_getter = now_getter
        self.timeout = timeout
        self.max_size = max_size
        self.state = state

    def on_item(self, value: V) -> Tuple[Iterable[List[V]], bool]:
        self.state.acc.append(value)
        if self.state.timeout_at is None:
            self.state.timeout_at = self.now_getter() + self.timeout
        if len(self.state.acc) >= self.max_size:
            acc = self.state.acc
            self.state.acc = []
            self.state.timeout_at = None
            return ((acc,), StatefulLogic.DISCARD)
        return ((), StatefulLogic.RETAIN)

    def on_notify(self) -> Tuple[Iterable[List[V]], bool]:
        acc = self.state.acc
        self.state.acc = []
        self.state.timeout_at = None
        return ((acc,), StatefulLogic.DISCARD)

    def on_eof(self) -> Tuple[Iterable[List[V]], bool]:
        acc = self.state.acc
        self.state.acc = []
        self.state.timeout_at = None
        return ((acc,), StatefulLogic.DISCARD)

    def notify_at(self) -> Optional[datetime]:
        return self.state.timeout_at

    def snapshot(self) -> _CollectState[V]:
        return copy.deepcopy(self.state)


@operator
def collect(
    step_id: str, up: KeyedStream[V], timeout: timedelta, max_size: int
) -> KeyedStream[List[V]]:
    """Collect items into a list up to a size or a timeout.

    :arg timeout: Timeout before emitting the list, even if `max_size`
        was not reached.
    :arg max_size: Emit the list once it reaches this size, even if
        `timeout` was not reached.

    Example:

    >>> flow = Dataflow("collect_eg")
    >>> s = op.input(
    ...     "inp", flow, TestingSource([("a", 1), ("a", 2), ("a", 3)])
    ... )
    >>> s = op.collect(
    ...     "collect", s, timeout=timedelta(seconds=10), max_size=2
    ... )
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', [1, 2])
    ('a', [3])
    """

    def shim_builder(
        resume_state: Optional[_CollectState[V]],
    ) -> _CollectLogic[V]:
        state = resume_state if resume_state is not None else _CollectState()
        return _CollectLogic(step_id, _get_system_utc, timeout, max_size, state)

    return stateful("stateful", up, shim_builder)


class _FoldFinalLogic(StatefulBatchLogic[V, S, S]):
    """Batch-level fold (the engine's hottest host-path logic:
    count/max/min/reduce_final all route here) — a tight fold loop
    instead of the per-item StatefulLogic shim."""

    def __init__(self, step_id: str, folder: Callable[[S, V], S], state: S):
        self.step_id = step_id
        self.folder = folder
        self.state = state

    def on_batch(self, values: List[V]) -> Tuple[Iterable[S], bool]:
        folder = self.folder
        state = self.state
        for v in values:
            state = folder(state, v)
        self.state = state
        return ((), StatefulBatchLogic.RETAIN)

    def on_item(self, value: V) -> Tuple[Iterable[S], bool]:
        """Per-item compatibility entry (the engine uses on_batch;
        the reference's internals tests drive on_item directly)."""
        self.state = self.folder(self.state, value)
        return ((), StatefulBatchLogic.RETAIN)

    def on_notify(self) -> Tuple[Iterable[S], bool]:
        return ((), StatefulBatchLogic.RETAIN)

    def on_eof(self) -> Tuple[Iterable[S], bool]:
        return ((self.state,), StatefulBatchLogic.DISCARD)

    def notify_at(self) -> Optional[datetime]:
        return None

    def snapshot(self) -> S:
        return copy.deepcopy(self.state)


@operator
def fold_final(
    step_id: str,
    up: KeyedStream[V],
    builder: Callable[[], S],
    folder: Callable[[S, V], S],
) -> KeyedStream[S]:
    """Build an empty accumulator, then combine values into it.

    Only returns results once the upstream is EOF:

    ```python
    folded = op.fold_final("fold", keyed, lambda: 0, lambda acc, v: acc + v)
    # ("key1", 1), ("key1", 2), ("key2", 3) -> ("key1", 3), ("key2", 3)
    ```

    Example:

    >>> flow = Dataflow("fold_final_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("a", 2)]))
    >>> s = op.fold_final("fold", s, list, lambda acc, v: acc + [v])
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', [1, 2])
    """

    def shim_builder(resume_state: Optional[S]) -> _FoldFinalLogic[V, S]:
        state = resume_state if resume_state is not None else builder()
        return _FoldFinalLogic(step_id, folder, state)

    return stateful_batch("stateful_batch", up, shim_builder)


@operator
def reduce_final(
    step_id: str,
    up: KeyedStream[V],
    reducer: Callable[[V, V], V],
) -> KeyedStream[V]:
    """Distill all values for a key down into a single value.

    Like {py:obj}`fold_final` but the first value is the initial
    accumulator.

    Example:

    >>> flow = Dataflow("reduce_final_eg")
    >>> s = op.input(
    ...     "inp", flow, TestingSource([("a", 1), ("a", 2), ("b", 5)])
    ... )
    >>> s = op.reduce_final("sum", s, lambda a, b: a + b)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 3)
    ('b', 5)
    """

    def pre_folder(acc: List[V], v: V) -> List[V]:
        if len(acc) < 1:
            return [v]
        acc[0] = reducer(acc[0], v)
        return acc

    folded = fold_final("fold_final", up, list, pre_folder)

    def extract(acc_v: List[V]) -> V:
        return acc_v[0]

    return map_value("unwrap", folded, extract)


@operator
def count_final(
    step_id: str, up: Stream[X], key: Callable[[X], str]
) -> KeyedStream[int]:
    """Count the number of occurrences of items in the entire stream.

    Only works on finite streams; results are only emitted once the
    upstream is EOF.

    Example:

    >>> flow = Dataflow("count_final_eg")
    >>> s = op.input("inp", flow, TestingSource(["a", "a", "b"]))
    >>> s = op.count_final("count", s, lambda x: x)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 2)
    ('b', 1)
    """
    keyed = map("extract_key", up, lambda x: (key(x), 1))
    return fold_final("count", keyed, int, lambda s, x: s + x)


@operator
def max_final(
    step_id: str,
    up: KeyedStream[V],
    by: Callable[[V], Any] = _identity,
) -> KeyedStream[V]:
    """Find the maximum value for each key; emitted at EOF.

    Example:

    >>> flow = Dataflow("max_final_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("a", 4)]))
    >>> s = op.max_final("max", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 4)
    """
    return reduce_final(
        "reduce_final", up, lambda a, b: a if by(a) >= by(b) else b
    )


@operator
def min_final(
    step_id: str,
    up: KeyedStream[V],
    by: Callable[[V], Any] = _identity,
) -> KeyedStream[V]:
    """Find the minimum value for each key; emitted at EOF.

    Example:

    >>> flow = Dataflow("min_final_eg")
    >>> s = op.input("inp", flow, TestingSource([("a", 1), ("a", 4)]))
    >>> s = op.min_final("min", s)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 1)
    """
    return reduce_final(
        "reduce_final", up, lambda a, b: a if by(a) <= by(b) else b
    )


class _StatefulMapLogic(StatefulLogic[V, W, S]):
    def __init__(
        self,
        step_id: str,
        mapper: Callable[[Optional[S], V], Tuple[Optional[S], Iterable[W]]],
        state: Optional[S],
        single: bool,
    ):
        self.step_id = step_id
        self.mapper = mapper
        self.state = state
        self.single = single

    def on_item(self, value: V) -> Tuple[Iterable[W], bool]:
        res = self.mapper(self.state, value)
        try:
            self.state, out = res
        except (TypeError, ValueError) as ex:
            msg = (
                f"return value of `mapper` {f_repr(self.mapper)} "
                f"in step {self.step_id!r} must be a 2-tuple of "
                f"`(updated_state, emit)`; got a {type(res)!r} instead"
            )
            raise BytewaxTypeError(msg) from ex
        if self.single:
            emitted: Iterable[W] = (out,)  # type: ignore[assignment]
        else:
            emitted = out
        return (emitted, self.state is None)

    def snapshot(self) -> Optional[S]:
        return copy.deepcopy(self.state)


@operator
def stateful_flat_map(
    step_id: str,
    up: KeyedStream[V],
    mapper: Callable[[Optional[S], V], Tuple[Optional[S], Iterable[W]]],
) -> KeyedStream[W]:
    """Transform values one-to-many, referencing a persistent state.

    :arg mapper: Called whenever a value is encountered from upstream
        with the last state or `None`, and then the value.  Should
        return a 2-tuple of `(updated_state, emit_values)`.  If the
        updated state is `None`, discard it.

    Example:

    >>> def dedupe(state, v):
    ...     seen = state or set()
    ...     out = [] if v in seen else [v]
    ...     seen.add(v)
    ...     return (seen, out)
    >>> flow = Dataflow("stateful_flat_map_eg")
    >>> s = op.input(
    ...     "inp", flow, TestingSource([("a", 1), ("a", 1), ("a", 2)])
    ... )
    >>> s = op.stateful_flat_map("dedupe", s, dedupe)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 1)
    ('a', 2)
    """

    def shim_builder(resume_state: Optional[S]) -> _StatefulMapLogic[V, W, S]:
        return _StatefulMapLogic(step_id, mapper, resume_state, single=False)

    return stateful("stateful", up, shim_builder)


@operator
def stateful_map(
    step_id: str,
    up: KeyedStream[V],
    mapper: Callable[[Optional[S], V], Tuple[Optional[S], W]],
) -> KeyedStream[W]:
    """Transform values one-to-one, referencing a persistent state.

    ```python
    def running_sum(state, v):
        state = (state or 0) + v
        return (state, state)

    sums = op.stateful_map("sum", keyed, running_sum)
    # ("a", 1), ("a", 2) -> ("a", 1), ("a", 3)
    ```

    :arg mapper: Called whenever a value is encountered from upstream
        with the last state or `None`, and then the value.  Should
        return a 2-tuple of `(updated_state, emit_value)`.  If the
        updated state is `None`, discard it.

    Example:

    >>> def running_sum(state, v):
    ...     state = (state or 0) + v
    ...     return (state, state)
    >>> flow = Dataflow("stateful_map_eg")
    >>> s = op.input(
    ...     "inp", flow, TestingSource([("a", 1), ("a", 2), ("b", 5)])
    ... )
    >>> s = op.stateful_map("sum", s, running_sum)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('a', 1)
    ('a', 3)
    ('b', 5)
    """

    def shim_builder(resume_state: Optional[S]) -> _StatefulMapLogic[V, W, S]:
        return _StatefulMapLogic(step_id, mapper, resume_state, single=True)

    return stateful("stateful", up, shim_builder)


# ---------------------------------------------------------------------------
# TTL-cache enrichment
# ---------------------------------------------------------------------------


@dataclass
class TTLCache(Generic[DK, DV]):
    """A simple TTL cache around a getter function."""

    v_getter: Callable[[DK], DV]
    now_getter: Callable[[], datetime]
    ttl: timedelta
    _cache: Dict[DK, Tuple[datetime, DV]] = field(default_factory=dict)

    def get(self, k: DK) -> DV:
        """Get the cached value for a key, refreshing if expired."""
        hit = self._cache.get(k)
        if hit is not None:
            asof, v = hit
            if self.now_getter() - asof < self.ttl:
                return v
        v = self.v_getter(k)
        self._cache[k] = (self.now_getter(), v)
        return v

    def remove(self, k: DK) -> None:
        """Remove the cached value for a key."""
        del self._cache[k]


@operator
def enrich_cached(
    step_id: str,
    up: Stream[X],
    getter: Callable[[DK], DV],
    mapper: Callable[[TTLCache[DK, DV], X], Y],
    ttl: timedelta = timedelta.max,
    _now_getter: Callable[[], datetime] = _get_system_utc,
) -> Stream[Y]:
    """Enrich / join items using a cached lookup to an external service.

    Example:

    >>> def lookup(k):
    ...     return {"1": "alice", "2": "bob"}[k]
    >>> flow = Dataflow("enrich_eg")
    >>> s = op.input("inp", flow, TestingSource(["1", "2", "1"]))
    >>> s = op.enrich_cached(
    ...     "names", s, lookup, lambda cache, k: cache.get(k)
    ... )
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    alice
    bob
    alice
    """
    cache = TTLCache(getter, _now_getter, ttl)

    def shim_mapper(xs: Iterable[X]) -> Iterable[Y]:
        return [mapper(cache, x) for x in xs]

    return flat_map_batch("flat_map_batch", up, shim_mapper)


# ---------------------------------------------------------------------------
# Joins
# ---------------------------------------------------------------------------

#: How to handle multiple values for a side between emits.
JoinInsertMode = typing.Literal["first", "last", "product"]
#: When to emit the joined row.
JoinEmitMode = typing.Literal["complete", "final", "running"]

_UNSET = object()


@dataclass
class _JoinState:
    seen: List[List[Any]]

    @classmethod
    def for_side_count(cls, side_count: int) -> "_JoinState":
        return cls(seen=[[] for _ in range(side_count)])

    def set_val(self, side: int, value: Any) -> None:
        self.seen[side] = [value]

    def add_val(self, side: int, value: Any) -> None:
        self.seen[side].append(value)

    def is_set(self, side: int) -> bool:
        return len(self.seen[side]) > 0

    def all_set(self) -> bool:
        return all(len(s) > 0 for s in self.seen)

    def astuples(self) -> List[Tuple]:
        import itertools

        sides = [s if len(s) > 0 else [None] for s in self.seen]
        return list(itertools.product(*sides))

    def clear(self) -> None:
        self.seen = [[] for _ in self.seen]


class _JoinLogic(StatefulLogic[Tuple[int, Any], Tuple, _JoinState]):
    def __init__(
        self,
        insert_mode: str,
        emit_mode: str,
        state: _JoinState,
    ):
        self.insert_mode = insert_mode
        self.emit_mode = emit_mode
        self.state = state

    def on_item(self, value: Tuple[int, Any]) -> Tuple[Iterable[Tuple], bool]:
        side, v = value
        if self.insert_mode == "first":
            if not self.state.is_set(side):
                self.state.set_val(side, v)
        elif self.insert_mode == "last":
            self.state.set_val(side, v)
        else:  # product
            self.state.add_val(side, v)

        if self.emit_mode == "running":
            return (self.state.astuples(), StatefulLogic.RETAIN)
        if self.emit_mode == "complete" and self.state.all_set():
            out = self.state.astuples()
            self.state.clear()
            return (out, StatefulLogic.DISCARD)
        return ((), StatefulLogic.RETAIN)

    def on_eof(self) -> Tuple[Iterable[Tuple], bool]:
        if self.emit_mode == "final":
            return (self.state.astuples(), StatefulLogic.DISCARD)
        return ((), StatefulLogic.RETAIN)

    def snapshot(self) -> _JoinState:
        return copy.deepcopy(self.state)


@operator
def _join_label_merge(
    step_id: str,
    *sides: KeyedStream[Any],
) -> KeyedStream[Tuple[int, Any]]:
    """Label values with their side index and merge the streams."""
    labeled = [
        map_value(f"label_{i}", side, lambda v, _i=i: (_i, v))
        for i, side in enumerate(sides)
    ]
    return merge("merge", *labeled)


@operator
def join(
    step_id: str,
    *sides: KeyedStream[Any],
    insert_mode: JoinInsertMode = "last",
    emit_mode: JoinEmitMode = "complete",
) -> KeyedStream[Tuple]:
    """Gather together the value for a key on multiple streams.

    :arg insert_mode: "first" keeps the first value per side, "last"
        the most recent, "product" all of them.
    :arg emit_mode: "complete" emits once all sides have a value and
        resets; "final" emits at EOF; "running" emits on every item.

    Example:

    >>> flow = Dataflow("join_eg")
    >>> names = op.input("n", flow, TestingSource([("1", "alice")]))
    >>> emails = op.input("e", flow, TestingSource([("1", "a@x.io")]))
    >>> s = op.join("join", names, emails)
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('1', ('alice', 'a@x.io'))

    With ``emit_mode="running"`` every update emits the current
    (possibly incomplete) tuple:

    >>> flow = Dataflow("join_running_eg")
    >>> a = op.input("a", flow, TestingSource([("1", "x1"), ("1", "x2")]))
    >>> b = op.input("b", flow, TestingSource([("1", "y1")]))
    >>> s = op.join("join", a, b, emit_mode="running")
    >>> op.output("out", s, StdOutSink())
    >>> run_main(flow)
    ('1', ('x1', None))
    ('1', ('x1', 'y1'))
    ('1', ('x2', 'y1'))
    """
    if insert_mode not in typing.get_args(JoinInsertMode):
        msg = f"unknown join insert mode {insert_mode!r}"
        raise ValueError(msg)
    if emit_mode not in typing.get_args(JoinEmitMode):
        msg = f"unknown join emit mode {emit_mode!r}"
        raise ValueError(msg)

    side_count = len(sides)

    def shim_builder(
        resume_state: Optional[_JoinState],
    ) -> _JoinLogic:
        state = (
            resume_state
            if resume_state is not None
            else _JoinState.for_side_count(side_count)
        )
        return _JoinLogic(insert_mode, emit_mode, state)

    merged = _join_label_merge("add_names", *sides)
    return stateful("join", merged, shim_builder)
