"""Low-level input interfaces and implementation helpers.

API parity target: ``bytewax.inputs`` (reference pysrc/bytewax/
inputs.py:57-629).  Sources are polled cooperatively by the engine's
worker loop; partitions report a ``next_awake`` time so idle sources
cost nothing.
"""

import asyncio
import queue
from abc import ABC, abstractmethod
from datetime import datetime, timedelta, timezone
from typing import (
    Any,
    AsyncIterator,
    Callable,
    Generic,
    Iterable,
    Iterator,
    List,
    Optional,
    TypeVar,
)

X = TypeVar("X")
S = TypeVar("S")

__all__ = [
    "AbortExecution",
    "DynamicSource",
    "FixedPartitionedSource",
    "Source",
    "SimplePollingSource",
    "StatefulSourcePartition",
    "StatelessSourcePartition",
    "batch",
    "batch_async",
    "batch_getter",
    "batch_getter_ex",
]


class AbortExecution(RuntimeError):
    """Raise this from any source to abort the execution immediately.

    No state for the current epoch is snapshotted, so resuming replays
    the aborted epoch.  Used for fault-injection testing.
    """


class Source(ABC, Generic[X]):
    """A location to read input items from.

    Base class for all input sources.  Do not subclass this directly;
    use one of the subclasses below.
    """

    def __json__(self):
        return {"type": type(self).__name__}


class StatefulSourcePartition(ABC, Generic[X, S]):
    """Input partition that maintains recoverable state of its position."""

    @abstractmethod
    def next_batch(self) -> Iterable[X]:
        """Attempt to get the next batch of input items.

        This must never block; return an empty list if no items are
        ready.  Raise `StopIteration` when the partition is complete
        for this execution.
        """
        ...

    def next_awake(self) -> Optional[datetime]:
        """Next system time this partition should be polled, if known."""
        return None

    @abstractmethod
    def snapshot(self) -> S:
        """Snapshot the position of the next read of this partition."""
        ...

    def close(self) -> None:
        """Cleanup when the execution completes."""
        return


class FixedPartitionedSource(Source[X], Generic[X, S]):
    """An input source with a fixed number of independent partitions."""

    @abstractmethod
    def list_parts(self) -> List[str]:
        """List all local partitions this worker has access to."""
        ...

    @abstractmethod
    def build_part(
        self,
        step_id: str,
        for_part: str,
        resume_state: Optional[S],
    ) -> StatefulSourcePartition[X, S]:
        """Build anew or resume an input partition."""
        ...


class StatelessSourcePartition(ABC, Generic[X]):
    """Input partition that is not recoverable."""

    @abstractmethod
    def next_batch(self) -> Iterable[X]:
        """Attempt to get the next batch of input items.

        Must never block; raise `StopIteration` when complete.
        """
        ...

    def next_awake(self) -> Optional[datetime]:
        """Next system time this partition should be polled, if known."""
        return None

    def close(self) -> None:
        """Cleanup when the execution completes."""
        return


class DynamicSource(Source[X]):
    """An input source where all workers can read distinct items.

    Each worker builds its own partition via
    {py:obj}`DynamicSource.build`.
    """

    @abstractmethod
    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> StatelessSourcePartition[X]:
        """Build an input partition for a worker."""
        ...


class _SimplePollingPartition(StatefulSourcePartition[X, S]):
    """Drift-free polling partition (reference inputs.py:285-330
    contract: `now` injected for deterministic tests; awakes advance
    from the SCHEDULED time, not from the current clock; an exactly
    on-mark `align_to` activates immediately)."""

    def __init__(
        self,
        now: datetime,
        interval: timedelta,
        align_to: Optional[datetime],
        getter: Callable[[], X],
        snapshot: Callable[[], Any] = lambda: None,
    ):
        self._interval = interval
        self._getter = getter
        self._snapshot = snapshot
        if align_to is not None:
            since_last_awake = (now - align_to) % interval
            if since_last_awake > timedelta(seconds=0):
                until_next_awake = interval - since_last_awake
            else:
                until_next_awake = timedelta(seconds=0)
            self._next_awake = now + until_next_awake
        else:
            self._next_awake = now

    def next_batch(self) -> List[X]:
        try:
            item = self._getter()
            self._next_awake += self._interval
            if item is None:
                return []
            return [item]
        except SimplePollingSource.Retry as ex:
            self._next_awake += ex.timeout
            return []

    def next_awake(self) -> Optional[datetime]:
        return self._next_awake

    def snapshot(self) -> Any:
        return self._snapshot()


class SimplePollingSource(FixedPartitionedSource[X, S]):
    """Calls a user-defined function at a regular interval.

    Subclass and override {py:obj}`next_item`.  Only one worker polls.

    Example:
        >>> from datetime import timedelta
        >>> from bytewax_amd.inputs import SimplePollingSource
        >>> class Counter(SimplePollingSource):
        ...     n = 0
        ...     def next_item(self):
        ...         Counter.n += 1
        ...         return Counter.n
        >>> src = Counter(timedelta(seconds=10))
        >>> src.list_parts()
        ['singleton']
        >>> part = src.build_part("input", "singleton", None)
        >>> part.next_batch()
        [1]
    """

    class Retry(Exception):
        """Raise from `next_item` to retry after a timeout."""

        def __init__(self, timeout: timedelta):
            self.timeout = timeout

    def __init__(
        self,
        interval: timedelta,
        align_to: Optional[datetime] = None,
    ):
        self._interval = interval
        self._align_to = align_to

    def list_parts(self) -> List[str]:
        return ["singleton"]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[None]
    ) -> "_SimplePollingPartition[X, None]":
        return _SimplePollingPartition(
            datetime.now(timezone.utc),
            self._interval,
            self._align_to,
            self.next_item,
        )

    @abstractmethod
    def next_item(self) -> X:
        """Called at the regular interval; return the next item or
        `None` for nothing, or raise {py:obj}`SimplePollingSource.Retry`
        to back off."""
        ...


def batch(ib: Iterable[X], batch_size: int) -> Iterator[List[X]]:
    """Batch an iterable into fixed-size lists.

    Use this to build `next_batch` for a partition over a plain
    iterator (reference documents the same helper on
    pysrc/bytewax/inputs.py).

    Example:
        >>> from bytewax_amd.inputs import batch
        >>> list(batch(range(5), 2))
        [[0, 1], [2, 3], [4]]
    """
    it = iter(ib)
    while True:
        out: List[X] = []
        for _ in range(batch_size):
            try:
                out.append(next(it))
            except StopIteration:
                if out:
                    yield out
                return
        yield out


def batch_getter(
    getter: Callable[[], X], batch_size: int, yield_on: Any = None
) -> Iterator[List[X]]:
    """Batch from a getter function that returns a sentinel when
    there are no items YET; the getter raises `StopIteration` at
    EOF, which yields the final partial batch and ends the iterator
    (reference inputs.py:477-510 contract).

    Example:
        >>> from bytewax_amd.inputs import batch_getter
        >>> q = [1, 2, 3]
        >>> def poll():
        ...     return q.pop(0) if q else None
        >>> it = batch_getter(poll, 2)
        >>> next(it), next(it), next(it)
        ([1, 2], [3], [])
    """
    while True:
        out: List[X] = []
        while len(out) < batch_size:
            try:
                item = getter()
            except StopIteration:
                yield out
                return
            if item == yield_on:
                break
            out.append(item)
        yield out


def batch_getter_ex(
    getter: Callable[[], X], batch_size: int, yield_ex: type = queue.Empty
) -> Iterator[List[X]]:
    """Batch from a getter that raises `yield_ex` (default
    `queue.Empty`) when there are no items YET and `StopIteration`
    at EOF — the final partial batch is yielded, then the iterator
    ends (reference inputs.py:513-547 contract).

    Example:
        >>> import queue
        >>> from bytewax_amd.inputs import batch_getter_ex
        >>> q = [1, 2, 3]
        >>> def poll():
        ...     if not q:
        ...         raise queue.Empty()
        ...     return q.pop(0)
        >>> it = batch_getter_ex(poll, 2)
        >>> next(it), next(it), next(it)
        ([1, 2], [3], [])
    """
    while True:
        out: List[X] = []
        while len(out) < batch_size:
            try:
                out.append(getter())
            except yield_ex:
                break
            except StopIteration:
                yield out
                return
        yield out


def batch_async(
    aib: AsyncIterator[X],
    timeout: timedelta,
    batch_size: int,
    loop: Optional[asyncio.AbstractEventLoop] = None,
) -> Iterator[List[X]]:
    """Batch an async iterator from a sync context.

    Each advance collects up to `batch_size` items, waiting at most
    `timeout` total; yields possibly-empty batches until the async
    iterator is exhausted.

    Example:
        >>> from datetime import timedelta
        >>> from bytewax_amd.inputs import batch_async
        >>> async def gen():
        ...     for i in range(3):
        ...         yield i
        >>> list(batch_async(gen(), timedelta(seconds=1), 2))
        [[0, 1], [2]]
    """
    loop = loop if loop is not None else asyncio.new_event_loop()

    task: Optional[asyncio.Task] = None

    async def anext_batch() -> List[X]:
        nonlocal task
        out: List[X] = []
        deadline = loop.time() + timeout.total_seconds()
        while len(out) < batch_size:
            if task is None:
                task = loop.create_task(aib.__anext__())
            budget = deadline - loop.time()
            if budget <= 0:
                break
            try:
                item = await asyncio.wait_for(asyncio.shield(task), budget)
            except asyncio.TimeoutError:
                break
            except StopAsyncIteration:
                task = None
                if out:
                    return out
                raise
            task = None
            out.append(item)
        return out

    while True:
        try:
            yield loop.run_until_complete(anext_batch())
        except StopAsyncIteration:
            return
