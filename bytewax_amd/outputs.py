"""Low-level output interfaces and implementation helpers.

API parity target: ``bytewax.outputs`` (reference pysrc/bytewax/
outputs.py:19-214).
"""

import zlib
from abc import ABC, abstractmethod
from typing import Any, Generic, List, Optional, TypeVar

X = TypeVar("X")
S = TypeVar("S")

__all__ = [
    "DynamicSink",
    "FixedPartitionedSink",
    "Sink",
    "StatefulSinkPartition",
    "StatelessSinkPartition",
]


class Sink(ABC, Generic[X]):
    """A location to write output items to.

    Base class for all output sinks.  Do not subclass this directly;
    use one of the subclasses below.
    """

    def __json__(self):
        return {"type": type(self).__name__}


class StatefulSinkPartition(ABC, Generic[X, S]):
    """Output partition that maintains recoverable state of its position."""

    @abstractmethod
    def write_batch(self, values: List[X]) -> None:
        """Write a batch of output values; must not return until
        durably written."""
        ...

    @abstractmethod
    def snapshot(self) -> S:
        """Snapshot the position of the next write of this partition."""
        ...

    def close(self) -> None:
        """Cleanup when the execution completes."""
        return


class FixedPartitionedSink(Sink[Any], Generic[X, S]):
    """An output sink with a fixed number of independent partitions.

    Items are ``(key, value)`` 2-tuples; the key is used to route each
    value to a partition via {py:obj}`part_fn`.
    """

    @abstractmethod
    def list_parts(self) -> List[str]:
        """List all local partitions this worker has access to."""
        ...

    def part_fn(self, item_key: str) -> int:
        """Route incoming `(key, value)` pairs to partitions.

        The default hash is `zlib.adler32`, which is consistent across
        processes (unlike Python's builtin `hash`).  The return value
        is wrapped modulo the partition count.

        Example:
            >>> from bytewax_amd.outputs import FixedPartitionedSink
            >>> FixedPartitionedSink.part_fn(None, "user-1") % 3
            1
        """
        return zlib.adler32(item_key.encode())

    @abstractmethod
    def build_part(
        self,
        step_id: str,
        for_part: str,
        resume_state: Optional[S],
    ) -> StatefulSinkPartition[X, S]:
        """Build anew or resume an output partition."""
        ...


class StatelessSinkPartition(ABC, Generic[X]):
    """Output partition that is not recoverable."""

    @abstractmethod
    def write_batch(self, items: List[X]) -> None:
        """Write a batch of output items."""
        ...

    def close(self) -> None:
        """Cleanup when the execution completes."""
        return


class DynamicSink(Sink[X]):
    """An output sink where all workers write items concurrently.

    Each worker builds its own partition via
    {py:obj}`DynamicSink.build`.
    """

    @abstractmethod
    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> StatelessSinkPartition[X]:
        """Build an output partition for a worker."""
        ...
