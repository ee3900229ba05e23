"""Connectors for console IO.

Parity target: ``bytewax.connectors.stdio`` (reference
connectors/stdio.py:20-37).
"""

from typing import Any, List

from ..outputs import DynamicSink, StatelessSinkPartition

__all__ = ["StdOutSink"]


class _PrintSinkPartition(StatelessSinkPartition[Any]):
    def write_batch(self, items: List[Any]) -> None:
        for item in items:
            print(item, flush=True)


class StdOutSink(DynamicSink[Any]):
    """Write each output item to stdout on that worker."""

    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> _PrintSinkPartition:
        return _PrintSinkPartition()
