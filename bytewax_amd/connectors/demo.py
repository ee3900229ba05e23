"""Demo connectors for tutorials and experimentation.

Parity target: ``bytewax.connectors.demo`` (reference
connectors/demo.py:51-103).
"""

import random
from dataclasses import dataclass, field
from datetime import datetime, timedelta, timezone
from typing import List, Optional, Tuple

from ..inputs import FixedPartitionedSource, StatefulSourcePartition

__all__ = ["RandomMetricSource"]

_State = Tuple[int, float]


class _RandomMetricPartition(
    StatefulSourcePartition[Tuple[str, float], _State]
):
    def __init__(
        self,
        metric_name: str,
        interval: timedelta,
        count: int,
        next_random,
        resume_state: Optional[_State],
    ):
        self._metric_name = metric_name
        self._interval = interval
        self._count = count
        self._next_random = next_random
        self._i, _ = resume_state if resume_state is not None else (0, 0.0)
        self._next_awake = datetime.now(timezone.utc)

    def next_batch(self) -> List[Tuple[str, float]]:
        if self._i >= self._count:
            raise StopIteration()
        self._i += 1
        value = self._next_random()
        self._next_awake = datetime.now(timezone.utc) + self._interval
        return [(self._metric_name, value)]

    def next_awake(self) -> Optional[datetime]:
        return self._next_awake

    def snapshot(self) -> _State:
        return (self._i, 0.0)


@dataclass
class RandomMetricSource(FixedPartitionedSource[Tuple[str, float], _State]):
    """Demo source of random metric values `(metric_name, val)` at a
    regular interval.

    :arg metric_name: To attach to each value.
    :arg interval: Emit a value on this cadence.
    :arg count: Number of values to generate.
    :arg next_random: Callable generating the next value.
    """

    metric_name: str
    interval: timedelta = timedelta(seconds=0.7)
    count: int = 20
    next_random: object = field(
        default_factory=lambda: (lambda: random.randrange(0, 10))
    )

    def list_parts(self) -> List[str]:
        return [self.metric_name]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[_State]
    ) -> _RandomMetricPartition:
        return _RandomMetricPartition(
            self.metric_name,
            self.interval,
            self.count,
            self.next_random,
            resume_state,
        )
