"""Helper tools for testing dataflows.

API parity target: ``bytewax.testing`` (reference pysrc/bytewax/
testing.py:38-383): `TestingSource` with EOF/ABORT/PAUSE sentinel
items, `TestingSink`, `TimeTestingGetter`, `poll_next_batch`, and the
`python -m bytewax_amd.testing` multi-process cluster launcher.
"""

import argparse
import time
from dataclasses import dataclass
from datetime import datetime, timedelta, timezone
from itertools import islice
from typing import Any, Iterable, Iterator, List, Optional, TypeVar

from ._engine import cluster_main, run_main
from .inputs import (
    AbortExecution,
    FixedPartitionedSource,
    StatefulSourcePartition,
)
from .outputs import DynamicSink, StatelessSinkPartition

X = TypeVar("X")

__all__ = [
    "TestingSink",
    "TestingSource",
    "TimeTestingGetter",
    "cluster_main",
    "ffwd_iter",
    "poll_next_batch",
    "run_main",
]


@dataclass
class TimeTestingGetter:
    """Wrapper to provide a modifiable system clock for unit tests."""

    now: datetime

    def advance(self, td: timedelta) -> None:
        """Advance the current time by this amount."""
        self.now += td

    def get(self) -> datetime:
        """Return the "current time"."""
        return self.now


def ffwd_iter(it: Iterator[Any], n: int) -> None:
    """Skip an iterator forward some number of items."""
    next(islice(it, n, n), None)


class _IterSourcePartition(StatefulSourcePartition[X, int]):
    def __init__(
        self,
        ib: Iterable[Any],
        batch_size: int,
        resume_state: Optional[int],
    ):
        self._start_idx = 0 if resume_state is None else resume_state
        self._batch_size = batch_size
        self._next_awake: Optional[datetime] = None
        self._it = iter(ib)
        # Resume to one after the last completed read index.
        ffwd_iter(self._it, self._start_idx)
        self._raise: Optional[Exception] = None

    def next_batch(self) -> List[X]:
        if self._raise is not None:
            raise self._raise
        self._next_awake = None

        batch: List[X] = []
        batch_append = batch.append
        batch_size = self._batch_size
        sentinels = _SENTINELS
        for item in self._it:
            # Hot path: one combined isinstance for the three
            # sentinel classes instead of three checks per item.
            if not isinstance(item, sentinels):
                batch_append(item)
                if len(batch) >= batch_size:
                    break
            elif isinstance(item, TestingSource.EOF):
                self._raise = StopIteration()
                # Skip over this on continuation.
                self._start_idx += 1
                break
            elif isinstance(item, TestingSource.ABORT):
                if not item._triggered:
                    self._raise = AbortExecution()
                    # Only trigger once; skipped on resume executions.
                    item._triggered = True
                    break
            else:  # PAUSE
                self._next_awake = datetime.now(tz=timezone.utc) + item.for_duration
                self._start_idx += 1
                break

        if len(batch) > 0 or self._raise is not None or self._next_awake is not None:
            self._start_idx += len(batch)
            return batch
        raise StopIteration()

    def next_awake(self) -> Optional[datetime]:
        return self._next_awake

    def snapshot(self) -> int:
        return self._start_idx


class TestingSource(FixedPartitionedSource[X, int]):
    """Produce input from a Python iterable.

    Only use this for unit testing; the iterable must be identical on
    all workers.  There is no parallelism; only one worker consumes the
    iterable.
    """

    __test__ = False

    @dataclass
    class EOF:
        """Signal the input to EOF; the next execution continues from
        the item after this."""

    @dataclass
    class ABORT:
        """Abort the execution when the input processes this item.

        Each abort only triggers once; it is skipped on resume
        executions.  Can't be used in multi-worker executions.
        """

        _triggered: bool = False

    @dataclass
    class PAUSE:
        """Signal this input to not emit items for a duration."""

        for_duration: timedelta

    def __init__(self, ib: Iterable[Any], batch_size: int = 1):
        self._ib = ib
        self._batch_size = batch_size

    def list_parts(self) -> List[str]:
        return ["iterable"]

    def build_part(
        self, step_id: str, for_part: str, resume_state: Optional[int]
    ) -> _IterSourcePartition[X]:
        return _IterSourcePartition(self._ib, self._batch_size, resume_state)


#: Sentinel classes checked once per item on the hot read path.
_SENTINELS = (TestingSource.EOF, TestingSource.ABORT, TestingSource.PAUSE)


class _ListSinkPartition(StatelessSinkPartition[X]):
    def __init__(self, ls: List[X]):
        self._ls = ls

    def write_batch(self, items: List[X]) -> None:
        self._ls += items


class TestingSink(DynamicSink[X]):
    """Append each output item to a list.

    Only use this for unit testing.  The list is not cleared between
    executions.
    """

    __test__ = False

    def __init__(self, ls: List[X]):
        self._ls = ls

    def build(
        self, step_id: str, worker_index: int, worker_count: int
    ) -> _ListSinkPartition[X]:
        return _ListSinkPartition(self._ls)


def poll_next_batch(part, timeout: timedelta = timedelta(seconds=5)) -> List[Any]:
    """Repeatedly poll a source partition until it returns a batch.

    Raises `TimeoutError` on timeout.
    """
    deadline = time.monotonic() + timeout.total_seconds()
    batch: List[Any] = []
    while len(batch) <= 0:
        if time.monotonic() > deadline:
            raise TimeoutError()
        batch = list(part.next_batch())
    return batch


def _parse_args():
    parser = argparse.ArgumentParser(
        prog="python -m bytewax_amd.testing",
        description="Run a dataflow as a local multi-process cluster for testing.",
    )
    parser.add_argument("import_str", help="dataflow import string")
    parser.add_argument(
        "-p", "--processes", type=int, default=1, help="process count"
    )
    parser.add_argument(
        "-w", "--workers-per-process", type=int, default=1, help="workers per process"
    )
    parser.add_argument("-r", "--recovery-directory", type=None, default=None)
    parser.add_argument(
        "-s",
        "--snapshot-interval",
        type=float,
        default=None,
        help="snapshot interval (sec)",
    )
    parser.add_argument(
        "-b",
        "--backup-interval",
        type=float,
        default=None,
        help="backup interval (sec)",
    )
    return parser.parse_args()


def _main() -> None:
    import subprocess
    import sys

    args = _parse_args()
    addresses = [f"127.0.0.1:{2101 + i}" for i in range(args.processes)]
    procs = []
    for i in range(args.processes):
        cmd = [
            sys.executable,
            "-m",
            "bytewax_amd.run",
            args.import_str,
            "-w",
            str(args.workers_per_process),
            "-i",
            str(i),
            "-a",
            ";".join(addresses),
        ]
        if args.recovery_directory is not None:
            cmd += ["-r", str(args.recovery_directory)]
        if args.snapshot_interval is not None:
            cmd += ["-s", str(args.snapshot_interval)]
        if args.backup_interval is not None:
            cmd += ["-b", str(args.backup_interval)]
        procs.append(subprocess.Popen(cmd))
    codes = [p.wait() for p in procs]
    if any(codes):
        sys.exit(max(codes))


if __name__ == "__main__":
    _main()
