"""Failure recovery: epoch-consistent snapshots in SQLite partitions.

Format parity target: the reference's recovery DB (reference
src/recovery.rs:456-513) — N ``part-{i}.sqlite3`` files, WAL mode, five
tables ``parts`` / ``exs`` / ``fronts`` / ``commits`` / ``snaps``;
snapshot rows are ``(step_id, state_key, snap_epoch, ser_change)`` with
``ser_change`` = pickled state for an upsert and NULL for a discard
(reference recovery.rs:1529-1616).

Run ``python -m bytewax_amd.recovery db_dir N`` to create a new set of
recovery partitions before running a dataflow with recovery enabled.
"""

import argparse
import pickle
import sqlite3
import threading
from dataclasses import dataclass
from datetime import timedelta
from pathlib import Path
from typing import Any, Dict, Iterable, List, Optional, Tuple

import zlib

__all__ = [
    "InconsistentPartitionsError",
    "MissingPartitionsError",
    "NoPartitionsError",
    "RecoveryConfig",
    "init_db_dir",
]


class NoPartitionsError(FileNotFoundError):
    """Raised when no recovery partitions are found on any worker."""


class MissingPartitionsError(FileNotFoundError):
    """Raised when an incomplete set of recovery partitions is found."""


class InconsistentPartitionsError(ValueError):
    """Raised when recovery partitions have been GC'd past the resume
    epoch and resuming would lose data."""


_SCHEMA = [
    """CREATE TABLE IF NOT EXISTS parts (
       created_at TEXT NOT NULL DEFAULT CURRENT_TIMESTAMP,
       part_index INTEGER PRIMARY KEY NOT NULL CHECK (part_index >= 0),
       part_count INTEGER NOT NULL CHECK (part_count > 0),
       CHECK (part_index < part_count)
       )""",
    """CREATE TABLE IF NOT EXISTS exs (
       created_at TEXT NOT NULL DEFAULT CURRENT_TIMESTAMP,
       ex_num INTEGER NOT NULL PRIMARY KEY,
       worker_count INTEGER NOT NULL CHECK (worker_count > 0),
       resume_epoch INTEGER NOT NULL
       )""",
    """CREATE TABLE IF NOT EXISTS fronts (
       created_at TEXT NOT NULL DEFAULT CURRENT_TIMESTAMP,
       ex_num INTEGER NOT NULL,
       worker_index INTEGER NOT NULL CHECK (worker_index >= 0),
       worker_frontier INTEGER NOT NULL,
       PRIMARY KEY (ex_num, worker_index)
       )""",
    """CREATE TABLE IF NOT EXISTS commits (
       created_at TEXT NOT NULL DEFAULT CURRENT_TIMESTAMP,
       part_index INTEGER PRIMARY KEY NOT NULL,
       commit_epoch INTEGER NOT NULL
       )""",
    """CREATE TABLE IF NOT EXISTS snaps (
       created_at TEXT NOT NULL DEFAULT CURRENT_TIMESTAMP,
       step_id TEXT NOT NULL,
       state_key TEXT NOT NULL,
       snap_epoch INTEGER NOT NULL,
       ser_change BLOB,
       PRIMARY KEY (step_id, state_key, snap_epoch)
       )""",
]


def _open_conn(path: Path) -> sqlite3.Connection:
    conn = sqlite3.connect(str(path), timeout=5.0, check_same_thread=False)
    conn.execute("PRAGMA journal_mode = WAL")
    conn.execute("PRAGMA busy_timeout = 5000")
    conn.execute("PRAGMA foreign_keys = ON")
    return conn


def init_db_dir(db_dir: Path, count: int) -> List[Path]:
    """Create and init a set of recovery partitions.

    :arg db_dir: Local directory to create partitions in.
    :arg count: Number of partitions to create.
    :returns: All the file names created.
    """
    db_dir = Path(db_dir)
    if not db_dir.is_dir():
        msg = f"recovery directory {db_dir} does not exist"
        raise FileNotFoundError(msg)
    paths = []
    for i in range(count):
        path = db_dir / f"part-{i}.sqlite3"
        conn = _open_conn(path)
        with conn:
            for stmt in _SCHEMA:
                conn.execute(stmt)
            conn.execute(
                "INSERT OR IGNORE INTO parts (part_index, part_count) "
                "VALUES (?, ?)",
                (i, count),
            )
        conn.close()
        paths.append(path)
    return paths


@dataclass(frozen=True)
class RecoveryConfig:
    """Configuration settings for recovery.

    :arg db_dir: Directory containing pre-initialized recovery
        partitions (see {py:obj}`init_db_dir`).
    :arg backup_interval: Amount of system time to wait to permanently
        delete a state snapshot after it is no longer needed, to allow
        off-machine backups of the partition files to stay mutually
        consistent.  Defaults to zero.
    """

    db_dir: Path
    backup_interval: Optional[timedelta] = None
    # Set to an object with a `.backup(path)` method to hook backups.
    backup: Any = None


def _epochs_per(duration: timedelta, epoch_interval: timedelta) -> int:
    """Number of epochs that fully cover `duration` (reference
    inputs.rs:64-77)."""
    if duration <= timedelta(0):
        return 0
    count, rem = divmod(duration, epoch_interval)
    return int(count) + (1 if rem > timedelta(0) else 0)


def _route_part(step_id: str, state_key: str, part_count: int) -> int:
    return zlib.adler32(f"{step_id}\x00{state_key}".encode()) % part_count


class RecoveryStore:
    """Engine-side handle over a set of recovery partitions.

    All worker threads of one process funnel through this object; a
    lock serializes SQLite access (the reference serializes per
    partition through owning workers instead — behaviorally equivalent
    in-process).
    """

    def __init__(self, config: RecoveryConfig, epoch_interval: timedelta):
        self.config = config
        db_dir = Path(config.db_dir)
        if not db_dir.is_dir():
            msg = f"recovery directory {db_dir} does not exist"
            raise NoPartitionsError(msg)
        self.paths = sorted(db_dir.glob("*.sqlite3"))
        if not self.paths:
            msg = (
                "No recovery partitions found on any worker; can't resume"
            )
            raise NoPartitionsError(msg)
        self.conns = [_open_conn(p) for p in self.paths]
        self.lock = threading.Lock()
        backup = config.backup_interval or timedelta(0)
        self.backup_delay_epochs = _epochs_per(backup, epoch_interval)

        # Validate the partition census.
        counts = set()
        found = set()
        for conn in self.conns:
            for part_index, part_count in conn.execute(
                "SELECT part_index, part_count FROM parts"
            ):
                counts.add(part_count)
                found.add(part_index)
        if not counts:
            msg = "No recovery partitions found on any worker; can't resume"
            raise NoPartitionsError(msg)
        if len(counts) > 1:
            msg = (
                "Inconsistent partition counts in recovery partitions; "
                "can't resume"
            )
            raise ValueError(msg)
        self.part_count = counts.pop()
        missing = set(range(self.part_count)) - found
        if missing:
            msg = (
                f"Missing recovery partitions {sorted(missing)} of "
                f"{self.part_count}; can't resume"
            )
            raise MissingPartitionsError(msg)
        # part_index -> connection holding it
        self.part_conns: Dict[int, sqlite3.Connection] = {}
        for conn in self.conns:
            for (part_index,) in conn.execute("SELECT part_index FROM parts"):
                self.part_conns[part_index] = conn

    def close(self) -> None:
        for conn in self.conns:
            conn.close()

    # -- resume calculation (reference recovery.rs:1180-1275) --

    def resume_from(self) -> Tuple[int, int]:
        """Compute `(ex_num, resume_epoch)` for the next execution."""
        rows: Dict[Tuple[int, int], int] = {}
        max_ex: Optional[Tuple[int, int, int]] = None
        with self.lock:
            for conn in self.conns:
                for ex_num, worker_count, resume_epoch in conn.execute(
                    "SELECT ex_num, worker_count, resume_epoch FROM exs"
                ):
                    if max_ex is None or ex_num > max_ex[0]:
                        max_ex = (ex_num, worker_count, resume_epoch)
            if max_ex is None:
                return (0, 1)  # default: first execution, epoch 1
            ex_num, worker_count, default_epoch = max_ex
            # Default frontier for every worker of that execution.
            progress = {w: default_epoch for w in range(worker_count)}
            for conn in self.conns:
                for w, f in conn.execute(
                    "SELECT worker_index, MAX(worker_frontier) FROM fronts "
                    "WHERE ex_num = ? GROUP BY worker_index",
                    (ex_num,),
                ):
                    if w in progress:
                        progress[w] = max(progress[w], f)
                    else:
                        progress[w] = f
            resume_epoch = min(progress.values())
            # Consistency: no partition may have GC'd past the resume
            # epoch.
            stale = []
            for conn in self.conns:
                for (part_index,) in conn.execute(
                    "SELECT part_index FROM commits WHERE commit_epoch > ?",
                    (resume_epoch,),
                ):
                    stale.append(part_index)
            if stale:
                ok = sorted(
                    set(range(self.part_count)) - set(stale)
                )
                msg = (
                    f"Recovery partitions {ok} of {self.part_count} are too "
                    f"old to resume from epoch {resume_epoch} without data "
                    "loss; do you have a newer backup of these partitions?"
                )
                raise InconsistentPartitionsError(msg)
            return (ex_num + 1, resume_epoch)

    # -- writers --

    def write_ex(self, ex_num: int, worker_count: int, resume_epoch: int) -> None:
        part = ex_num % self.part_count
        with self.lock:
            conn = self.part_conns[part]
            with conn:
                conn.execute(
                    "INSERT OR REPLACE INTO exs "
                    "(ex_num, worker_count, resume_epoch) VALUES (?, ?, ?)",
                    (ex_num, worker_count, resume_epoch),
                )

    def write_snaps(
        self,
        snaps: Iterable[Tuple[str, str, int, Optional[bytes]]],
    ) -> None:
        """Write `(step_id, state_key, snap_epoch, ser_change)` rows."""
        by_part: Dict[int, List[Tuple]] = {}
        for step_id, state_key, snap_epoch, ser_change in snaps:
            p = _route_part(step_id, state_key, self.part_count)
            by_part.setdefault(p, []).append(
                (step_id, state_key, snap_epoch, ser_change)
            )
        with self.lock:
            for p, rows in by_part.items():
                conn = self.part_conns[p]
                with conn:
                    conn.executemany(
                        "INSERT OR REPLACE INTO snaps "
                        "(step_id, state_key, snap_epoch, ser_change) "
                        "VALUES (?, ?, ?, ?)",
                        rows,
                    )

    def write_frontier(
        self, ex_num: int, worker_index: int, worker_frontier: int
    ) -> None:
        part = (ex_num + worker_index) % self.part_count
        with self.lock:
            conn = self.part_conns[part]
            with conn:
                conn.execute(
                    "INSERT OR REPLACE INTO fronts "
                    "(ex_num, worker_index, worker_frontier) VALUES (?, ?, ?)",
                    (ex_num, worker_index, worker_frontier),
                )

    def commit_and_gc(self, cluster_frontier: int) -> None:
        """Commit a GC horizon and delete superseded snapshots.

        The horizon is delayed by the backup interval (reference
        recovery.rs:943-989).
        """
        commit_epoch = cluster_frontier - 1 - self.backup_delay_epochs
        if commit_epoch <= 0:
            return
        with self.lock:
            for part_index, conn in self.part_conns.items():
                with conn:
                    conn.execute(
                        "INSERT OR REPLACE INTO commits "
                        "(part_index, commit_epoch) VALUES (?, ?)",
                        (part_index, commit_epoch),
                    )
                    # Delete rows superseded by a newer snapshot that is
                    # itself within the committed horizon.
                    conn.execute(
                        """DELETE FROM snaps WHERE EXISTS (
                             SELECT 1 FROM snaps AS newer
                             WHERE newer.step_id = snaps.step_id
                               AND newer.state_key = snaps.state_key
                               AND newer.snap_epoch > snaps.snap_epoch
                               AND newer.snap_epoch <= ?
                           )""",
                        (commit_epoch,),
                    )

    # -- loaders --

    def load_resume_snaps(
        self, resume_epoch: int
    ) -> Iterable[Tuple[str, str, int, Optional[bytes]]]:
        """Latest snapshot per `(step_id, state_key)` strictly before
        the resume epoch (reference recovery.rs:800-903)."""
        best: Dict[Tuple[str, str], Tuple[int, Optional[bytes]]] = {}
        with self.lock:
            for conn in self.conns:
                for step_id, state_key, snap_epoch, ser_change in conn.execute(
                    """SELECT step_id, state_key, snap_epoch, ser_change
                       FROM snaps WHERE snap_epoch < ?""",
                    (resume_epoch,),
                ):
                    k = (step_id, state_key)
                    if k not in best or snap_epoch > best[k][0]:
                        best[k] = (snap_epoch, ser_change)
        for (step_id, state_key), (snap_epoch, ser_change) in best.items():
            yield (step_id, state_key, snap_epoch, ser_change)


def ser_state(state: Any) -> bytes:
    """Pickle operator state for a snapshot row."""
    return pickle.dumps(state)


def de_state(ser_change: bytes) -> Any:
    return pickle.loads(ser_change)


def _parse_args():
    parser = argparse.ArgumentParser(
        prog="python -m bytewax_amd.recovery",
        description="Create and init a set of recovery partitions.",
        epilog="See the `bytewax_amd.recovery` module docstring for more info.",
    )
    parser.add_argument(
        "db_dir",
        type=Path,
        help="Local directory to create partitions in",
    )
    parser.add_argument(
        "part_count",
        type=int,
        help="Number of partitions to create",
    )
    return parser.parse_args()


if __name__ == "__main__":
    args = _parse_args()
    init_db_dir(args.db_dir, args.part_count)
