"""Recovery / resume behavior (parity: reference pytests/test_recovery.py)."""

import shutil
from datetime import timedelta
from pathlib import Path

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.recovery import (
    InconsistentPartitionsError,
    MissingPartitionsError,
    NoPartitionsError,
    RecoveryConfig,
    init_db_dir,
)
from bytewax_amd.testing import TestingSink, TestingSource, cluster_main, run_main

ZERO_TD = timedelta(seconds=0)
FIVE_TD = timedelta(seconds=5)


def test_abort_no_snapshots(recovery_config):
    inp = [0, 1, 2, TestingSource.ABORT(), 3, 4]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))
        op.output("out", s, TestingSink(out))
        return flow

    # 5-sec epoch interval: no snapshot when the abort happens.
    run_main(build(), epoch_interval=FIVE_TD, recovery_config=recovery_config)
    assert out == [0, 1, 2]

    # So resume replays all input.
    out.clear()
    run_main(build(), epoch_interval=FIVE_TD, recovery_config=recovery_config)
    assert out == [0, 1, 2, 3, 4]


def test_abort_with_snapshots(recovery_config):
    inp = [0, 1, 2, TestingSource.ABORT(), 3, 4]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))
        op.output("out", s, TestingSink(out))
        return flow

    # 0-sec epoch interval: snapshot after each item.
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [0, 1, 2]

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [3, 4]


def test_continuation(recovery_config):
    inp = [0, 1, 2, TestingSource.EOF(), 3, 4]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))
        op.output("out", s, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [0, 1, 2]

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [3, 4]

    # Running again after full consumption emits nothing.
    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == []


def test_stateful_state_resumes(recovery_config):
    inp = [("a", 1), ("a", 2), TestingSource.EOF(), ("a", 10)]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))

        def running_sum(state, v):
            state = (state or 0) + v
            return (state, state)

        s = op.stateful_map("sum", s, running_sum)
        op.output("out", s, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [("a", 1), ("a", 3)]

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    # State 3 resumed; 3 + 10 = 13.
    assert out == [("a", 13)]


def test_rescale(tmp_path: Path):
    """State rendezvous across changing worker counts (reference
    pytests/test_recovery.py:151-201)."""
    init_db_dir(tmp_path, 4)
    rc = RecoveryConfig(tmp_path)
    inp = [
        ("a", 1),
        ("b", 10),
        TestingSource.EOF(),
        ("a", 2),
        ("b", 20),
        TestingSource.EOF(),
        ("a", 3),
        ("b", 30),
    ]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))

        def running_sum(state, v):
            state = (state or 0) + v
            return (state, state)

        s = op.stateful_map("sum", s, running_sum)
        op.output("out", s, TestingSink(out))
        return flow

    cluster_main(
        build(), [], 0, worker_count_per_proc=3,
        epoch_interval=ZERO_TD, recovery_config=rc,
    )
    assert sorted(out) == [("a", 1), ("b", 10)]

    out.clear()
    cluster_main(
        build(), [], 0, worker_count_per_proc=5,
        epoch_interval=ZERO_TD, recovery_config=rc,
    )
    assert sorted(out) == [("a", 3), ("b", 30)]

    out.clear()
    cluster_main(
        build(), [], 0, worker_count_per_proc=1,
        epoch_interval=ZERO_TD, recovery_config=rc,
    )
    assert sorted(out) == [("a", 6), ("b", 60)]


def test_no_partitions_raises(tmp_path: Path):
    rc = RecoveryConfig(tmp_path)
    flow = Dataflow("test_df")
    s = op.input("inp", flow, TestingSource([1]))
    op.output("out", s, TestingSink([]))
    with pytest.raises(NoPartitionsError):
        run_main(flow, recovery_config=rc)


def test_missing_partition_raises(tmp_path: Path):
    init_db_dir(tmp_path, 3)
    (tmp_path / "part-1.sqlite3").unlink()
    rc = RecoveryConfig(tmp_path)
    flow = Dataflow("test_df")
    s = op.input("inp", flow, TestingSource([1]))
    op.output("out", s, TestingSink([]))
    with pytest.raises(MissingPartitionsError):
        run_main(flow, recovery_config=rc)


def test_inconsistent_partitions_raises(tmp_path: Path):
    """GC'ing a partition past the resume epoch must refuse resume."""
    db_a = tmp_path / "a"
    db_a.mkdir()
    init_db_dir(db_a, 2)
    rc = RecoveryConfig(db_a)
    inp = [0, 1, 2, TestingSource.EOF(), 3, 4]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))
        op.output("out", s, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=rc)

    # Simulate a partition whose commit horizon is past the resume
    # epoch (e.g. restored from a too-new copy).
    import sqlite3

    conn = sqlite3.connect(db_a / "part-0.sqlite3")
    with conn:
        conn.execute(
            "INSERT OR REPLACE INTO commits (part_index, commit_epoch) "
            "VALUES (0, 99999)"
        )
    conn.close()

    with pytest.raises(InconsistentPartitionsError):
        run_main(build(), epoch_interval=ZERO_TD, recovery_config=rc)


def test_gc_keeps_latest_snapshot(recovery_config):
    """After GC only the newest snapshot per (step, key) remains, and
    resume still works."""
    inp = [("a", 1), ("a", 2), ("a", 3), TestingSource.EOF(), ("a", 4)]
    out = []

    def build():
        flow = Dataflow("test_df")
        s = op.input("inp", flow, TestingSource(inp))

        def running_sum(state, v):
            state = (state or 0) + v
            return (state, state)

        s = op.stateful_map("sum", s, running_sum)
        op.output("out", s, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)

    # Count snapshot rows for the stateful step across partitions:
    # GC should have deleted superseded epochs.
    import sqlite3

    rows = []
    for p in sorted(Path(recovery_config.db_dir).glob("*.sqlite3")):
        conn = sqlite3.connect(p)
        rows += conn.execute(
            "SELECT step_id, state_key, snap_epoch FROM snaps "
            "WHERE state_key = 'a'"
        ).fetchall()
        conn.close()
    sum_rows = [r for r in rows if "sum" in r[0]]
    assert len(sum_rows) == 1

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [("a", 10)]


def test_backup_interval_delays_gc(tmp_path: Path):
    init_db_dir(tmp_path, 1)
    rc = RecoveryConfig(tmp_path, backup_interval=timedelta(hours=1))
    inp = [("a", 1), ("a", 2), ("a", 3)]
    out = []

    flow = Dataflow("test_df")
    s = op.input("inp", flow, TestingSource(inp))

    def running_sum(state, v):
        state = (state or 0) + v
        return (state, state)

    s2 = op.stateful_map("sum", s, running_sum)
    op.output("out", s2, TestingSink(out))

    run_main(flow, epoch_interval=ZERO_TD, recovery_config=rc)

    import sqlite3

    conn = sqlite3.connect(tmp_path / "part-0.sqlite3")
    rows = conn.execute(
        "SELECT COUNT(*) FROM snaps WHERE state_key = 'a' "
        "AND step_id LIKE '%sum%'"
    ).fetchone()[0]
    conn.close()
    # With a long backup interval nothing is GC'd.
    assert rows >= 2
