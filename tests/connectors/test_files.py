"""File connector tests (parity: reference pytests/connectors/)."""

from pathlib import Path

import bytewax_amd.operators as op
from bytewax_amd.connectors.files import (
    CSVSource,
    DirSink,
    DirSource,
    FileSink,
    FileSource,
)
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.recovery import RecoveryConfig, init_db_dir
from bytewax_amd.testing import TestingSink, TestingSource, run_main
from tests.conftest import ZERO_TD


def test_file_source(tmp_path: Path, entry_point):
    f = tmp_path / "in.txt"
    f.write_text("one\ntwo\nthree\n")
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, FileSource(f))
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == ["one", "three", "two"]


def test_dir_source(tmp_path: Path, entry_point):
    (tmp_path / "a.txt").write_text("a1\na2\n")
    (tmp_path / "b.txt").write_text("b1\n")
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, DirSource(tmp_path, glob_pat="*.txt"))
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == ["a1", "a2", "b1"]


def test_csv_source(tmp_path: Path):
    f = tmp_path / "in.csv"
    f.write_text("id,name\n1,ada\n2,grace\n")
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, CSVSource(f))
    op.output("out", s, TestingSink(out))
    run_main(flow)
    assert out == [
        {"id": "1", "name": "ada"},
        {"id": "2", "name": "grace"},
    ]


def test_file_source_resume(tmp_path: Path):
    db = tmp_path / "db"
    db.mkdir()
    init_db_dir(db, 1)
    rc = RecoveryConfig(db)
    f = tmp_path / "in.txt"
    f.write_text("one\ntwo\nthree\nfour\n")
    out = []

    def build(abort_after):
        from bytewax_amd.inputs import AbortExecution

        seen = {"n": 0}

        def maybe_abort(x):
            seen["n"] += 1
            if abort_after is not None and seen["n"] > abort_after:
                raise AbortExecution()
            return x

        flow = Dataflow("f")
        s = op.input("inp", flow, FileSource(f, batch_size=1))
        s = op.map("chk", s, maybe_abort)
        op.output("out", s, TestingSink(out))
        return flow

    run_main(build(2), epoch_interval=ZERO_TD, recovery_config=rc)
    assert out == ["one", "two"]
    out.clear()
    run_main(build(None), epoch_interval=ZERO_TD, recovery_config=rc)
    # Offset snapshot from the epoch before the abort resumes at line 3
    # (line ordering is exactly-once, no duplicates).
    assert out == ["three", "four"]


def test_file_sink(tmp_path: Path, entry_point):
    target = tmp_path / "out.txt"
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["x", "y"]))
    keyed = op.key_on("k", s, lambda x: x)
    op.output("out", keyed, FileSink(target))
    entry_point(flow)
    assert sorted(target.read_text().splitlines()) == ["x", "y"]


def test_dir_sink(tmp_path: Path):
    outdir = tmp_path / "out"
    outdir.mkdir()
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([("a", "1"), ("b", "2")]))
    op.output("out", s, DirSink(outdir, file_count=2))
    run_main(flow)
    written = sorted(
        line
        for p in outdir.glob("part_*")
        for line in p.read_text().splitlines()
    )
    assert written == ["1", "2"]


def test_stdout_sink(capfd, entry_point):
    from bytewax_amd.connectors.stdio import StdOutSink

    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2]))
    op.output("out", s, StdOutSink())
    entry_point(flow)
    captured = capfd.readouterr()
    assert sorted(captured.out.splitlines()) == ["1", "2"]


def test_demo_source():
    from datetime import timedelta

    from bytewax_amd.connectors.demo import RandomMetricSource

    out = []
    flow = Dataflow("f")
    s = op.input(
        "inp",
        flow,
        RandomMetricSource(
            "m", interval=timedelta(0), count=5, next_random=lambda: 7
        ),
    )
    op.output("out", s, TestingSink(out))
    run_main(flow)
    assert out == [("m", 7)] * 5


def test_file_sink_truncates_to_resume_offset(tmp_path: Path):
    """Exactly-once in a batch context: on resume the sink seeks back
    to the snapshotted offset and truncates, dropping rows written
    after the last committed snapshot (reference files.py DirSink
    semantics)."""
    from bytewax_amd.connectors.files import FileSink

    path = tmp_path / "out.txt"
    sink = FileSink(path)
    part = sink.build_part("s", str(path), None)
    part.write_batch(["a", "b"])
    offset = part.snapshot()
    # Uncommitted tail beyond the snapshot.
    part.write_batch(["lost1", "lost2"])
    part.close()
    assert path.read_text() == "a\nb\nlost1\nlost2\n"
    # Restart from the committed offset: the tail is truncated away.
    part2 = sink.build_part("s", str(path), offset)
    part2.write_batch(["c"])
    part2.close()
    assert path.read_text() == "a\nb\nc\n"


def test_file_sink_end_to_end_recovery(tmp_path: Path, recovery_config):
    """FileSink + recovery: an aborted execution's uncommitted rows do
    not appear twice after resume."""
    import bytewax_amd.operators as op
    from bytewax_amd.connectors.files import FileSink
    from bytewax_amd.dataflow import Dataflow
    from bytewax_amd.testing import TestingSource, run_main

    out_path = tmp_path / "sunk.txt"

    def mk_flow(source_items):
        flow = Dataflow("sink_rec")
        s = op.input("inp", flow, TestingSource(source_items))
        keyed = op.key_on("k", s, lambda _x: "p")
        fmt = op.map_value("fmt", keyed, str)
        op.output("out", fmt, FileSink(out_path))
        return flow

    items = [1, 2, TestingSource.ABORT(), 3, 4]
    flow = mk_flow(items)
    try:
        run_main(flow, recovery_config=recovery_config)
    except SystemExit:
        pass
    except Exception:
        pass
    flow2 = mk_flow(items)
    run_main(flow2, recovery_config=recovery_config)
    lines = [ln for ln in out_path.read_text().splitlines() if ln]
    assert sorted(lines) == ["1", "2", "3", "4"]
