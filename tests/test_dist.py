"""Multi-process cluster execution over the gloo transport.

Covers the `torch.distributed` exchange path (the CPU twin of the
RCCL/xGMI GPU exchange) end-to-end through the `python -m
bytewax_amd.run` CLI: keyed state is partitioned across 2 OS
processes and results are written through a partitioned file sink.
"""

import os
import subprocess
import sys
import textwrap
from pathlib import Path

import pytest


@pytest.mark.timeout(180)
def test_two_process_cluster_keyed_state(tmp_path: Path):
    out_file = tmp_path / "out.txt"
    flow_file = tmp_path / "flowdef.py"
    flow_file.write_text(
        textwrap.dedent(
            f"""
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.files import FileSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            inp = [(str(i % 4), 1) for i in range(20)]

            flow = Dataflow("dist_test")
            s = op.input("inp", flow, TestingSource(inp))

            def running_sum(state, v):
                state = (state or 0) + v
                return (state, state)

            s = op.stateful_map("sum", s, running_sum)
            s = op.map("fmt", s, lambda kv: (kv[0], f"{{kv[0]}}={{kv[1]}}"))
            op.output("out", s, FileSink({str(out_file)!r}))
            """
        )
    )
    port = 29400 + os.getpid() % 500
    addresses = f"127.0.0.1:{port};127.0.0.1:{port + 1}"
    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    procs = [
        subprocess.Popen(
            [
                sys.executable,
                "-m",
                "bytewax_amd.run",
                f"{flow_file}:flow",
                "-i",
                str(i),
                "-a",
                addresses,
            ],
            env=env,
            stdout=subprocess.PIPE,
            stderr=subprocess.PIPE,
        )
        for i in range(2)
    ]
    outs = []
    for p in procs:
        stdout, stderr = p.communicate(timeout=150)
        outs.append((p.returncode, stdout, stderr))
    for rc, stdout, stderr in outs:
        assert rc == 0, f"proc failed: {stderr.decode()[-2000:]}"

    lines = sorted(out_file.read_text().splitlines())
    # 4 keys x 5 items each, running sums 1..5 per key.
    expected = sorted(f"{k}={v}" for k in "0123" for v in range(1, 6))
    assert lines == expected


@pytest.mark.timeout(240)
def test_two_process_cluster_recovery_resume(tmp_path: Path):
    """EOF + resume across a 2-process gloo cluster sharing one
    recovery directory: keyed state and source positions restore with
    exactly-once results.  (ABORT is single-process-only by its
    documented contract — its one-shot latch is in-process state.)"""
    from bytewax_amd.recovery import init_db_dir

    db = tmp_path / "db"
    db.mkdir()
    init_db_dir(db, 4)
    out_file = tmp_path / "out.txt"
    flow_file = tmp_path / "flowdef.py"
    flow_file.write_text(
        textwrap.dedent(
            f"""
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.files import FileSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            inp = [(str(i % 2), 1) for i in range(6)]
            inp.insert(4, TestingSource.EOF())

            flow = Dataflow("dist_rec")
            s = op.input("inp", flow, TestingSource(inp))

            def running_sum(state, v):
                state = (state or 0) + v
                return (state, state)

            s = op.stateful_map("sum", s, running_sum)
            s = op.map("fmt", s, lambda kv: (kv[0], f"{{kv[0]}}={{kv[1]}}"))
            op.output("out", s, FileSink({str(out_file)!r}))
            """
        )
    )
    port = 29100 + os.getpid() % 300

    def launch():
        addresses = f"127.0.0.1:{port};127.0.0.1:{port + 1}"
        env = dict(os.environ)
        env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
        procs = [
            subprocess.Popen(
                [
                    sys.executable,
                    "-m",
                    "bytewax_amd.run",
                    f"{flow_file}:flow",
                    "-i",
                    str(i),
                    "-a",
                    addresses,
                    "-r",
                    str(db),
                    "-s",
                    "0",
                    "-b",
                    "0",
                ],
                env=env,
                stdout=subprocess.PIPE,
                stderr=subprocess.PIPE,
            )
            for i in range(2)
        ]
        for p in procs:
            _so, se = p.communicate(timeout=200)
            assert p.returncode == 0, se.decode()[-2000:]

    launch()  # runs until the EOF sentinel
    first = sorted(out_file.read_text().splitlines())
    assert first == ["0=1", "0=2", "1=1", "1=2"]

    launch()  # continuation: state 2 per key, remaining items
    lines = sorted(out_file.read_text().splitlines())
    # FileSink truncates to its snapshot offset on resume, so the file
    # holds the pre-EOF lines plus the continued ones.
    assert lines == ["0=1", "0=2", "0=3", "1=1", "1=2", "1=3"]


@pytest.mark.timeout(180)
def test_testing_module_cluster_launcher(tmp_path: Path):
    """`python -m bytewax_amd.testing -p2` spawns a working local
    cluster (parity: reference testing.py:287-383)."""
    flow_file = tmp_path / "lflow.py"
    out_file = tmp_path / "out.txt"
    flow_file.write_text(
        textwrap.dedent(
            f"""
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.files import FileSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            flow = Dataflow("launcher")
            s = op.input("inp", flow, TestingSource([("a", "x"), ("b", "y")]))
            op.output("out", s, FileSink({str(out_file)!r}))
            """
        )
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    res = subprocess.run(
        [
            sys.executable,
            "-m",
            "bytewax_amd.testing",
            f"{flow_file}:flow",
            "-p",
            "2",
            "-w",
            "1",
        ],
        env=env,
        capture_output=True,
        timeout=150,
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert sorted(out_file.read_text().splitlines()) == ["x", "y"]


@pytest.mark.timeout(180)
def test_one_rank_failure_aborts_cluster_cleanly(tmp_path: Path):
    """A user exception on one rank votes the cluster into an abort:
    the failing rank re-raises, the peer exits cleanly and promptly
    (no hang on a broken collective). Reference run.rs:273-304 panic
    hook semantics."""
    flow_file = tmp_path / "failing_flow.py"
    flow_file.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.stdio import StdOutSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.inputs import (
                FixedPartitionedSource,
                StatefulSourcePartition,
            )

            class _Part(StatefulSourcePartition):
                def __init__(self, part, resume):
                    self.part = part
                    self.i = resume if resume is not None else 0

                def next_batch(self):
                    if self.i >= 5:
                        raise StopIteration()
                    self.i += 1
                    return [(self.part, self.i)]

                def snapshot(self):
                    return self.i

            class Src(FixedPartitionedSource):
                def list_parts(self):
                    return ["p0", "p1"]

                def build_part(self, step_id, part, resume):
                    return _Part(part, resume)

            def boom(item):
                part, i = item
                if part == "p1" and i == 3:
                    raise ValueError("poison item")
                return item

            flow = Dataflow("fail_test")
            s = op.input("inp", flow, Src())
            s = op.map("boom", s, boom)
            op.output("out", s, StdOutSink())
            """
        )
    )
    port = 29500 + os.getpid() % 300
    addresses = f"127.0.0.1:{port};127.0.0.1:{port + 1}"
    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    procs = [
        subprocess.Popen(
            [
                sys.executable,
                "-m",
                "bytewax_amd.run",
                f"{flow_file}:flow",
                "-i",
                str(i),
                "-a",
                addresses,
            ],
            env=env,
            stdout=subprocess.PIPE,
            stderr=subprocess.PIPE,
        )
        for i in range(2)
    ]
    outs = [p.communicate(timeout=120) for p in procs]
    rcs = [p.returncode for p in procs]
    # Parts assign round-robin over sorted names: p1 -> worker 1.
    assert rcs[1] != 0
    assert "poison item" in outs[1][1].decode()
    assert rcs[0] == 0, outs[0][1].decode()[-1500:]


@pytest.mark.timeout(180)
def test_threads_times_processes_cluster(tmp_path: Path):
    """`-w 2 -a <2 procs>` = 4 global workers: 2 worker THREADS per
    process over the gloo mesh (reference Cluster{threads, process,
    addresses}, src/run.rs:259-271).  Keyed state partitions across
    all 4; results are exact."""
    out_file = tmp_path / "out.txt"
    flow_file = tmp_path / "flowdef.py"
    flow_file.write_text(
        textwrap.dedent(
            f"""
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.files import FileSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            inp = [(str(i % 8), 1) for i in range(40)]

            flow = Dataflow("dist_hybrid")
            s = op.input("inp", flow, TestingSource(inp))

            def running_sum(state, v):
                state = (state or 0) + v
                return (state, state)

            s = op.stateful_map("sum", s, running_sum)
            s = op.map("fmt", s, lambda kv: (kv[0], f"{{kv[0]}}={{kv[1]}}"))
            op.output("out", s, FileSink({str(out_file)!r}))
            """
        )
    )
    port = 29700 + os.getpid() % 200
    addresses = f"127.0.0.1:{port};127.0.0.1:{port + 1}"
    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    procs = [
        subprocess.Popen(
            [
                sys.executable,
                "-m",
                "bytewax_amd.run",
                f"{flow_file}:flow",
                "-i",
                str(i),
                "-a",
                addresses,
                "-w",
                "2",
            ],
            env=env,
            stdout=subprocess.PIPE,
            stderr=subprocess.PIPE,
        )
        for i in range(2)
    ]
    for p in procs:
        _so, se = p.communicate(timeout=150)
        assert p.returncode == 0, f"proc failed: {se.decode()[-2000:]}"

    lines = sorted(out_file.read_text().splitlines())
    expected = sorted(
        f"{k}={v}" for k in "01234567" for v in range(1, 6)
    )
    assert lines == expected
