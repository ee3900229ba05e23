"""Visualize + run-CLI behavior (parity: reference pytests/test_parse.py,
test_visualize.py)."""

import json
import subprocess
import sys
import textwrap
from pathlib import Path

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.run import _locate_dataflow, _prepare_import
from bytewax_amd.testing import TestingSink, TestingSource
from bytewax_amd.visualize import render_dataflow, to_json, to_mermaid


def _flow():
    flow = Dataflow("viz")
    s = op.input("inp", flow, TestingSource([1]))
    s = op.map("double", s, lambda x: x * 2)
    op.output("out", s, TestingSink([]))
    return flow


def test_render_dataflow():
    r = render_dataflow(_flow())
    assert r.flow_id == "viz"
    names = [s.step_id for s in r.substeps]
    assert names == ["viz.inp", "viz.double", "viz.out"]
    # Composite steps have substeps.
    double = r.substeps[1]
    assert double.substeps[0].op_type == "flat_map_batch"


def test_to_json_roundtrips():
    d = json.loads(to_json(_flow()))
    assert d["flow_id"] == "viz"
    assert len(d["substeps"]) == 3


def test_to_mermaid():
    m = to_mermaid(_flow())
    assert "flowchart TD" in m
    assert 'viz.inp -- "down → up" --> viz.double' in m
    assert 'viz.double -- "down → up" --> viz.out' in m


def test_prepare_import_py_path(tmp_path: Path):
    f = tmp_path / "myflow.py"
    f.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSink, TestingSource

            flow = Dataflow("fromfile")
            s = op.input("inp", flow, TestingSource([1]))
            op.output("out", s, TestingSink([]))
            """
        )
    )
    flow = _locate_dataflow(*_prepare_import(f"{f}:flow"))
    assert flow.flow_id == "fromfile"


def test_prepare_import_default_name(tmp_path: Path):
    f = tmp_path / "defflow.py"
    f.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSink, TestingSource

            flow = Dataflow("deffy")
            s = op.input("inp", flow, TestingSource([1]))
            op.output("out", s, TestingSink([]))
            """
        )
    )
    flow = _locate_dataflow(*_prepare_import(str(f)))
    assert flow.flow_id == "deffy"


def test_locate_dataflow_factory(tmp_path: Path):
    f = tmp_path / "factoryflow.py"
    f.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSink, TestingSource

            def build(name="fact"):
                flow = Dataflow(name)
                s = op.input("inp", flow, TestingSource([1]))
                op.output("out", s, TestingSink([]))
                return flow
            """
        )
    )
    sys.path.insert(0, str(tmp_path))
    try:
        flow = _locate_dataflow("factoryflow", "build('custom')")
        assert flow.flow_id == "custom"
        flow2 = _locate_dataflow("factoryflow", "build")
        assert flow2.flow_id == "fact"
    finally:
        sys.path.remove(str(tmp_path))


def test_locate_dataflow_bad_name(tmp_path: Path):
    f = tmp_path / "badflow.py"
    f.write_text("x = 1\n")
    sys.path.insert(0, str(tmp_path))
    try:
        with pytest.raises(AttributeError):
            _locate_dataflow("badflow", "nope")
        with pytest.raises(TypeError):
            _locate_dataflow("badflow", "x")
    finally:
        sys.path.remove(str(tmp_path))


def test_cli_end_to_end(tmp_path: Path):
    f = tmp_path / "cliflow.py"
    f.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.stdio import StdOutSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            flow = Dataflow("cli")
            s = op.input("inp", flow, TestingSource(["hello"]))
            op.output("out", s, StdOutSink())
            """
        )
    )
    import os

    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    res = subprocess.run(
        [sys.executable, "-m", "bytewax_amd.run", f"{f}:flow"],
        capture_output=True,
        timeout=120,
        env=env,
    )
    assert res.returncode == 0, res.stderr.decode()
    assert "hello" in res.stdout.decode()


def test_visualize_cli(tmp_path: Path):
    f = tmp_path / "vizflow.py"
    f.write_text(
        textwrap.dedent(
            """
            import bytewax_amd.operators as op
            from bytewax_amd.connectors.stdio import StdOutSink
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.testing import TestingSource

            flow = Dataflow("vizcli")
            s = op.input("inp", flow, TestingSource([1]))
            op.output("out", s, StdOutSink())
            """
        )
    )
    import os

    env = dict(os.environ)
    env["PYTHONPATH"] = str(Path(__file__).resolve().parent.parent)
    res = subprocess.run(
        [
            sys.executable,
            "-m",
            "bytewax_amd.visualize",
            f"{f}:flow",
            "--format",
            "json",
        ],
        capture_output=True,
        timeout=120,
        env=env,
    )
    assert res.returncode == 0, res.stderr.decode()
    assert json.loads(res.stdout)["flow_id"] == "vizcli"
