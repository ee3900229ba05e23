"""Graph-builder tests (parity: reference pytests/test_dataflow.py)."""

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow, Stream
from bytewax_amd.testing import TestingSource


def test_flow_id_no_period():
    with pytest.raises(ValueError):
        Dataflow("a.b")


def test_step_id_no_period():
    flow = Dataflow("flow")
    s = op.input("inp", flow, TestingSource([]))
    with pytest.raises(ValueError):
        op.map("a.b", s, lambda x: x)


def test_duplicate_step_id_raises():
    flow = Dataflow("flow")
    s = op.input("inp", flow, TestingSource([]))
    op.map("dup", s, lambda x: x)
    with pytest.raises(ValueError, match="duplicate"):
        op.map("dup", s, lambda x: x)


def test_nested_step_ids_are_scoped():
    flow = Dataflow("flow")
    s = op.input("inp", flow, TestingSource([]))
    mapped = op.map("go", s, lambda x: x)
    # `map` is a composite over flat_map_batch; the substep nests.
    (map_step,) = [st for st in flow.substeps if st.step_name == "go"]
    assert map_step.step_id == "flow.go"
    assert map_step.substeps[0].step_id.startswith("flow.go.")
    assert isinstance(mapped, Stream)


def test_then_chaining():
    flow = Dataflow("flow")
    s = op.input("inp", flow, TestingSource([1]))
    s2 = s.then(op.map, "x", lambda x: x + 1)
    assert isinstance(s2, Stream)
    assert s2.flow() is flow


def test_stream_ids_unique():
    flow = Dataflow("flow")
    a = op.input("a", flow, TestingSource([]))
    b = op.input("b", flow, TestingSource([]))
    assert a.stream_id != b.stream_id


def test_input_requires_source():
    flow = Dataflow("flow")
    with pytest.raises(TypeError):
        op.input("inp", flow, [1, 2, 3])


def test_output_requires_sink():
    flow = Dataflow("flow")
    s = op.input("inp", flow, TestingSource([]))
    with pytest.raises(TypeError):
        op.output("out", s, [])


def test_operator_decorator_requires_step_id():
    from bytewax_amd.dataflow import operator

    with pytest.raises(TypeError):

        @operator
        def bad_op(up):  # missing step_id
            return up
