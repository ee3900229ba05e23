"""Stateless operator behavior (parity: reference pytests/operators/)."""

import re

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource


def test_map(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2, 3]))
    s = op.map("add", s, lambda x: x + 1)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [2, 3, 4]


def test_filter(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2, 3, 4]))
    s = op.filter("odd", s, lambda x: x % 2 == 1)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [1, 3]


def test_filter_non_bool_raises(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1]))
    s = op.filter("bad", s, lambda x: x)
    op.output("out", s, TestingSink(out))
    with pytest.raises(TypeError):
        entry_point(flow)


def test_flat_map(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["split me up"]))
    s = op.flat_map("split", s, str.split)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == ["me", "split", "up"]


def test_flat_map_batch(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2, 3], batch_size=3))
    s = op.flat_map_batch("dbl", s, lambda xs: [x * 2 for x in xs])
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [2, 4, 6]


def test_filter_map(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([0, 1, 2, 3]))
    s = op.filter_map("fm", s, lambda x: x * 10 if x % 2 == 0 else None)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [0, 20]


def test_flatten(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([[1, 2], [3]]))
    s = op.flatten("flat", s)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [1, 2, 3]


def test_flatten_non_iterable_raises(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1]))
    s = op.flatten("flat", s)
    op.output("out", s, TestingSink(out))
    with pytest.raises(TypeError):
        entry_point(flow)


def test_branch(entry_point):
    trues = []
    falses = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2, 3, 4]))
    b = op.branch("b", s, lambda x: x % 2 == 0)
    op.output("t", b.trues, TestingSink(trues))
    op.output("f", b.falses, TestingSink(falses))
    entry_point(flow)
    assert sorted(trues) == [2, 4]
    assert sorted(falses) == [1, 3]


def test_merge(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([1, 2]))
    b = op.input("b", flow, TestingSource([3, 4]))
    m = op.merge("m", a, b)
    op.output("out", m, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [1, 2, 3, 4]


def test_redistribute(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(list(range(10))))
    s = op.redistribute("shuffle", s)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == list(range(10))


def test_key_on_key_rm(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2]))
    keyed = op.key_on("k", s, lambda x: str(x))
    unkeyed = op.key_rm("unk", keyed)
    op.output("out", unkeyed, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [1, 2]


def test_key_on_non_str_raises(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1]))
    keyed = op.key_on("k", s, lambda x: x)
    op.output("out", keyed, TestingSink(out))
    with pytest.raises(TypeError):
        entry_point(flow)


def test_map_value(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([("a", 1), ("b", 2)]))
    s = op.map_value("mv", s, lambda v: v * 10)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 10), ("b", 20)]


def test_filter_value(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([("a", 1), ("b", 2)]))
    s = op.filter_value("fv", s, lambda v: v > 1)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert out == [("b", 2)]


def test_inspect(entry_point, capfd):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["a"]))
    s = op.inspect("help", s)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    captured = capfd.readouterr()
    assert "f.help: 'a'" in captured.out
    assert out == ["a"]


def test_inspect_debug(entry_point, capfd):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["a"]))
    s = op.inspect_debug("dbg", s)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    captured = capfd.readouterr()
    assert re.search(r"f\.dbg W\d+ @\d+: 'a'", captured.out)
    assert out == ["a"]


def test_raises(entry_point):
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1]))
    op.raises("die", s)
    with pytest.raises(RuntimeError):
        entry_point(flow)


def test_enrich_cached(entry_point):
    lookups = []

    def mock_service(k):
        lookups.append(k)
        return {"a": 10, "b": 20}.get(k)

    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["a", "b", "a"]))
    s = op.enrich_cached(
        "enrich", s, mock_service, lambda cache, item: (item, cache.get(item))
    )
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 10), ("a", 10), ("b", 20)]


def test_user_exception_propagates(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1, 2]))

    def boom(x):
        raise ValueError("boom")

    s = op.map("boom", s, boom)
    op.output("out", s, TestingSink(out))
    # User exceptions surface WRAPPED as BytewaxRuntimeError with the
    # original chained as __cause__ (the reference's convention; its
    # pytests assert the custom exception does not escape raw).
    from bytewax_amd.errors import BytewaxRuntimeError

    with pytest.raises(BytewaxRuntimeError) as exc_info:
        entry_point(flow)
    assert isinstance(exc_info.value.__cause__, ValueError)
    assert "boom" in str(exc_info.value.__cause__)


def test_mid_flow_merge_of_keyed_streams(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([("k", 1)]))
    b = op.input("b", flow, TestingSource([("k", 2)]))
    m = op.merge("m", a, b)
    summed = op.fold_final("sum", m, int, lambda acc, v: acc + v)
    op.output("out", summed, TestingSink(out))
    entry_point(flow)
    # fold_final keys on the tuple's key; both items share "k".
    assert out == [("k", 3)]


def test_merge_many_streams(entry_point):
    out = []
    flow = Dataflow("f")
    ins = [
        op.input(f"i{i}", flow, TestingSource([i * 10, i * 10 + 1]))
        for i in range(4)
    ]
    m = op.merge("m", *ins)
    op.output("out", m, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [0, 1, 10, 11, 20, 21, 30, 31]
