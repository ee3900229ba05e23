"""Stateful operator behavior (parity: reference pytests/operators/)."""

from datetime import timedelta

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource


def test_stateful_map(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([("a", 1), ("a", 2), ("b", 5)]))

    def running_sum(state, v):
        state = (state or 0) + v
        return (state, state)

    s = op.stateful_map("sum", s, running_sum)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 1), ("a", 3), ("b", 5)]


def test_stateful_map_discard_state(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input(
        "inp", flow, TestingSource([("a", 1), ("a", 1), ("a", 1)])
    )

    def tally_resets(state, v):
        state = (state or 0) + v
        if state >= 2:
            return (None, state)
        return (state, state)

    s = op.stateful_map("tally", s, tally_resets)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert out == [("a", 1), ("a", 2), ("a", 1)]


def test_stateful_flat_map(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([("a", 2), ("a", 3)]))

    def dup(state, v):
        return (state, [v] * v)

    s = op.stateful_flat_map("dup", s, dup)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 2)] * 2 + [("a", 3)] * 3


def test_fold_final(entry_point):
    out = []
    flow = Dataflow("f")
    source = [("key1", 1), ("key1", 2), ("key2", 3), ("key2", 5)]
    s = op.input("inp", flow, TestingSource(source))
    s = op.fold_final("fold", s, lambda: 0, lambda acc, v: acc + v)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("key1", 3), ("key2", 8)]


def test_reduce_final(entry_point):
    out = []
    flow = Dataflow("f")
    source = [("a", 1), ("a", 2), ("b", 10)]
    s = op.input("inp", flow, TestingSource(source))
    s = op.reduce_final("red", s, lambda a, b: a + b)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 3), ("b", 10)]


def test_count_final(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(["a", "b", "a"]))
    s = op.count_final("count", s, lambda x: x)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", 2), ("b", 1)]


def test_max_final_min_final(entry_point):
    outmax = []
    outmin = []
    flow = Dataflow("f")
    source = [("a", 3), ("a", 1), ("a", 2)]
    s = op.input("inp", flow, TestingSource(source))
    mx = op.max_final("mx", s)
    mn = op.min_final("mn", s)
    op.output("om", mx, TestingSink(outmax))
    op.output("on", mn, TestingSink(outmin))
    entry_point(flow)
    assert outmax == [("a", 3)]
    assert outmin == [("a", 1)]


def test_collect_max_size(entry_point):
    out = []
    flow = Dataflow("f")
    source = [("a", 1), ("a", 2), ("a", 3), ("a", 4)]
    s = op.input("inp", flow, TestingSource(source))
    s = op.collect("c", s, timedelta(seconds=100), max_size=2)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert out == [("a", [1, 2]), ("a", [3, 4])]


def test_collect_eof_flush(entry_point):
    out = []
    flow = Dataflow("f")
    source = [("a", 1), ("a", 2), ("a", 3)]
    s = op.input("inp", flow, TestingSource(source))
    s = op.collect("c", s, timedelta(seconds=100), max_size=2)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    assert out == [("a", [1, 2]), ("a", [3])]


def test_collect_timeout():
    # Timer-driven flush: a PAUSE longer than the timeout forces the
    # notify path.  Single-worker only to keep timing deterministic.
    from bytewax_amd.testing import run_main

    out = []
    flow = Dataflow("f")
    source = [
        ("a", 1),
        TestingSource.PAUSE(timedelta(milliseconds=300)),
        ("a", 2),
    ]
    s = op.input("inp", flow, TestingSource(source))
    s = op.collect("c", s, timedelta(milliseconds=50), max_size=10)
    op.output("out", s, TestingSink(out))
    run_main(flow)
    assert out == [("a", [1]), ("a", [2])]


def test_join_complete(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([("k", 1)]))
    b = op.input("b", flow, TestingSource([("k", "x")]))
    j = op.join("j", a, b)
    op.output("out", j, TestingSink(out))
    entry_point(flow)
    assert out == [("k", (1, "x"))]


def test_join_running(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([("k", 1), ("k", 2)]))
    b = op.input("b", flow, TestingSource([], batch_size=1))
    j = op.join("j", a, b, emit_mode="running")
    op.output("out", j, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("k", (1, None)), ("k", (2, None))]


def test_join_final(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([("k", 1)]))
    b = op.input("b", flow, TestingSource([("j", "y")]))
    j = op.join("j", a, b, emit_mode="final")
    op.output("out", j, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("j", (None, "y")), ("k", (1, None))]


def test_join_product(entry_point):
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([("k", 1), ("k", 2)]))
    b = op.input("b", flow, TestingSource([("k", "x")]))
    j = op.join("j", a, b, insert_mode="product", emit_mode="final")
    op.output("out", j, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("k", (1, "x")), ("k", (2, "x"))]


def test_join_bad_mode_raises():
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource([]))
    with pytest.raises(ValueError):
        op.join("j", a, a, insert_mode="nope")


def test_stateful_requires_kv(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([1]))
    s = op.stateful_map("sm", s, lambda st, v: (st, v))
    op.output("out", s, TestingSink(out))
    with pytest.raises(TypeError):
        entry_point(flow)


def test_stateful_requires_str_key(entry_point):
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource([(1, 1)]))
    s = op.stateful_map("sm", s, lambda st, v: (st, v))
    op.output("out", s, TestingSink(out))
    with pytest.raises(TypeError):
        entry_point(flow)


def test_stateful_flat_map_expands_and_discards(entry_point):
    """stateful_flat_map: one-to-many with per-key state; returning
    None state discards the key (reference operators/__init__.py
    stateful_flat_map)."""
    import bytewax_amd.operators as op
    from bytewax_amd.dataflow import Dataflow
    from bytewax_amd.testing import TestingSink, TestingSource

    out = []
    flow = Dataflow("sfm")
    s = op.input(
        "inp", flow, TestingSource([("k", 1), ("k", 2), ("k", 3)])
    )

    def dup_until_two(state, v):
        seen = (state or 0) + 1
        # Emit v repeated `seen` times; discard state after 2 values.
        new_state = None if seen >= 2 else seen
        return (new_state, [v] * seen)

    s = op.stateful_flat_map("fm", s, dup_until_two)
    op.output("out", s, TestingSink(out))
    entry_point(flow)
    # k=1 (seen 1 -> [1]), k=2 (seen 2 -> [2,2], discard),
    # k=3 (fresh, seen 1 -> [3]).
    assert sorted(out) == [("k", 1), ("k", 2), ("k", 2), ("k", 3)]


def test_stateful_midbatch_discard_keeps_remaining_values():
    """A per-item logic that completes mid-batch must NOT drop the
    batch's remaining values: the shim rebuilds a fresh logic for
    them (reference pysrc _StatefulLogic.on_batch rebuilds and
    continues).  Regression: join complete-mode with two right-side
    values in one batch previously lost the second."""
    out = []
    flow = Dataflow("midbatch_discard")
    l = op.input(
        "l", flow, TestingSource([("k", "a1"), ("k", "a3")], batch_size=1)
    )
    r = op.input(
        "r", flow, TestingSource([("k", "b1"), ("k", "b2")], batch_size=10)
    )
    j = op.join("j", l, r)
    op.output("out", j, TestingSink(out))
    from bytewax_amd.testing import run_main

    run_main(flow)
    # b1 completes the first pair (discard); b2 must persist in a
    # fresh logic and join the later a3.
    assert ("k", ("a1", "b1")) in out
    assert ("k", ("a3", "b2")) in out


def test_collect_multiple_max_size_completions_in_one_batch():
    """collect(max_size=3) fed 7 same-key items in one source batch
    must emit [0,1,2], [3,4,5] and (at EOF) [6] — each max_size
    completion discards the logic mid-batch and a fresh one collects
    the rest (same shim path as the join regression above)."""
    out = []
    flow = Dataflow("collect_midbatch")
    s = op.input(
        "inp",
        flow,
        TestingSource([("k", i) for i in range(7)], batch_size=10),
    )
    c = op.collect(
        "collect", s, timeout=timedelta(seconds=10), max_size=3
    )
    op.output("out", c, TestingSink(out))
    from bytewax_amd.testing import run_main

    run_main(flow)
    assert [v for _k, v in out] == [[0, 1, 2], [3, 4, 5], [6]]
