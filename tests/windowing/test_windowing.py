"""Windowing operator behavior (parity: reference
pytests/operators/windowing/)."""

from datetime import datetime, timedelta, timezone

import pytest

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as w
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.operators.windowing import (
    LATE_SESSION_ID,
    EventClock,
    SessionWindower,
    SlidingWindower,
    SystemClock,
    TumblingWindower,
    WindowMetadata,
)
from bytewax_amd.testing import TestingSink, TestingSource, TimeTestingGetter, run_main

ALIGN_TO = datetime(2022, 1, 1, tzinfo=timezone.utc)
ZERO_TD = timedelta(seconds=0)


def ec(wait=ZERO_TD):
    return EventClock(ts_getter=lambda x: x[0], wait_for_system_duration=wait)


def ts(secs):
    return ALIGN_TO + timedelta(seconds=secs)


def test_fold_window_tumbling(entry_point):
    inp = [(ts(1), "a"), (ts(2), "b"), (ts(61), "c")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    wo = w.fold_window(
        "fw",
        keyed,
        ec(),
        TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1)),
        list,
        lambda acc, x: acc + [x[1]],
        lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("ALL", (0, ["a", "b"])), ("ALL", (1, ["c"]))]


def test_count_window(entry_point):
    inp = [ts(0), ts(30), ts(65), ts(90)]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    clock = EventClock(ts_getter=lambda x: x, wait_for_system_duration=ZERO_TD)
    wo = w.count_window(
        "cw",
        s,
        clock,
        TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1)),
        lambda x: "ALL",
    )
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("ALL", (0, 2)), ("ALL", (1, 2))]


def test_reduce_window_max_min(entry_point):
    inp = [(ts(0), 3), (ts(1), 7), (ts(2), 1)]
    outr, outmx, outmn = [], [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wr = w.reduce_window(
        "rw", keyed, ec(), tw, lambda a, b: (a[0], a[1] + b[1])
    )
    wmx = w.max_window("mx", keyed, ec(), tw, by=lambda x: x[1])
    wmn = w.min_window("mn", keyed, ec(), tw, by=lambda x: x[1])
    op.output("or_", wr.down, TestingSink(outr))
    op.output("omx", wmx.down, TestingSink(outmx))
    op.output("omn", wmn.down, TestingSink(outmn))
    entry_point(flow)
    assert outr == [("ALL", (0, (ts(0), 11)))]
    assert outmx == [("ALL", (0, (ts(1), 7)))]
    assert outmn == [("ALL", (0, (ts(2), 1)))]


def test_collect_window_list_set_dict(entry_point):
    inp = [(ts(0), 1), (ts(1), 2), (ts(2), 1)]
    outl, outs = [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    vals = op.map_value("v", keyed, lambda x: x[1])
    # values lose timestamps; clock must read original; use list of
    # pairs for list collect on the keyed (ts, v) stream instead
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wl = w.collect_window("cl", keyed, ec(), tw, into=list)
    op.output("ol", wl.down, TestingSink(outl))
    entry_point(flow)
    assert outl == [("ALL", (0, [(ts(0), 1), (ts(1), 2), (ts(2), 1)]))]


def test_sliding_window_overlap(entry_point):
    # length 60 s, offset 30 s: an item at t=45 is in windows 0 and 1.
    inp = [(ts(45), "x")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    sw = SlidingWindower(
        length=timedelta(seconds=60),
        offset=timedelta(seconds=30),
        align_to=ALIGN_TO,
    )
    wo = w.collect_window("cw", keyed, ec(), sw, into=list)
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [
        ("ALL", (0, [(ts(45), "x")])),
        ("ALL", (1, [(ts(45), "x")])),
    ]


def test_sliding_offset_longer_than_length_raises():
    with pytest.raises(ValueError):
        SlidingWindower(
            length=timedelta(seconds=10),
            offset=timedelta(seconds=20),
            align_to=ALIGN_TO,
        )


def test_session_window_gap(entry_point):
    inp = [(ts(0), "a"), (ts(1), "b"), (ts(100), "c")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    sw = SessionWindower(gap=timedelta(seconds=10))
    wo = w.collect_window("cw", keyed, ec(), sw, into=list)
    vals = op.map_value(
        "strip", wo.down, lambda wid_acc: (wid_acc[0], [v for _t, v in wid_acc[1]])
    )
    op.output("out", vals, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("ALL", (0, ["a", "b"])), ("ALL", (1, ["c"]))]


def test_session_window_merge():
    # Items arrive out of order within the watermark wait: two
    # sessions form then merge when a bridging item arrives.
    inp = [(ts(0), "a"), (ts(30), "c"), (ts(12), "b")]
    out = []
    metas = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    clock = ec(wait=timedelta(seconds=120))
    sw = SessionWindower(gap=timedelta(seconds=15))
    wo = w.collect_window("cw", keyed, clock, sw, into=list)
    vals = op.map_value(
        "strip", wo.down, lambda wid_acc: (wid_acc[0], [v for _t, v in wid_acc[1]])
    )
    op.output("out", vals, TestingSink(out))
    op.output("meta", wo.meta, TestingSink(metas))
    run_main(flow)
    # "a"@0 and "b"@12 merge (gap 12 <= 15); "c"@30 is 18 s after the
    # extended close at 12, which exceeds the 15 s gap: two sessions.
    assert sorted(out) == [("ALL", (0, ["a", "b"])), ("ALL", (1, ["c"]))]


def test_late_items():
    inp = [(ts(100), "on-time"), (ts(0), "late")]
    down, late = [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(seconds=10))
    wo = w.collect_window("cw", keyed, ec(), tw, into=list)
    op.output("down", wo.down, TestingSink(down))
    op.output("late", wo.late, TestingSink(late))
    run_main(flow)
    assert late == [("ALL", (0, (ts(0), "late")))]
    assert down == [("ALL", (10, [(ts(100), "on-time")]))]


def test_window_metadata():
    inp = [(ts(0), "a")]
    metas = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.collect_window("cw", keyed, ec(), tw, into=list)
    op.output("meta", wo.meta, TestingSink(metas))
    run_main(flow)
    assert metas == [
        (
            "ALL",
            (
                0,
                WindowMetadata(
                    open_time=ALIGN_TO,
                    close_time=ALIGN_TO + timedelta(minutes=1),
                ),
            ),
        )
    ]


def test_join_window(entry_point):
    a_inp = [(ts(0), 1)]
    b_inp = [(ts(5), "x")]
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource(a_inp))
    b = op.input("b", flow, TestingSource(b_inp))
    ka = op.key_on("ka", a, lambda x: "K")
    kb = op.key_on("kb", b, lambda x: "K")
    clock = EventClock(
        ts_getter=lambda x: x[0], wait_for_system_duration=ZERO_TD
    )
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.join_window("jw", clock, tw, ka, kb)
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert out == [("K", (0, ((ts(0), 1), (ts(5), "x"))))]


def test_event_clock_watermark_wait():
    """With a long wait duration and a controllable clock, windows only
    close once system time advances past close + wait."""
    getter = TimeTestingGetter(now=datetime(2024, 1, 1, tzinfo=timezone.utc))
    clock = EventClock(
        ts_getter=lambda x: x[0],
        wait_for_system_duration=timedelta(seconds=10),
        now_getter=getter.get,
        to_system_utc=lambda _ts: None,
    )
    logic = clock.build(None)
    logic.before_batch()
    ts0, watermark = logic.on_item((ts(60), "a"))
    assert ts0 == ts(60)
    assert watermark == ts(50)
    # Advance the fake system clock: watermark moves with it.
    getter.advance(timedelta(seconds=5))
    assert logic.on_notify() == ts(55)


def test_system_clock(entry_point):
    inp = list(range(5))
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    # Huge windows: everything falls in one window, closed at EOF.
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(days=100_000))
    wo = w.collect_window("cw", keyed, SystemClock(), tw, into=list)
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert len(out) == 1
    k, (_wid, vals) = out[0]
    assert k == "ALL"
    assert sorted(vals) == inp


def test_fold_window_recovery_open_window_resumes(recovery_config):
    """An ABORT mid-stream leaves window 0 open in the snapshot; the
    resumed execution folds into the *recovered* accumulator."""
    inp = [
        (ts(0), "a"),
        (ts(1), "b"),
        TestingSource.ABORT(),
        (ts(2), "c"),
        (ts(61), "d"),
    ]
    out = []

    def build():
        flow = Dataflow("f")
        s = op.input("inp", flow, TestingSource(inp))
        keyed = op.key_on("k", s, lambda x: "ALL")
        clock = EventClock(
            ts_getter=lambda x: x[0],
            wait_for_system_duration=timedelta(hours=1),
        )
        tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
        wo = w.fold_window(
            "fw", keyed, clock, tw,
            list,
            lambda acc, x: acc + [x[1]],
            lambda x, y: x + y,
        )
        op.output("out", wo.down, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    # Aborted before EOF: nothing closed yet.
    assert out == []

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    # Window 0 resumed with ["a", "b"], folds "c"; "d" opens window 1;
    # EOF closes both.
    assert out == [("ALL", (0, ["a", "b", "c"])), ("ALL", (1, ["d"]))]


def test_fold_window_eof_discards_state(recovery_config):
    """EOF closes all windows and discards logic state; a resumed
    execution starts windows fresh (parity with reference
    `_WindowLogic` is_empty semantics)."""
    inp = [
        (ts(0), "a"),
        TestingSource.EOF(),
        (ts(2), "c"),
    ]
    out = []

    def build():
        flow = Dataflow("f")
        s = op.input("inp", flow, TestingSource(inp))
        keyed = op.key_on("k", s, lambda x: "ALL")
        tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
        wo = w.fold_window(
            "fw", keyed, ec(), tw,
            list,
            lambda acc, x: acc + [x[1]],
            lambda x, y: x + y,
        )
        op.output("out", wo.down, TestingSink(out))
        return flow

    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [("ALL", (0, ["a"]))]

    out.clear()
    run_main(build(), epoch_interval=ZERO_TD, recovery_config=recovery_config)
    assert out == [("ALL", (0, ["c"]))]


def test_fold_window_ordered_replays_by_timestamp():
    """With ordered=True (default) values reach the fold in timestamp
    order even when they arrive shuffled within the watermark wait."""
    inp = [(ts(5), "c"), (ts(1), "a"), (ts(3), "b")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    clock = ec(wait=timedelta(hours=1))
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.fold_window(
        "fw", keyed, clock, tw,
        list, lambda acc, x: acc + [x[1]], lambda a, b: a + b,
        ordered=True,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    assert out == [("ALL", (0, ["a", "b", "c"]))]


def test_fold_window_unordered_keeps_arrival_order():
    inp = [(ts(5), "c"), (ts(1), "a"), (ts(3), "b")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    clock = ec(wait=timedelta(hours=1))
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.fold_window(
        "fw", keyed, clock, tw,
        list, lambda acc, x: acc + [x[1]], lambda a, b: a + b,
        ordered=False,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    assert out == [("ALL", (0, ["c", "a", "b"]))]


def test_collect_window_set_and_dict(entry_point):
    inp = [(ts(0), ("k1", 1)), (ts(1), ("k2", 2)), (ts(2), ("k1", 3))]
    outs, outd = [], []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    keyed = op.key_on("k", s, lambda x: "ALL")
    vals = op.map_value("strip_ts", keyed, lambda tv: tv[1])
    # The clock must see timestamps, so collect on the (ts, pair)
    # stream for `set` and on pairs for `dict` via a second window.
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    ws = w.collect_window("cs", keyed, ec(), tw, into=set)
    wd = w.collect_window(
        "cd",
        vals,
        EventClock(
            ts_getter=lambda _v: ts(0),
            wait_for_system_duration=timedelta(hours=1),
        ),
        tw,
        into=dict,
    )
    op.output("os", ws.down, TestingSink(outs))
    op.output("od", wd.down, TestingSink(outd))
    entry_point(flow)
    assert outs[0][1][1] == set(inp)
    assert outd[0][1][1] == {"k1": 3, "k2": 2}


def test_join_window_running_mode(entry_point):
    a_inp = [(ts(0), 1), (ts(2), 2)]
    b_inp = [(ts(1), "x")]
    out = []
    flow = Dataflow("f")
    a = op.input("a", flow, TestingSource(a_inp))
    b = op.input("b", flow, TestingSource(b_inp))
    ka = op.key_on("ka", a, lambda x: "K")
    kb = op.key_on("kb", b, lambda x: "K")
    clock = EventClock(
        ts_getter=lambda x: x[0], wait_for_system_duration=ZERO_TD
    )
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.join_window("jw", clock, tw, ka, kb, emit_mode="running")
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    # Running mode emits on every value (ordered by timestamp).
    vals = [v for _k, (_w, v) in out]
    assert vals == [
        ((ts(0), 1), None),
        ((ts(0), 1), (ts(1), "x")),
        ((ts(2), 2), (ts(1), "x")),
    ]


def test_count_window_multiple_keys(entry_point):
    inp = [(ts(0), "a"), (ts(1), "b"), (ts(2), "a")]
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(inp))
    clock = EventClock(
        ts_getter=lambda x: x[0], wait_for_system_duration=ZERO_TD
    )
    tw = TumblingWindower(align_to=ALIGN_TO, length=timedelta(minutes=1))
    wo = w.count_window("cw", s, clock, tw, lambda x: x[1])
    op.output("out", wo.down, TestingSink(out))
    entry_point(flow)
    assert sorted(out) == [("a", (0, 2)), ("b", (0, 1))]
