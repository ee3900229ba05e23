"""Windower logics as plain objects (parity: reference
pytests/operators/windowing/test_sliding_windower.py etc.)."""

from datetime import datetime, timedelta, timezone

from bytewax_amd.operators.windowing import (
    LATE_SESSION_ID,
    SessionWindower,
    SlidingWindower,
    TumblingWindower,
    WindowMetadata,
)

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)


def ts(secs):
    return ALIGN + timedelta(seconds=secs)


def test_sliding_intersects_overlapping():
    logic = SlidingWindower(
        length=timedelta(seconds=10),
        offset=timedelta(seconds=5),
        align_to=ALIGN,
    ).build(None)
    # t=7 is within [0,10) (win 0) and [5,15) (win 1).
    assert logic.intersects(ts(7)) == [0, 1]
    # t=3 only within [-5,5) (win -1) and [0,10) (win 0).
    assert logic.intersects(ts(3)) == [-1, 0]


def test_sliding_open_close_notify():
    logic = SlidingWindower(
        length=timedelta(seconds=10),
        offset=timedelta(seconds=10),
        align_to=ALIGN,
    ).build(None)
    assert list(logic.open_for(ts(1))) == [0]
    assert logic.notify_at() == ts(10)
    assert not logic.is_empty()
    # Not closed before close time.
    assert list(logic.close_for(ts(9))) == []
    closed = list(logic.close_for(ts(10)))
    assert closed == [(0, WindowMetadata(ts(0), ts(10)))]
    assert logic.is_empty()


def test_tumbling_is_sliding_with_offset_eq_length():
    logic = TumblingWindower(
        length=timedelta(seconds=10), align_to=ALIGN
    ).build(None)
    assert list(logic.open_for(ts(25))) == [2]
    assert list(logic.late_for(ts(25))) == [2]


def test_session_opens_and_extends():
    logic = SessionWindower(gap=timedelta(seconds=10)).build(None)
    (w0,) = logic.open_for(ts(0))
    assert w0 == 0
    # Within gap after close: same session, close time extends.
    (w1,) = logic.open_for(ts(8))
    assert w1 == 0
    # Outside the gap: new session.
    (w2,) = logic.open_for(ts(30))
    assert w2 == 1
    # Late items get the sentinel.
    assert list(logic.late_for(ts(0))) == [LATE_SESSION_ID]
    # Sessions close only once the watermark passes close + gap.
    assert list(logic.close_for(ts(17))) == []
    closed = list(logic.close_for(ts(19)))
    assert [w for w, _m in closed] == [0]
    meta = closed[0][1]
    assert meta.open_time == ts(0)
    assert meta.close_time == ts(8)


def test_session_merge_reports_pairs():
    logic = SessionWindower(gap=timedelta(seconds=10)).build(None)
    logic.open_for(ts(0))  # session 0: [0, 0]
    logic.open_for(ts(18))  # session 1: [18, 18] (18 > gap from 0)
    assert list(logic.merged()) == []
    # Bridging item at 9 extends session 0's close to 9, which is now
    # within the gap of session 1's open (18 - 9 <= 10) -> merge 1
    # into 0.
    assert list(logic.open_for(ts(9))) == [0]
    merges = list(logic.merged())
    assert merges == [(1, 0)]
    # The merged metadata spans both.
    closed = list(logic.close_for(ts(100)))
    assert len(closed) == 1
    wid, meta = closed[0]
    assert wid == 0
    assert meta.open_time == ts(0)
    assert meta.close_time == ts(18)
    assert meta.merged_ids == {1}


def test_session_never_empty():
    logic = SessionWindower(gap=timedelta(seconds=1)).build(None)
    assert not logic.is_empty()


def test_snapshot_roundtrip_sliding():
    windower = SlidingWindower(
        length=timedelta(seconds=10),
        offset=timedelta(seconds=10),
        align_to=ALIGN,
    )
    a = windower.build(None)
    a.open_for(ts(3))
    snap = a.snapshot()
    b = windower.build(snap)
    assert list(b.close_for(ts(10))) == [(0, WindowMetadata(ts(0), ts(10)))]


def test_event_clock_watermark_monotonic_and_waits():
    """EventClock watermark = max event ts - wait + elapsed system
    time, and never regresses (reference windowing.py
    _EventClockLogic)."""
    from datetime import datetime, timedelta, timezone

    from bytewax_amd.operators.windowing import EventClock
    from bytewax_amd.testing import TimeTestingGetter

    t0 = datetime(2024, 1, 1, tzinfo=timezone.utc)
    getter = TimeTestingGetter(t0)
    clock = EventClock(
        ts_getter=lambda x: x,
        wait_for_system_duration=timedelta(seconds=10),
        now_getter=getter.get,
    )
    logic = clock.build(None)
    ev = t0 + timedelta(seconds=100)
    logic.before_batch()
    ts, _wm = logic.on_item(ev)
    assert ts == ev
    wm0 = logic.on_notify()
    assert wm0 == ev - timedelta(seconds=10)
    # System time advances without events: watermark advances too.
    getter.advance(timedelta(seconds=4))
    assert logic.on_notify() == wm0 + timedelta(seconds=4)
    # An older event cannot regress the watermark.
    logic.before_batch()
    logic.on_item(t0)
    assert logic.on_notify() >= wm0 + timedelta(seconds=4)
