"""Property test: host session windows (gap-merge logic incl.
`_session_find_merges`) vs a brute-force session builder, over
arbitrary out-of-order arrivals kept in-allowance by a large
EventClock wait (reference tests sessions similarly heavily;
pytests/operators/windowing/test_session_windower.py)."""

from collections import Counter
from datetime import datetime, timedelta, timezone

import pytest

hyp = pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st  # noqa: E402

import bytewax_amd.operators as op  # noqa: E402
import bytewax_amd.operators.windowing as w  # noqa: E402
from bytewax_amd.dataflow import Dataflow  # noqa: E402
from bytewax_amd.testing import (  # noqa: E402
    TestingSink,
    TestingSource,
    run_main,
)

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)


def _brute_sessions(items, gap_ms):
    """Maximal groups of items whose sorted timestamps never gap by
    more than `gap_ms`; returns a list of (ts, val) Counters ordered
    by session start."""
    if not items:
        return []
    srt = sorted(items)
    sessions = [[srt[0]]]
    for it in srt[1:]:
        if it[0] - sessions[-1][-1][0] <= gap_ms:
            sessions[-1].append(it)
        else:
            sessions.append([it])
    return [Counter(sess) for sess in sessions]


@settings(max_examples=60, deadline=None, derandomize=True)
@given(
    items=st.lists(
        st.tuples(
            st.integers(min_value=0, max_value=300),  # ts offset (ms)
            st.integers(min_value=0, max_value=9),    # value
        ),
        min_size=0,
        max_size=50,
    ),
    gap_ms=st.integers(min_value=1, max_value=60),
)
def test_session_windows_match_brute_force(items, gap_ms):
    out = []
    flow = Dataflow("prop_sessions")
    src = [
        (ALIGN + timedelta(milliseconds=ts), ts, v) for ts, v in items
    ]
    s = op.input("inp", flow, TestingSource(src))
    keyed = op.key_on("k", s, lambda it: "ALL")
    clock = w.EventClock(
        ts_getter=lambda it: it[0],
        # Large allowance: arbitrary arrival disorder stays
        # in-window, so every merge path is exercised.
        wait_for_system_duration=timedelta(seconds=600),
    )
    wo = w.fold_window(
        "fw",
        keyed,
        clock,
        w.SessionWindower(gap=timedelta(milliseconds=gap_ms)),
        list,
        lambda acc, it: acc + [(it[1], it[2])],
        lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)

    got = sorted(
        (min(ts for ts, _v in pairs), Counter(pairs))
        for _key, (_wid, pairs) in out
        if pairs
    )
    expected = [
        (min(c)[0], c) for c in _brute_sessions(items, gap_ms)
    ]
    assert [c for _t, c in got] == [c for _t, c in expected]
