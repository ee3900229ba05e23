"""Property-based check: the window state's CPU twin equals a brute
-force fold for arbitrary batches (hypothesis; derandomized so CI is
stable)."""

from collections import Counter

import pytest

torch = pytest.importorskip("torch")
hyp = pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st  # noqa: E402

from bytewax_amd.gpu import AGG_SUM, RecordBatch, WindowAggState  # noqa: E402

events = st.lists(
    st.tuples(
        st.integers(min_value=0, max_value=7),      # key
        st.integers(min_value=0, max_value=500),    # ts
        st.integers(min_value=-10, max_value=10),   # val
    ),
    min_size=0,
    max_size=60,
)


@settings(max_examples=60, derandomize=True, deadline=None)
@given(batches=st.lists(events, min_size=1, max_size=4),
       len_ms=st.sampled_from([7, 50, 100]))
def test_window_sum_twin_matches_bruteforce(batches, len_ms):
    stt = WindowAggState(torch.device("cpu"), 0, len_ms, AGG_SUM)
    ref = Counter()
    for evs in batches:
        if not evs:
            continue
        keys = torch.tensor([e[0] for e in evs], dtype=torch.int32)
        ts = torch.tensor([e[1] for e in evs], dtype=torch.int64)
        vals = torch.tensor([e[2] for e in evs], dtype=torch.int64)
        stt.insert(RecordBatch(keys, ts, vals, max_ts=int(ts.max())))
        for k, t, v in evs:
            ref[(k, t // len_ms)] += v
    out = stt.close_all()
    got = Counter()
    if out is not None:
        for k, t, v in zip(
            out.keys.tolist(), out.ts.tolist(), out.vals.tolist()
        ):
            got[(k, t // len_ms)] += v
    assert got == ref
