"""Property test: the WindowAggState CPU twin (the semantic reference
for the HIP window kernels, including the round-2 sliding expansion)
against a brute-force dict over random batches/windows/offsets."""

import pytest

torch = pytest.importorskip("torch")
hyp = pytest.importorskip("hypothesis")

from hypothesis import given, settings, strategies as st  # noqa: E402

from bytewax_amd.gpu import AGG_SUM, RecordBatch, WindowAggState  # noqa: E402

ALIGN = 1_700_000_000_000


@st.composite
def window_case(draw):
    off = draw(st.integers(min_value=1, max_value=6)) * 10
    len_mult = draw(st.integers(min_value=1, max_value=4))
    length = off * len_mult
    n = draw(st.integers(min_value=0, max_value=120))
    events = [
        (
            draw(st.integers(min_value=0, max_value=9)),
            draw(st.integers(min_value=0, max_value=500)),
            draw(st.integers(min_value=-5, max_value=50)),
        )
        for _ in range(n)
    ]
    n_batches = draw(st.integers(min_value=1, max_value=4))
    return length, off, events, n_batches


@settings(max_examples=120, deadline=None)
@given(window_case())
def test_cpu_twin_matches_brute_force(case):
    length, off, events, n_batches = case
    state = WindowAggState(
        torch.device("cpu"), ALIGN, length, AGG_SUM, off_ms=off
    )
    per = max(1, -(-len(events) // n_batches))
    for i in range(0, len(events), per):
        chunk = events[i : i + per]
        if not chunk:
            continue
        state.insert(
            RecordBatch(
                torch.tensor([k for k, _, _ in chunk], dtype=torch.int32),
                torch.tensor(
                    [ALIGN + t for _, t, _ in chunk], dtype=torch.int64
                ),
                torch.tensor([v for _, _, v in chunk], dtype=torch.int64),
            )
        )
    out = state.close_all()

    expected = {}
    for k, t, v in events:
        hi = t // off  # newest window containing t
        lo = (t - length) // off + 1
        for w in range(lo, hi + 1):
            key = (k, ALIGN + w * off)
            expected[key] = expected.get(key, 0) + v
    got = {}
    if out is not None:
        for k, ts, v in zip(
            out.keys.tolist(), out.ts.tolist(), out.vals.tolist()
        ):
            got[(k, ts)] = got.get((k, ts), 0) + v
    assert got == expected
