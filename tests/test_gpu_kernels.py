"""HIP kernel numerics vs plain-Python/torch references (GPU only).

Every kernel result is checked against an exact host-side reference of
the same op, per the test strategy in SURVEY.md §4.
"""

from collections import Counter
from datetime import datetime, timedelta, timezone

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)


def _skip_no_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")


def _ref_counts(keys, ts, align_ms, len_ms):
    c = Counter()
    for k, t in zip(keys.tolist(), ts.tolist()):
        win = (t - align_ms) // len_ms
        c[(k, win)] += 1
    return c


def _ref_sums(keys, ts, vals, align_ms, len_ms):
    c = Counter()
    for k, t, v in zip(keys.tolist(), ts.tolist(), vals.tolist()):
        win = (t - align_ms) // len_ms
        c[(k, win)] += v
    return c


def _extract_to_counter(state):
    batch = state.close_all()
    if batch is None:
        return Counter()
    keys = batch.keys.cpu().tolist()
    wins = ((batch.ts.cpu() - state.align_ms) // state.len_ms).tolist()
    vals = batch.vals.cpu().tolist()
    return Counter({(k, w): v for k, w, v in zip(keys, wins, vals)})


@pytest.mark.parametrize("dedup", [False, True])
def test_window_count_matches_reference(dedup):
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    torch.manual_seed(0)
    n = 200_000
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    keys = torch.randint(0, 500, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 300_000, (n,), dtype=torch.int64)
    ref = _ref_counts(keys, ts, align_ms, len_ms)

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT,
        slots_pow=14, dedup=dedup,
    )
    state.insert(RecordBatch(keys.cuda(), ts.cuda()))
    got = _extract_to_counter(state)
    assert got == ref


@pytest.mark.parametrize("mode_name", ["count", "sum"])
def test_window_radix_matches_reference(mode_name):
    """The radix-partitioned LDS-staged path must agree exactly with
    the host reference (and hence with the single-pass path)."""
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, AGG_SUM, RecordBatch, WindowAggState, _ms

    torch.manual_seed(9)
    n = 500_000
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    keys = torch.randint(0, 20_000, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 300_000, (n,), dtype=torch.int64)
    vals = torch.randint(0, 100, (n,), dtype=torch.int64)
    mode = AGG_COUNT if mode_name == "count" else AGG_SUM
    if mode == AGG_COUNT:
        ref = _ref_counts(keys, ts, align_ms, len_ms)
    else:
        ref = Counter(
            {
                k: v
                for k, v in _ref_sums(
                    keys, ts, vals, align_ms, len_ms
                ).items()
            }
        )

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, mode,
        slots_pow=18, radix=True, region_bits=11, max_batch=n,
    )
    state.insert(
        RecordBatch(keys.cuda(), ts.cuda(), vals.cuda())
    )
    got = _extract_to_counter(state)
    assert got == ref


def test_window_sum_matches_reference():
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_SUM, RecordBatch, WindowAggState, _ms

    torch.manual_seed(1)
    n = 100_000
    align_ms = _ms(ALIGN)
    len_ms = 10_000
    keys = torch.randint(0, 100, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 100_000, (n,), dtype=torch.int64)
    vals = torch.randint(0, 1000, (n,), dtype=torch.int64)
    ref = _ref_sums(keys, ts, vals, align_ms, len_ms)

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_SUM, slots_pow=12
    )
    state.insert(RecordBatch(keys.cuda(), ts.cuda(), vals.cuda()))
    got = _extract_to_counter(state)
    assert got == ref


def test_window_sum_dedup_matches_reference():
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_SUM, RecordBatch, WindowAggState, _ms

    torch.manual_seed(5)
    n = 64_000
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    # 2 keys: heavy contention, exercises the wave segmented-sum path.
    keys = torch.randint(0, 2, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 120_000, (n,), dtype=torch.int64)
    vals = torch.randint(0, 50, (n,), dtype=torch.int64)
    ref = _ref_sums(keys, ts, vals, align_ms, len_ms)

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_SUM,
        slots_pow=10, dedup=True,
    )
    state.insert(RecordBatch(keys.cuda(), ts.cuda(), vals.cuda()))
    got = _extract_to_counter(state)
    assert got == ref


def test_watermark_close_partial():
    """Only windows below the watermark horizon close; the rest stay."""
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    align_ms = _ms(ALIGN)
    len_ms = 60_000
    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT, slots_pow=10
    )
    # Window 0 events, then window 1 events.
    k0 = torch.zeros(100, dtype=torch.int32).cuda()
    t0 = torch.full((100,), align_ms + 5_000, dtype=torch.int64).cuda()
    state.insert(RecordBatch(k0, t0, max_ts=align_ms + 5_000))
    assert state.close_due() is None  # watermark still inside window 0

    t1 = torch.full((50,), align_ms + 65_000, dtype=torch.int64).cuda()
    state.insert(
        RecordBatch(k0[:50], t1, max_ts=align_ms + 65_000)
    )
    closed = state.close_due()
    assert closed is not None
    assert len(closed) == 1
    assert int(closed.vals[0].item()) == 100  # window 0 count
    # Window 1 still open; closes at EOF.
    rest = state.close_all()
    assert rest is not None and int(rest.vals[0].item()) == 50


def test_snapshot_restore_roundtrip():
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    torch.manual_seed(2)
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    keys = torch.randint(0, 50, (10_000,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 60_000, (10_000,), dtype=torch.int64)
    ref = _ref_counts(keys, ts, align_ms, len_ms)

    a = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT, slots_pow=10
    )
    a.insert(RecordBatch(keys.cuda(), ts.cuda(), max_ts=int(ts.max())))
    snap = a.snapshot_to_host()

    b = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT, slots_pow=10
    )
    b.restore_from_host(snap)
    got = _extract_to_counter(b)
    assert got == ref
    assert b.max_ts_host == a.max_ts_host


def test_bucket_exchange_kernels():
    """Hist + scatter produce consistent per-destination segments."""
    _skip_no_gpu()
    from bytewax_amd.gpu import ext

    k = ext()
    torch.manual_seed(3)
    n = 100_000
    world = 8
    keys = torch.randint(0, 10_000, (n,), dtype=torch.int32).cuda()
    ts = torch.arange(n, dtype=torch.int64).cuda()
    counts = torch.zeros(world, dtype=torch.int32, device="cuda")
    k.bucket_hist(keys, world, counts)
    assert int(counts.sum().item()) == n
    offsets = torch.cumsum(counts, 0, dtype=torch.int32) - counts
    cursors = offsets.clone()
    out_keys = torch.empty(n, dtype=torch.int32, device="cuda")
    out_ts = torch.empty(n, dtype=torch.int64, device="cuda")
    out_vals = torch.empty(0, dtype=torch.int64, device="cuda")
    k.bucket_scatter(keys, ts, None, world, cursors, out_keys, out_ts, out_vals)
    # Multiset of (key, ts) preserved.
    a = sorted(zip(keys.cpu().tolist(), ts.cpu().tolist()))
    b = sorted(zip(out_keys.cpu().tolist(), out_ts.cpu().tolist()))
    assert a == b
    # Each segment contains only keys routed to that destination.
    off = offsets.cpu().tolist()
    cnt = counts.cpu().tolist()
    ok = out_keys.cpu().tolist()

    def mix64(x):
        x &= (1 << 64) - 1
        x ^= x >> 33
        x = (x * 0xFF51AFD7ED558CCD) & ((1 << 64) - 1)
        x ^= x >> 33
        x = (x * 0xC4CEB9FE1A85EC53) & ((1 << 64) - 1)
        x ^= x >> 33
        return x

    for d in range(world):
        for key in ok[off[d] : off[d] + cnt[d]]:
            assert mix64(key) % world == d


def test_engine_pipeline_on_gpu():
    """Full engine run: synthetic source -> window count -> sink."""
    _skip_no_gpu()
    from __graft_entry__ import smoke

    smoke()


def test_filter_compact_gpu():
    _skip_no_gpu()
    from bytewax_amd.gpu import RecordBatch
    from bytewax_amd.gpu.operators import filter_batch  # noqa: F401
    from bytewax_amd.gpu import ext

    torch.manual_seed(11)
    n = 1_000_000
    keys = torch.randint(0, 1000, (n,), dtype=torch.int32).cuda()
    ts = torch.arange(n, dtype=torch.int64).cuda()
    mask = (keys % 3 == 0)
    out_keys = torch.empty(n, dtype=torch.int32, device="cuda")
    out_ts = torch.empty(n, dtype=torch.int64, device="cuda")
    out_vals = torch.empty(0, dtype=torch.int64, device="cuda")
    out_n = torch.zeros(1, dtype=torch.int32, device="cuda")
    ext().filter_compact(
        keys, ts, None, mask.to(torch.uint8), out_keys, out_ts, out_vals,
        out_n,
    )
    kept = int(out_n.item())
    assert kept == int(mask.sum().item())
    got = sorted(zip(out_keys[:kept].cpu().tolist(), out_ts[:kept].cpu().tolist()))
    ref = sorted(
        (k, t)
        for k, t, m in zip(keys.cpu().tolist(), ts.cpu().tolist(), mask.cpu().tolist())
        if m
    )
    assert got == ref


def test_sliding_window_gpu_matches_reference():
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    torch.manual_seed(13)
    n = 100_000
    align_ms = _ms(ALIGN)
    len_ms, off_ms = 60_000, 20_000
    keys = torch.randint(0, 200, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 200_000, (n,), dtype=torch.int64)

    ref = Counter()
    for k, t in zip(keys.tolist(), ts.tolist()):
        hi = (t - align_ms) // off_ms
        lo = (t - align_ms - len_ms) // off_ms + 1
        for wn in range(lo, hi + 1):
            ref[(k, wn)] += 1

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT,
        slots_pow=14, off_ms=off_ms,
    )
    state.insert(RecordBatch(keys.cuda(), ts.cuda()))
    batch = state.close_all()
    got = Counter()
    for k, t, v in zip(
        batch.keys.cpu().tolist(),
        batch.ts.cpu().tolist(),
        batch.vals.cpu().tolist(),
    ):
        got[(k, (t - align_ms) // off_ms)] = v
    # Negative-window cells (events near align with lo < 0) exist in
    # both; compare only non-negative windows where ref is exact too.
    assert got == ref


def test_window_radix_v2_matches_reference():
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    torch.manual_seed(17)
    n = 800_000
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    keys = torch.randint(0, 30_000, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 300_000, (n,), dtype=torch.int64)
    ref = _ref_counts(keys, ts, align_ms, len_ms)

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, AGG_COUNT,
        slots_pow=18, radix_v2=True, max_batch=n,
    )
    state.insert(RecordBatch(keys.cuda(), ts.cuda()))
    got = _extract_to_counter(state)
    assert got == ref


@pytest.mark.parametrize("mode_name", ["count", "sum"])
def test_window_radix_segmented_insert_matches_reference(mode_name):
    """`insert_lazy` consumes the exchange wire format (int32 ts
    deltas in per-source-rank segments with scalar bases) and must
    agree with the host reference exactly — this is the world>1 insert
    path."""
    _skip_no_gpu()
    from bytewax_amd.gpu import (
        AGG_COUNT,
        AGG_SUM,
        WindowAggState,
        _LazyTsBatch,
        _ms,
    )

    torch.manual_seed(11)
    n = 300_000
    align_ms = _ms(ALIGN)
    len_ms = 60_000
    keys = torch.randint(0, 20_000, (n,), dtype=torch.int32)
    ts = align_ms + torch.randint(0, 300_000, (n,), dtype=torch.int64)
    vals = torch.randint(0, 100, (n,), dtype=torch.int64)
    mode = AGG_COUNT if mode_name == "count" else AGG_SUM
    if mode == AGG_COUNT:
        ref = _ref_counts(keys, ts, align_ms, len_ms)
    else:
        ref = Counter(_ref_sums(keys, ts, vals, align_ms, len_ms))

    # Fake a 3-rank exchange: uneven segments, distinct bases.
    seg_counts = [n // 2, 0, n // 3, n - n // 2 - n // 3]
    seg_bases = [align_ms - 7, align_ms + 123_456, align_ms, align_ms + 9]
    ts32 = torch.empty(n, dtype=torch.int32)
    off = 0
    for cnt, base in zip(seg_counts, seg_bases):
        ts32[off : off + cnt] = (ts[off : off + cnt] - base).to(torch.int32)
        off += cnt
    lz = _LazyTsBatch(
        keys.cuda(), ts32.cuda(), seg_counts, seg_bases,
        vals.cuda() if mode == AGG_SUM else None, int(ts.max()),
    )

    state = WindowAggState(
        torch.device("cuda:0"), align_ms, len_ms, mode,
        slots_pow=18, radix=True, region_bits=11, max_batch=n,
    )
    state.insert_lazy(lz)
    assert int(state.max_ts_dev.item()) == int(ts.max())
    got = _extract_to_counter(state)
    assert got == ref


def test_native_pipelined_matches_serial():
    """The two-stream scatter/agg pipeline must produce byte-identical
    results to the serial native loop."""
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, WindowAggState, _ms

    align_ms = _ms(ALIGN)
    dev = torch.device("cuda:0")
    n = 500_000
    g = torch.Generator(device="cuda").manual_seed(3)
    key_pool = [
        torch.randint(0, 5_000, (n,), dtype=torch.int32, generator=g,
                      device=dev)
        for _ in range(4)
    ]
    ts_pool = [
        (torch.arange(n, dtype=torch.int64, device=dev) * 5000) // n
        for _ in range(4)
    ]

    def run(pipelined):
        st = WindowAggState(
            dev, align_ms, 1000, AGG_COUNT, slots_pow=16, radix=True,
            region_bits=9, max_batch=n,
        )
        rows, _ = st.native_run(
            key_pool, ts_pool, 0, 24, 5000, pipelined=pipelined
        )
        rest = st.close_all()
        tail = sorted(
            zip(rest.keys.cpu().tolist(), rest.ts.cpu().tolist(),
                rest.vals.cpu().tolist())
        ) if rest is not None else []
        return rows, tail

    rows_a, tail_a = run(False)
    rows_b, tail_b = run(True)
    assert rows_a == rows_b
    assert tail_a == tail_b
    # 24 steps x 5 windows/step worth of events over 5k keys: every
    # (key, window) cell must appear exactly once in closed+tail.
    assert rows_a > 0


def test_python_engine_pipelined_insert_matches_serial(monkeypatch):
    """The per-step engine's two-stream insert must equal the serial
    insert exactly (same flow, env-toggled)."""
    _skip_no_gpu()
    from datetime import timedelta

    import bytewax_amd.operators as op
    from bytewax_amd.dataflow import Dataflow
    from bytewax_amd.gpu.operators import (
        CollectCountsSink,
        SyntheticEventSource,
        keyed_window_agg,
    )
    from bytewax_amd.testing import run_main

    def run(pipe: str):
        monkeypatch.setenv("BYTEWAX_PY_PIPELINE", pipe)
        out = []
        flow = Dataflow("pipe_eq")
        s = op.input(
            "inp",
            flow,
            SyntheticEventSource(
                events_per_batch=2_000_000,
                n_batches=12,
                vocab=50_000,
                align_to=ALIGN,
                sim_ms_per_batch=30_000,
            ),
        )
        agg = keyed_window_agg(
            "agg",
            s,
            align_to=ALIGN,
            length=timedelta(minutes=1),
            mode="count",
            slots_pow=18,
            radix=True,
            out_cap=1 << 21,
        )
        op.output("out", agg, CollectCountsSink(out))
        run_main(flow)
        rows = sorted(
            (k, t, v)
            for b in out
            for k, t, v in zip(
                b.keys.cpu().tolist(),
                b.ts.cpu().tolist(),
                b.vals.cpu().tolist(),
            )
        )
        return rows

    serial = run("0")
    piped = run("1")
    assert serial == piped
    assert len(serial) > 0


def test_insert_pipelined_buffer_growth_matches_serial():
    """Growing batches force a mid-run scatter-buffer reallocation on
    the pipelined path; results must still match the serial insert."""
    _skip_no_gpu()
    from bytewax_amd.gpu import AGG_COUNT, RecordBatch, WindowAggState, _ms

    align_ms = _ms(ALIGN)
    dev = torch.device("cuda:0")
    sizes = [100_000, 150_000, 450_000, 200_000]
    g = torch.Generator(device="cuda").manual_seed(21)
    batches = []
    for i, n in enumerate(sizes):
        keys = torch.randint(0, 10_000, (n,), dtype=torch.int32,
                             generator=g, device=dev)
        ts = torch.randint(0, 5_000, (n,), dtype=torch.int64,
                           generator=g, device=dev)
        batches.append(
            RecordBatch(keys, ts, None, max_ts=(i + 1) * 5_000 - 1,
                        ts_base=align_ms + i * 5_000)
        )

    def run(pipelined):
        st = WindowAggState(
            dev, align_ms, 1000, AGG_COUNT, slots_pow=19, radix=True,
            region_bits=9, max_batch=sizes[0],
        )
        for b in batches:
            if pipelined:
                st.insert_pipelined(
                    b.keys, b.ts, b.vals, b.ts_base, [], [], b.max_ts
                )
            else:
                st.insert(b)
        rest = st.close_all()
        return sorted(
            zip(rest.keys.cpu().tolist(), rest.ts.cpu().tolist(),
                rest.vals.cpu().tolist())
        )

    assert run(False) == run(True)
