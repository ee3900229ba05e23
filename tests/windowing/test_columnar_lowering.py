"""Public windowing API lowering onto the GPU kernels (VERDICT r1
item 3): ONE `fold_window`/`count_window` flow definition runs on both
engines, chosen by stream type — host `_WindowLogic` for
Python-object values, `WindowAggState` HIP kernels for RecordBatch
values.
"""

import random
from datetime import datetime, timedelta, timezone

import pytest

torch = pytest.importorskip("torch")

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as w
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.gpu import RecordBatch
from bytewax_amd.operators.windowing import (
    COLUMNAR_WINDOW_ID,
    EventClock,
    TumblingWindower,
)
from bytewax_amd.testing import TestingSink, TestingSource, run_main

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ALIGN_MS = int(ALIGN.timestamp() * 1000)
WINDOW = timedelta(seconds=60)


def _mk_events(n, vocab=20, seed=3):
    rng = random.Random(seed)
    return [
        (rng.randrange(vocab), ALIGN_MS + i * 700, rng.randrange(100))
        for i in range(n)
    ]


def _windowed(flow_name, source_items, folder, ts_getter, key_fn):
    """The ONE flow definition: input -> key_on -> fold_window."""
    out = []
    flow = Dataflow(flow_name)
    s = op.input("inp", flow, TestingSource(source_items))
    keyed = op.key_on("k", s, key_fn)
    clock = EventClock(
        ts_getter=ts_getter, wait_for_system_duration=timedelta(0)
    )
    wo = w.fold_window(
        "fw", keyed, clock,
        TumblingWindower(align_to=ALIGN, length=WINDOW),
        int, folder, lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    return out


def _host_run(events, folder):
    items = [
        (datetime.fromtimestamp(ms / 1000, tz=timezone.utc), k, v)
        for k, ms, v in events
    ]
    out = _windowed(
        "host", items, folder,
        ts_getter=lambda it: it[0], key_fn=lambda it: str(it[1]),
    )
    win_len = int(WINDOW.total_seconds() * 1000)
    return {
        (int(key), ALIGN_MS + wid * win_len): acc
        for key, (wid, acc) in out
    }


def _columnar_run(events, folder, device="cpu", batch=41):
    batches = []
    for i in range(0, len(events), batch):
        chunk = events[i : i + batch]
        batches.append(
            RecordBatch(
                torch.tensor([k for k, _, _ in chunk], dtype=torch.int32,
                             device=device),
                torch.tensor([ms for _, ms, _ in chunk], dtype=torch.int64,
                             device=device),
                torch.tensor([v for _, _, v in chunk], dtype=torch.int64,
                             device=device),
            )
        )
    out = _windowed(
        "columnar", batches, folder,
        ts_getter=lambda it: it, key_fn=lambda b: "shard-0",
    )
    res = {}
    for _key, (wid, rb) in out:
        assert wid == COLUMNAR_WINDOW_ID
        for k, ms, v in zip(
            rb.keys.cpu().tolist(), rb.ts.cpu().tolist(),
            rb.vals.cpu().tolist(),
        ):
            res[(k, ms)] = res.get((k, ms), 0) + v
    return res


def test_fold_window_sum_lowering_cpu_twin():
    events = _mk_events(800)
    folder = w.device_sum(lambda it: it[2])
    assert _columnar_run(events, folder) == _host_run(events, folder)


def test_fold_window_count_lowering_cpu_twin():
    events = _mk_events(800)
    folder = w.device_count()
    assert _columnar_run(events, folder) == _host_run(events, folder)


def test_plain_python_folder_stays_on_host():
    # A non-sentinel folder must NOT lower; the flow still works on
    # host items exactly as before.
    events = _mk_events(100)
    res = _host_run(events, lambda acc, it: acc + it[2])
    assert sum(res.values()) == sum(v for _k, _ms, v in events)


@pytest.mark.gpu
def test_fold_window_sum_lowering_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=500)
    folder = w.device_sum(lambda it: it[2])
    assert _columnar_run(events, folder, "cuda:0") == _host_run(
        events, folder
    )


@pytest.mark.gpu
def test_count_window_lowering_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=500)
    # count_window over a columnar stream: the key callable returns
    # the shard key for batches.
    batches = []
    for i in range(0, len(events), 64):
        chunk = events[i : i + 64]
        batches.append(
            RecordBatch(
                torch.tensor([k for k, _, _ in chunk], dtype=torch.int32,
                             device="cuda:0"),
                torch.tensor([ms for _, ms, _ in chunk], dtype=torch.int64,
                             device="cuda:0"),
            )
        )
    out = []
    flow = Dataflow("cw_col")
    s = op.input("inp", flow, TestingSource(batches))
    clock = EventClock(
        ts_getter=lambda it: it, wait_for_system_duration=timedelta(0)
    )
    wo = w.count_window(
        "cw", s, clock,
        TumblingWindower(align_to=ALIGN, length=WINDOW),
        key=lambda b: "shard-0",
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    total = 0
    for _key, (wid, rb) in out:
        assert wid == COLUMNAR_WINDOW_ID
        total += int(rb.vals.sum().item())
    assert total == len(events)


def _sliding_run(events, folder, device="cpu", batch=41):
    from bytewax_amd.operators.windowing import SlidingWindower

    batches = []
    for i in range(0, len(events), batch):
        chunk = events[i : i + batch]
        batches.append(
            RecordBatch(
                torch.tensor([k for k, _, _ in chunk], dtype=torch.int32,
                             device=device),
                torch.tensor([ms for _, ms, _ in chunk], dtype=torch.int64,
                             device=device),
                torch.tensor([v for _, _, v in chunk], dtype=torch.int64,
                             device=device),
            )
        )
    out = []
    flow = Dataflow("columnar_sliding")
    s = op.input("inp", flow, TestingSource(batches))
    keyed = op.key_on("k", s, lambda b: "shard-0")
    clock = EventClock(
        ts_getter=lambda it: it, wait_for_system_duration=timedelta(0)
    )
    wo = w.fold_window(
        "fw", keyed, clock,
        SlidingWindower(
            align_to=ALIGN, length=WINDOW, offset=timedelta(seconds=20)
        ),
        int, folder, lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    res = {}
    for _key, (wid, rb) in out:
        assert wid == COLUMNAR_WINDOW_ID
        for k, ms, v in zip(
            rb.keys.cpu().tolist(), rb.ts.cpu().tolist(),
            rb.vals.cpu().tolist(),
        ):
            res[(k, ms)] = res.get((k, ms), 0) + v
    return res


def _sliding_host(events, folder):
    from bytewax_amd.operators.windowing import SlidingWindower

    items = [
        (datetime.fromtimestamp(ms / 1000, tz=timezone.utc), k, v)
        for k, ms, v in events
    ]
    out = []
    flow = Dataflow("host_sliding")
    s = op.input("inp", flow, TestingSource(items))
    keyed = op.key_on("k", s, lambda it: str(it[1]))
    clock = EventClock(
        ts_getter=lambda it: it[0], wait_for_system_duration=timedelta(0)
    )
    wo = w.fold_window(
        "fw", keyed, clock,
        SlidingWindower(
            align_to=ALIGN, length=WINDOW, offset=timedelta(seconds=20)
        ),
        int, folder, lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    off_ms = 20_000
    return {
        (int(key), ALIGN_MS + wid * off_ms): acc
        for key, (wid, acc) in out
    }


def test_sliding_lowering_cpu_twin_matches_host():
    events = _mk_events(600)
    folder = w.device_sum(lambda it: it[2])
    assert _sliding_run(events, folder) == _sliding_host(events, folder)


@pytest.mark.gpu
def test_sliding_lowering_gpu_matches_host():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=300)
    folder = w.device_sum(lambda it: it[2])
    assert _sliding_run(events, folder, "cuda:0") == _sliding_host(
        events, folder
    )


def _minmax_host(events, folder, builder):
    items = [
        (datetime.fromtimestamp(ms / 1000, tz=timezone.utc), k, v)
        for k, ms, v in events
    ]
    out = []
    flow = Dataflow("host_minmax")
    s = op.input("inp", flow, TestingSource(items))
    keyed = op.key_on("k", s, lambda it: str(it[1]))
    clock = EventClock(
        ts_getter=lambda it: it[0], wait_for_system_duration=timedelta(0)
    )
    wo = w.fold_window(
        "fw", keyed, clock,
        TumblingWindower(align_to=ALIGN, length=WINDOW),
        builder, folder, lambda a, b: folder(a, (None, None, b)),
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    win_len = int(WINDOW.total_seconds() * 1000)
    return {
        (int(key), ALIGN_MS + wid * win_len): acc
        for key, (wid, acc) in out
    }


def test_fold_window_min_max_lowering_cpu_twin():
    events = _mk_events(700)
    for name, builder in (("min", lambda: 10**9), ("max", lambda: -1)):
        folder = (
            w.device_min(lambda it: it[2])
            if name == "min"
            else w.device_max(lambda it: it[2])
        )
        host = {
            k: v
            for k, v in _minmax_host(events, folder, builder).items()
        }
        col = _columnar_run(events, folder)
        assert col == host, name


@pytest.mark.gpu
def test_fold_window_min_max_lowering_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=400)
    folder = w.device_max(lambda it: it[2])
    assert _columnar_run(events, folder, "cuda:0") == _minmax_host(
        events, folder, lambda: -1
    )


def _mean_host(events, folder):
    """Host path for device_mean: (sum, count) accumulator + the
    finisher wired by fold_window (tests the finish map too)."""
    items = [
        (datetime.fromtimestamp(ms / 1000, tz=timezone.utc), k, v)
        for k, ms, v in events
    ]
    out = []
    flow = Dataflow("host_mean")
    s = op.input("inp", flow, TestingSource(items))
    keyed = op.key_on("k", s, lambda it: str(it[1]))
    clock = EventClock(
        ts_getter=lambda it: it[0], wait_for_system_duration=timedelta(0)
    )
    wo = w.fold_window(
        "fw", keyed, clock,
        TumblingWindower(align_to=ALIGN, length=WINDOW),
        lambda: (0, 0), folder,
        lambda a, b: (a[0] + b[0], a[1] + b[1]),
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    win_len = int(WINDOW.total_seconds() * 1000)
    return {
        (int(key), ALIGN_MS + wid * win_len): acc
        for key, (wid, acc) in out
    }


def test_fold_window_mean_lowering_cpu_twin():
    events = _mk_events(700)
    folder = w.device_mean(lambda it: it[2])
    host = _mean_host(events, folder)
    assert all(isinstance(v, float) for v in host.values())
    assert _columnar_run(events, folder) == host


@pytest.mark.gpu
def test_fold_window_mean_lowering_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=400)
    folder = w.device_mean(lambda it: it[2])
    assert _columnar_run(events, folder, "cuda:0") == _mean_host(
        events, folder
    )


def _minmax_columnar_run(events, which, device="cpu", batch=37):
    """max_window/min_window over RecordBatch streams (auto
    lowering through the reduce shim's DeviceFoldable)."""
    batches = []
    for i in range(0, len(events), batch):
        chunk = events[i : i + batch]
        batches.append(
            RecordBatch(
                torch.tensor([k for k, _, _ in chunk], dtype=torch.int32,
                             device=device),
                torch.tensor([ms for _, ms, _ in chunk], dtype=torch.int64,
                             device=device),
                torch.tensor([v for _, _, v in chunk], dtype=torch.int64,
                             device=device),
            )
        )
    out = []
    flow = Dataflow("columnar_minmax_pub")
    s = op.input("inp", flow, TestingSource(batches))
    keyed = op.key_on("k", s, lambda b: "shard-0")
    clock = EventClock(
        ts_getter=lambda it: it, wait_for_system_duration=timedelta(0)
    )
    fn = w.max_window if which == "max" else w.min_window
    wo = fn("mw", keyed, clock,
            TumblingWindower(align_to=ALIGN, length=WINDOW))
    op.output("out", wo.down, TestingSink(out))
    run_main(flow)
    res = {}
    for _key, (wid, rb) in out:
        assert wid == COLUMNAR_WINDOW_ID
        for k, ms, v in zip(
            rb.keys.cpu().tolist(), rb.ts.cpu().tolist(),
            rb.vals.cpu().tolist(),
        ):
            res[(k, ms)] = v
    return res


def test_public_max_min_window_lower_on_columnar_cpu_twin():
    events = _mk_events(600)
    win_len = int(WINDOW.total_seconds() * 1000)
    for which, red in (("max", max), ("min", min)):
        brute = {}
        for k, ms, v in events:
            cell = (k, ALIGN_MS + ((ms - ALIGN_MS) // win_len) * win_len)
            brute[cell] = red(brute.get(cell, v), v)
        assert _minmax_columnar_run(events, which) == brute, which


@pytest.mark.gpu
def test_public_max_window_lowers_on_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    events = _mk_events(20_000, vocab=300)
    win_len = int(WINDOW.total_seconds() * 1000)
    brute = {}
    for k, ms, v in events:
        cell = (k, ALIGN_MS + ((ms - ALIGN_MS) // win_len) * win_len)
        brute[cell] = max(brute.get(cell, v), v)
    assert _minmax_columnar_run(events, "max", "cuda:0") == brute


def test_sliding_mean_stays_on_host():
    """Sliding + device_mean must NOT lower (the stats backend is
    tumbling-only); the host path computes the overlapped windows."""
    from bytewax_amd.operators.windowing import SlidingWindower, _ColumnarSpec

    spec = _ColumnarSpec.resolve(
        "mean",
        EventClock(ts_getter=lambda it: it,
                   wait_for_system_duration=timedelta(0)),
        SlidingWindower(
            align_to=ALIGN, length=WINDOW, offset=timedelta(seconds=20)
        ),
    )
    assert spec is None
    # Tumbling mean does lower.
    spec = _ColumnarSpec.resolve(
        "mean",
        EventClock(ts_getter=lambda it: it,
                   wait_for_system_duration=timedelta(0)),
        TumblingWindower(align_to=ALIGN, length=WINDOW),
    )
    assert spec is not None
