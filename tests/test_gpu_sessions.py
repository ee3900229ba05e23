"""Device session windows: CPU twins (always) + device numerics vs a
host reference (gpu mark).  Parity: reference windowing.py
SessionWindower semantics under watermark-ordered input."""

from datetime import datetime, timedelta, timezone

import pytest

torch = pytest.importorskip("torch")

from bytewax_amd.gpu import AGG_COUNT, AGG_SUM, RecordBatch  # noqa: E402
from bytewax_amd.gpu.state import SessionAggState  # noqa: E402

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)


def _rows(out):
    if out is None:
        return []
    return sorted(
        zip(
            out["keys"].tolist(),
            out["start"].tolist(),
            out["end"].tolist(),
            out["vals"].tolist(),
        )
    )


def _ref_sessions(keys, ts, vals, gap_ms):
    """Serial reference: per-key in-order walk."""
    from collections import defaultdict

    per_key = defaultdict(list)
    for k, t, v in sorted(zip(keys, ts, vals), key=lambda r: (r[0], r[1])):
        per_key[k].append((t, v))
    out = []
    for k, evs in per_key.items():
        start, last, acc = None, None, 0
        for t, v in evs:
            if last is not None and t - last > gap_ms:
                out.append((k, start, last, acc))
                start, acc = None, 0
            if start is None:
                start = t
                acc = 0
            last = t
            acc += v
        if start is not None:
            out.append((k, start, last, acc))
    return sorted(out)


def test_session_gap_close_cpu():
    st = SessionAggState(torch.device("cpu"), gap_ms=100, mode=AGG_COUNT)
    st.insert(
        RecordBatch(
            torch.tensor([1, 1, 1, 2], dtype=torch.int32),
            torch.tensor([10, 50, 300, 20], dtype=torch.int64),
            None,
            max_ts=300,
        )
    )
    # Key 1's first session (10..50) closed by the 300 gap; key 2
    # idle-closed by the watermark (300 - gap = 200 > 20).
    closed = st.close_due()
    assert _rows(closed) == [(1, 10, 50, 2), (2, 20, 20, 1)]
    # EOF closes the rest.
    assert _rows(st.close_all()) == [(1, 300, 300, 1)]


def test_session_watermark_close_cpu():
    st = SessionAggState(torch.device("cpu"), gap_ms=100, mode=AGG_SUM)
    st.insert(
        RecordBatch(
            torch.tensor([5], dtype=torch.int32),
            torch.tensor([1000], dtype=torch.int64),
            torch.tensor([7], dtype=torch.int64),
            max_ts=1000,
        )
    )
    assert st.close_due() is None  # watermark 1000: still within gap
    st.insert(
        RecordBatch(
            torch.tensor([6], dtype=torch.int32),
            torch.tensor([5000], dtype=torch.int64),
            torch.tensor([1], dtype=torch.int64),
            max_ts=5000,
        )
    )
    # Watermark 5000 > 1000 + gap: key 5's session is idle-closed.
    assert _rows(st.close_due()) == [(5, 1000, 1000, 7)]


def test_session_snapshot_roundtrip_cpu():
    a = SessionAggState(torch.device("cpu"), gap_ms=100, mode=AGG_COUNT)
    a.insert(
        RecordBatch(
            torch.tensor([1, 2], dtype=torch.int32),
            torch.tensor([10, 20], dtype=torch.int64),
            None,
            max_ts=20,
        )
    )
    snap = a.snapshot_to_host()
    b = SessionAggState(torch.device("cpu"), gap_ms=100, mode=AGG_COUNT)
    b.restore_from_host(snap)
    assert _rows(b.close_all()) == _rows(a.close_all())


def test_keyed_session_agg_pipeline_cpu():
    import bytewax_amd.operators as op
    from bytewax_amd.dataflow import Dataflow
    from bytewax_amd.gpu.operators import keyed_session_agg
    from bytewax_amd.inputs import DynamicSource, StatelessSourcePartition
    from bytewax_amd.outputs import DynamicSink, StatelessSinkPartition
    from bytewax_amd.testing import run_main

    class _Src(StatelessSourcePartition):
        def __init__(self):
            self.i = 0

        def next_batch(self):
            if self.i >= 3:
                raise StopIteration()
            base = self.i * 1000
            self.i += 1
            return [
                RecordBatch(
                    torch.arange(4, dtype=torch.int32),
                    torch.full((4,), base, dtype=torch.int64),
                    None,
                    max_ts=base,
                )
            ]

    class Src(DynamicSource):
        def build(self, step_id, worker_index, worker_count):
            return _Src()

    class _Collect(StatelessSinkPartition):
        def __init__(self, ls):
            self._ls = ls

        def write_batch(self, items):
            self._ls.extend(items)

    class Sink(DynamicSink):
        def __init__(self, ls):
            self._ls = ls

        def build(self, step_id, worker_index, worker_count):
            return _Collect(self._ls)

    out = []
    flow = Dataflow("sess")
    s = op.input("inp", flow, Src())
    # Gap 500ms < 1000ms batch spacing: every event is its own session.
    agg = keyed_session_agg(
        "sess_agg", s, gap=timedelta(milliseconds=500), device="cpu"
    )
    op.output("out", agg, Sink(out))
    run_main(flow)
    rows = [r for d in out for r in _rows(d)]
    assert len(rows) == 12  # 3 batches x 4 keys, all singleton sessions
    assert all(v == 1 for _k, _s, _e, v in rows)


@pytest.mark.gpu
@pytest.mark.parametrize("mode_name", ["count", "sum"])
def test_session_gpu_matches_reference(mode_name):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    g = torch.Generator().manual_seed(12)
    n = 200_000
    vocab = 5_000
    gap = 500
    keys = torch.randint(0, vocab, (n,), dtype=torch.int32, generator=g)
    # Timestamps over a 20s span: plenty of per-key gaps > 500ms.
    ts = torch.randint(0, 20_000, (n,), dtype=torch.int64, generator=g)
    vals = torch.randint(1, 50, (n,), dtype=torch.int64, generator=g)
    mode = AGG_COUNT if mode_name == "count" else AGG_SUM
    ref = _ref_sessions(
        keys.tolist(),
        ts.tolist(),
        [1] * n if mode == AGG_COUNT else vals.tolist(),
        gap,
    )
    st = SessionAggState(
        torch.device("cuda:0"), gap_ms=gap, mode=mode, slots_pow=14,
        out_cap=n,
    )
    st.insert(
        RecordBatch(keys.cuda(), ts.cuda(), vals.cuda(), max_ts=20_000)
    )
    got = []
    closed = st.close_due()
    if closed is not None:
        got.extend(
            _rows({k: t.cpu() for k, t in closed.items()})
        )
    final = st.close_all()
    if final is not None:
        got.extend(_rows({k: t.cpu() for k, t in final.items()}))
    assert sorted(got) == ref


@pytest.mark.gpu
def test_session_gpu_snapshot_roundtrip():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    g = torch.Generator().manual_seed(13)
    n = 50_000
    keys = torch.randint(0, 1000, (n,), dtype=torch.int32, generator=g)
    ts = torch.randint(0, 5_000, (n,), dtype=torch.int64, generator=g)
    a = SessionAggState(
        torch.device("cuda:0"), gap_ms=10_000, mode=AGG_COUNT,
        slots_pow=12, out_cap=n,
    )
    a.insert(RecordBatch(keys.cuda(), ts.cuda(), None, max_ts=5_000))
    snap = a.snapshot_to_host()
    b = SessionAggState(
        torch.device("cuda:0"), gap_ms=10_000, mode=AGG_COUNT,
        slots_pow=12, out_cap=n,
    )
    b.restore_from_host(snap)
    ra = _rows({k: t.cpu() for k, t in a.close_all().items()})
    rb = _rows({k: t.cpu() for k, t in b.close_all().items()})
    assert ra == rb


@pytest.mark.timeout(300)
def test_sessions_two_ranks_gloo(tmp_path):
    """keyed_session_agg across 2 gloo ranks: the sync exchange is
    collective per step; totals must be exact."""
    import os
    import subprocess
    import sys
    import textwrap
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    prog = tmp_path / "sess2.py"
    prog.write_text(
        textwrap.dedent(
            """
            import os
            import torch
            import torch.distributed as dist

            import bytewax_amd.operators as op
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.gpu import RecordBatch
            from bytewax_amd.gpu.operators import keyed_session_agg
            from bytewax_amd.inputs import (
                DynamicSource,
                StatelessSourcePartition,
            )
            from bytewax_amd.outputs import (
                DynamicSink,
                StatelessSinkPartition,
            )
            from bytewax_amd.testing import run_main
            from datetime import timedelta

            dist.init_process_group("gloo")
            rank = dist.get_rank()

            class _Src(StatelessSourcePartition):
                def __init__(self):
                    self.i = 0

                def next_batch(self):
                    if self.i >= 3:
                        raise StopIteration()
                    base = self.i * 1000
                    self.i += 1
                    # Both ranks emit the same 6 keys each step.
                    return [
                        RecordBatch(
                            torch.arange(6, dtype=torch.int32),
                            torch.full((6,), base, dtype=torch.int64),
                            None,
                            max_ts=base,
                        )
                    ]

            class Src(DynamicSource):
                def build(self, step_id, wi, wc):
                    return _Src()

            total = []

            class _Sink(StatelessSinkPartition):
                def write_batch(self, items):
                    for d in items:
                        total.extend(d["vals"].tolist())

            class Sink(DynamicSink):
                def build(self, step_id, wi, wc):
                    return _Sink()

            flow = Dataflow("sess2")
            s = op.input("inp", flow, Src())
            agg = keyed_session_agg(
                "sess", s, gap=timedelta(milliseconds=500), device="cpu"
            )
            op.output("out", agg, Sink())
            run_main(flow)
            t = torch.tensor([sum(total)], dtype=torch.int64)
            dist.all_reduce(t)
            if rank == 0:
                # 3 steps x 6 keys x 2 ranks, singleton sessions of
                # count 2 (both ranks' copies merge on the owner).
                assert int(t.item()) == 36, int(t.item())
                print("SESS2 OK")
            dist.destroy_process_group()
            """
        )
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = str(repo)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(29880 + os.getpid() % 100)
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK=str(rank))
        procs.append(
            subprocess.Popen(
                [sys.executable, str(prog)],
                env=e,
                stdout=subprocess.PIPE,
                stderr=subprocess.PIPE,
            )
        )
    outs = [p.communicate(timeout=240) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se.decode()[-1500:]
    assert "SESS2 OK" in outs[0][0].decode()


@pytest.mark.gpu
def test_session_batch_fast_path_matches_sort_path():
    """gap >= batch span triggers the stats-kernel fast path; results
    must equal the sorted walk on identical data (absolute-ts batches
    force the sort path)."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    dev = torch.device("cuda:0")
    gap = 5_000
    span = 1_000
    bases = [0, 1_000, 10_000, 11_000, 30_000]  # jumps split sessions
    g = torch.Generator().manual_seed(17)
    raw = []
    for base in bases:
        n = 40_000
        keys = torch.randint(0, 2_000, (n,), dtype=torch.int32, generator=g)
        ts0 = torch.randint(0, span, (n,), dtype=torch.int64, generator=g)
        raw.append((keys, ts0, base))

    def run(fast):
        st = SessionAggState(
            dev, gap_ms=gap, mode=AGG_COUNT, slots_pow=13, out_cap=1 << 20
        )
        got = []
        for keys, ts0, base in raw:
            if fast:
                b = RecordBatch(
                    keys.cuda(), ts0.cuda(), None,
                    max_ts=base + span, ts_base=base,
                )
            else:
                b = RecordBatch(
                    keys.cuda(), (ts0 + base).cuda(), None,
                    max_ts=base + span,
                )
            st.insert(b)
            out = st.close_due()
            if out is not None:
                got.extend(_rows({k: t.cpu() for k, t in out.items()}))
        fin = st.close_all()
        if fin is not None:
            got.extend(_rows({k: t.cpu() for k, t in fin.items()}))
        return sorted(got)

    slow = run(False)
    fast = run(True)
    assert fast == slow
    ref = _ref_sessions(
        [k for keys, _t, _b in raw for k in keys.tolist()],
        [t + b for _k, ts0, b in raw for t in ts0.tolist()],
        [1] * sum(len(keys) for keys, _t, _b in raw),
        gap,
    )
    assert fast == ref
