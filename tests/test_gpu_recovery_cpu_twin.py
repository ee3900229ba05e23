"""Device-state recovery through the real engine machinery (CPU twin).

The window-agg operator's snapshot is the host spill of the HBM table;
here we verify it rides the SQLite recovery store with exactly-once
semantics: an ABORT mid-stream, then resume, must produce exactly the
same window counts as an uninterrupted run.
"""

from datetime import datetime, timedelta, timezone
from typing import List, Optional

import pytest

torch = pytest.importorskip("torch")

import bytewax_amd.operators as op  # noqa: E402
from bytewax_amd.dataflow import Dataflow  # noqa: E402
from bytewax_amd.gpu import RecordBatch  # noqa: E402
from bytewax_amd.gpu.operators import (  # noqa: E402
    CollectCountsSink,
    keyed_window_agg,
)
from bytewax_amd.inputs import (  # noqa: E402
    AbortExecution,
    FixedPartitionedSource,
    StatefulSourcePartition,
)
from bytewax_amd.testing import run_main  # noqa: E402

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ZERO_TD = timedelta(seconds=0)


class _ScriptedPartition(StatefulSourcePartition):
    def __init__(self, batches, abort_at, resume_state: Optional[int]):
        self.batches = batches
        self.abort_at = abort_at
        self.i = resume_state if resume_state is not None else 0

    def next_batch(self) -> List[RecordBatch]:
        if self.abort_at is not None and self.i == self.abort_at:
            # One-shot, like TestingSource.ABORT.
            self.abort_at = None
            raise AbortExecution()
        if self.i >= len(self.batches):
            raise StopIteration()
        b = self.batches[self.i]
        self.i += 1
        return [b]

    def snapshot(self) -> int:
        return self.i


class ScriptedSource(FixedPartitionedSource):
    """Replayable batch script with a snapshotted cursor."""

    def __init__(self, batches, abort_at=None):
        self.batches = batches
        self.abort_at = [abort_at]  # shared, one-shot across builds

    def list_parts(self):
        return ["script"]

    def build_part(self, step_id, for_part, resume_state):
        abort_at = self.abort_at[0]
        self.abort_at[0] = None
        return _ScriptedPartition(self.batches, abort_at, resume_state)


def _mk_batches(align_ms):
    def b(start_s, keys):
        n = len(keys)
        return RecordBatch(
            torch.tensor(keys, dtype=torch.int32),
            torch.full((n,), align_ms + start_s * 1000, dtype=torch.int64),
            max_ts=align_ms + start_s * 1000,
        )

    return [
        b(0, [1, 1, 2]),
        b(10, [1, 3]),
        b(70, [2, 2]),  # watermark passes window 0 here
        b(80, [1]),
    ]


def _counts(out, align_ms):
    got = {}
    for batch in out:
        for k, t, v in zip(
            batch.keys.tolist(), batch.ts.tolist(), batch.vals.tolist()
        ):
            w = (t - align_ms) // 60_000
            got[(k, w)] = got.get((k, w), 0) + v
    return got


def _run(recovery_config, source, device="cpu"):
    from bytewax_amd.gpu import _ms

    align_ms = _ms(ALIGN)
    out = []
    flow = Dataflow("gpurec")
    s = op.input("inp", flow, source)
    agg = keyed_window_agg(
        "agg",
        s,
        align_to=ALIGN,
        length=timedelta(minutes=1),
        device=device,
        exchange=False,
    )
    op.output("out", agg, CollectCountsSink(out))
    run_main(flow, epoch_interval=ZERO_TD, recovery_config=recovery_config)
    return _counts(out, align_ms)


EXPECTED = {
    (1, 0): 3,  # key 1: batches 0 (x2) + 1
    (2, 0): 1,
    (3, 0): 1,
    (2, 1): 2,  # batch 2 at t=70
    (1, 1): 1,  # batch 3 at t=80
}


def test_window_state_exactly_once_across_abort(recovery_config):
    from bytewax_amd.gpu import _ms

    align_ms = _ms(ALIGN)
    batches = _mk_batches(align_ms)

    # Aborts while window 0 is still open (after 2 batches).
    got1 = _run(recovery_config, ScriptedSource(batches, abort_at=2))
    assert got1 == {}  # nothing closed before the abort

    # Resume: source cursor and the window table snapshot both restore
    # from the epoch before the abort; the result must equal an
    # uninterrupted run — no lost or double-counted events.
    got2 = _run(recovery_config, ScriptedSource(batches))
    assert got2 == EXPECTED


def test_window_uninterrupted_baseline(tmp_path):
    from bytewax_amd.gpu import _ms
    from bytewax_amd.recovery import RecoveryConfig, init_db_dir

    db = tmp_path / "db"
    db.mkdir()
    init_db_dir(db, 2)
    align_ms = _ms(ALIGN)
    got = _run(RecoveryConfig(db), ScriptedSource(_mk_batches(align_ms)))
    assert got == EXPECTED


@pytest.mark.gpu
def test_window_state_exactly_once_across_abort_gpu(recovery_config):
    """Same exactly-once contract with the real device table: the
    snapshot is the pinned-host spill of HBM state, written into the
    SQLite store and restored on resume."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from bytewax_amd.gpu import _ms

    align_ms = _ms(ALIGN)
    batches = [
        RecordBatch(b.keys.cuda(), b.ts.cuda(), max_ts=b.max_ts)
        for b in _mk_batches(align_ms)
    ]
    got1 = _run(
        recovery_config, ScriptedSource(batches, abort_at=2), device="cuda"
    )
    assert got1 == {}
    got2 = _run(recovery_config, ScriptedSource(batches), device="cuda")
    assert got2 == EXPECTED
