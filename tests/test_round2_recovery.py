"""Recovery round-trips for the round-2 features: the string
dictionary's id assignment and the lowered `fold_window` columnar
state must both survive an abort + resume with exactly-once results.
"""

from datetime import datetime, timedelta, timezone

import pytest

torch = pytest.importorskip("torch")

import bytewax_amd.operators as op  # noqa: E402
import bytewax_amd.operators.windowing as win  # noqa: E402
from bytewax_amd.dataflow import Dataflow  # noqa: E402
from bytewax_amd.gpu import RecordBatch  # noqa: E402
from bytewax_amd.inputs import (  # noqa: E402
    AbortExecution,
    FixedPartitionedSource,
    StatefulSourcePartition,
)
from bytewax_amd.testing import TestingSink, run_main  # noqa: E402

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ALIGN_MS = int(ALIGN.timestamp() * 1000)
ZERO_TD = timedelta(seconds=0)


class _ScriptedPartition(StatefulSourcePartition):
    def __init__(self, items, abort_at, resume_state):
        self.items = items
        self.abort_at = abort_at
        self.i = resume_state if resume_state is not None else 0

    def next_batch(self):
        if self.abort_at is not None and self.i == self.abort_at:
            self.abort_at = None
            raise AbortExecution()
        if self.i >= len(self.items):
            raise StopIteration()
        item = self.items[self.i]
        self.i += 1
        return [item]

    def snapshot(self):
        return self.i


class ScriptedSource(FixedPartitionedSource):
    def __init__(self, items, abort_at=None):
        self.items = items
        self.abort_at = [abort_at]

    def list_parts(self):
        return ["p0"]

    def build_part(self, step_id, part, resume_state):
        abort_at = self.abort_at[0]
        self.abort_at[0] = None
        return _ScriptedPartition(self.items, abort_at, resume_state)


def _str_batches():
    words = ["apple", "pear", "fig", "apple", "plum", "fig", "apple", "pear"]
    out = []
    for i in range(0, len(words), 2):
        chunk = words[i : i + 2]
        out.append(
            (chunk, [ALIGN_MS + (i + j) * 500 for j in range(len(chunk))])
        )
    return out


def _run_str(recovery_config, source):
    from bytewax_amd.gpu.operators import keyed_window_agg_str

    out = []
    flow = Dataflow("str_rec")
    s = op.input("inp", flow, source)
    agg = keyed_window_agg_str(
        "agg", s, align_to=ALIGN, length=timedelta(seconds=60),
        device="cpu",
    )
    op.output("out", agg, TestingSink(out))
    run_main(flow, epoch_interval=ZERO_TD, recovery_config=recovery_config)
    res = {}
    for key, win_ms, val in out:
        res[(key, win_ms)] = res.get((key, win_ms), 0) + val
    return res


def test_string_dict_state_exactly_once_across_abort(recovery_config):
    batches = _str_batches()
    got1 = _run_str(recovery_config, ScriptedSource(batches, abort_at=2))
    got2 = _run_str(recovery_config, ScriptedSource(batches))
    combined = dict(got1)
    for k, v in got2.items():
        combined[k] = combined.get(k, 0) + v
    assert combined == {
        ("apple", ALIGN_MS): 3,
        ("pear", ALIGN_MS): 2,
        ("fig", ALIGN_MS): 2,
        ("plum", ALIGN_MS): 1,
    }


def _col_batches():
    out = []
    for i in range(6):
        out.append(
            RecordBatch(
                torch.tensor([i % 3, (i + 1) % 3], dtype=torch.int32),
                torch.tensor(
                    [ALIGN_MS + i * 700, ALIGN_MS + i * 700 + 10],
                    dtype=torch.int64,
                ),
                torch.tensor([1, 1], dtype=torch.int64),
            )
        )
    return out


def _run_lowered(recovery_config, source):
    out = []
    flow = Dataflow("lower_rec")
    s = op.input("inp", flow, source)
    keyed = op.key_on("k", s, lambda b: "shard-0")
    clock = win.EventClock(
        ts_getter=lambda it: it, wait_for_system_duration=ZERO_TD
    )
    wo = win.fold_window(
        "fw", keyed, clock,
        win.TumblingWindower(align_to=ALIGN, length=timedelta(seconds=60)),
        int, win.device_sum(), lambda a, b: a + b,
    )
    op.output("out", wo.down, TestingSink(out))
    run_main(flow, epoch_interval=ZERO_TD, recovery_config=recovery_config)
    res = {}
    for _k, (_wid, rb) in out:
        for key, ms, v in zip(
            rb.keys.cpu().tolist(), rb.ts.cpu().tolist(),
            rb.vals.cpu().tolist(),
        ):
            res[(key, ms)] = res.get((key, ms), 0) + v
    return res


def test_lowered_fold_window_exactly_once_across_abort(recovery_config):
    batches = _col_batches()
    got1 = _run_lowered(recovery_config, ScriptedSource(batches, abort_at=3))
    got2 = _run_lowered(recovery_config, ScriptedSource(batches))
    combined = dict(got1)
    for k, v in got2.items():
        combined[k] = combined.get(k, 0) + v
    # 6 batches x 2 events, keys 0..2 — all in the first window.
    assert sum(combined.values()) == 12
    assert combined == {
        (0, ALIGN_MS): 4,
        (1, ALIGN_MS): 4,
        (2, ALIGN_MS): 4,
    }
