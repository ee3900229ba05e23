"""The `bytewax` compatibility alias: reference-style user code runs
unchanged."""

import re


def test_reference_style_wordcount_runs():
    # Written exactly as a Bytewax user would write it.
    import bytewax.operators as op
    from bytewax.dataflow import Dataflow
    from bytewax.testing import TestingSink, TestingSource, run_main

    lines = ["to be or not to be", "that is the question"]

    def tokenize(line):
        return re.findall(r"[^\s]+", line)

    out = []
    flow = Dataflow("wordcount")
    s = op.input("inp", flow, TestingSource(lines))
    s = op.flat_map("tokenize", s, tokenize)
    counts = op.count_final("count", s, lambda w: w)
    op.output("out", counts, TestingSink(out))
    run_main(flow)
    assert ("to", 2) in out
    assert ("be", 2) in out
    assert ("question", 1) in out


def test_reference_style_windowing_runs():
    from datetime import datetime, timedelta, timezone

    import bytewax.operators as op
    import bytewax.operators.windowing as w
    from bytewax.dataflow import Dataflow
    from bytewax.operators.windowing import EventClock, TumblingWindower
    from bytewax.testing import TestingSink, TestingSource, run_main

    align_to = datetime(2022, 1, 1, tzinfo=timezone.utc)
    inp = [align_to + timedelta(seconds=i) for i in range(120)]
    clock = EventClock(
        ts_getter=lambda x: x, wait_for_system_duration=timedelta(0)
    )
    windower = TumblingWindower(align_to=align_to, length=timedelta(minutes=1))

    out = []
    flow = Dataflow("bench")
    wo = (
        op.input("in", flow, TestingSource(inp, 10))
        .then(op.key_on, "key-on", lambda _x: "ALL")
        .then(
            w.fold_window,
            "fold-window",
            clock,
            windower,
            list,
            lambda acc, x: acc + [x],
            lambda a, b: a + b,
        )
    )
    counts = op.map_value("count", wo.down, lambda wa: (wa[0], len(wa[1])))
    op.output("out", counts, TestingSink(out))
    run_main(flow)
    assert sorted(out) == [("ALL", (0, 60)), ("ALL", (1, 60))]


def test_connectors_alias():
    from bytewax.connectors.files import FileSource  # noqa: F401
    from bytewax.connectors.stdio import StdOutSink  # noqa: F401
    from bytewax.recovery import RecoveryConfig  # noqa: F401
    from bytewax.inputs import FixedPartitionedSource  # noqa: F401
    from bytewax.outputs import DynamicSink  # noqa: F401
