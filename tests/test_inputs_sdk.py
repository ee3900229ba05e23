"""Input SDK helper behavior (parity: reference pytests for inputs)."""

import asyncio
from datetime import datetime, timedelta, timezone

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.inputs import (
    SimplePollingSource,
    batch,
    batch_async,
    batch_getter,
    batch_getter_ex,
)
from bytewax_amd.testing import TestingSink, TestingSource, poll_next_batch, run_main


def test_batch():
    out = list(batch(range(7), 3))
    assert out == [[0, 1, 2], [3, 4, 5], [6]]


def test_batch_getter():
    items = [1, 2, 3, None, 4]
    it = iter(items)

    def getter():
        try:
            return next(it)
        except StopIteration:
            return None

    b = batch_getter(getter, 2)
    assert next(b) == [1, 2]
    assert next(b) == [3]


def test_batch_getter_ex():
    stack = [3, 2, 1]

    def getter():
        if stack:
            return stack.pop()
        raise IndexError()

    b = batch_getter_ex(getter, 10, yield_ex=IndexError)
    assert next(b) == [1, 2, 3]
    assert next(b) == []


def test_batch_async():
    async def agen():
        for i in range(5):
            yield i

    b = batch_async(agen(), timeout=timedelta(seconds=1), batch_size=2)
    got = list(b)
    assert [x for chunk in got for x in chunk] == [0, 1, 2, 3, 4]


def test_simple_polling_source():
    class CountSource(SimplePollingSource):
        def __init__(self):
            super().__init__(interval=timedelta(0))
            self.n = 0

        def next_item(self):
            self.n += 1
            if self.n > 3:
                raise StopIteration()
            return self.n

    # Drive the partition directly (the flow-level path would poll
    # forever since SimplePollingSource never ends).
    part = CountSource().build_part("s", "singleton", None)
    got = []
    for _ in range(3):
        got.extend(poll_next_batch(part))
    assert got == [1, 2, 3]


def test_simple_polling_source_retry():
    class FlakySource(SimplePollingSource):
        def __init__(self):
            super().__init__(interval=timedelta(0))
            self.calls = 0

        def next_item(self):
            self.calls += 1
            if self.calls == 1:
                raise SimplePollingSource.Retry(timedelta(milliseconds=1))
            return "ok"

    part = FlakySource().build_part("s", "singleton", None)
    assert part.next_batch() == []
    awake = part.next_awake()
    assert awake is not None
    got = poll_next_batch(part)
    assert got == ["ok"]


def test_pause_sentinel_resumes_emission():
    out = []
    flow = Dataflow("f")
    src = TestingSource(
        [1, TestingSource.PAUSE(timedelta(milliseconds=50)), 2]
    )
    s = op.input("inp", flow, src)
    op.output("out", s, TestingSink(out))
    run_main(flow)
    assert out == [1, 2]


def test_testing_source_batch_size():
    out = []
    flow = Dataflow("f")
    s = op.input("inp", flow, TestingSource(list(range(10)), batch_size=4))
    s = op.flat_map_batch("ident", s, lambda xs: [len(xs)])
    op.output("out", s, TestingSink(out))
    run_main(flow)
    assert out == [4, 4, 2]
