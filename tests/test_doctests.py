"""Documented-output exactness for the operator stdlib.

The reference runs `testcode`/`testoutput` blocks from every operator
docstring in CI (reference justfile:93-94; 88 blocks in
pysrc/bytewax/operators/__init__.py alone).  Our equivalent: operator
docstrings carry standard doctests (`Example:` sections) executed
here with a shared namespace, asserting the documented output
byte-for-byte.
"""

import doctest
from datetime import datetime, timedelta, timezone

import bytewax_amd.operators as op
import bytewax_amd.operators.windowing as win
from bytewax_amd.connectors.stdio import StdOutSink
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource, run_main

GLOBS = {
    "op": op,
    "win": win,
    "Dataflow": Dataflow,
    "TestingSource": TestingSource,
    "TestingSink": TestingSink,
    "StdOutSink": StdOutSink,
    "run_main": run_main,
    "datetime": datetime,
    "timedelta": timedelta,
    "timezone": timezone,
}


def _run(module, min_attempted: int):
    res = doctest.testmod(
        module,
        extraglobs=dict(GLOBS),
        optionflags=doctest.NORMALIZE_WHITESPACE,
        verbose=False,
    )
    assert res.failed == 0, f"{res.failed} doctest failures in {module}"
    assert res.attempted >= min_attempted, (
        f"expected >= {min_attempted} doctest statements in {module}, "
        f"ran {res.attempted}"
    )


def test_operator_docstring_examples():
    _run(op, 100)


def test_windowing_docstring_examples():
    _run(win, 40)


def test_inputs_sdk_docstring_examples():
    import bytewax_amd.inputs as inputs

    _run(inputs, 20)


def test_outputs_sdk_docstring_examples():
    import bytewax_amd.outputs as outputs

    _run(outputs, 2)


def test_kafka_message_docstring_examples():
    import bytewax_amd.connectors.kafka as kafka

    _run(kafka, 2)
