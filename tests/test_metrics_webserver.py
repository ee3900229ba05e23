"""Metrics + ops webserver tests (parity: reference webserver/metrics)."""

import json
import os
import time
import urllib.request

import pytest

import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource, run_main


def test_metrics_counters():
    from bytewax_amd._metrics import enable_metrics, generate_python_metrics

    enable_metrics()
    out = []
    flow = Dataflow("metrics_flow")
    s = op.input("inp", flow, TestingSource([1, 2, 3]))
    s = op.map("double", s, lambda x: x * 2)
    op.output("out", s, TestingSink(out))
    run_main(flow)
    text = generate_python_metrics().decode()
    assert "bytewax_item_inp_count" in text
    assert 'step_id="metrics_flow.double.flat_map_batch"' in text


def test_webserver_serves_dataflow_and_metrics(monkeypatch):
    port = 31000 + os.getpid() % 500
    monkeypatch.setenv("BYTEWAX_DATAFLOW_API_ENABLED", "1")
    monkeypatch.setenv("BYTEWAX_DATAFLOW_API_PORT", str(port))

    out = []
    flow = Dataflow("ws_flow")
    s = op.input("inp", flow, TestingSource(list(range(10))))
    op.output("out", s, TestingSink(out))
    run_main(flow)

    body = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/dataflow", timeout=5
    ).read()
    rendered = json.loads(body)
    assert rendered["flow_id"] == "ws_flow"
    metrics = urllib.request.urlopen(
        f"http://127.0.0.1:{port}/metrics", timeout=5
    ).read()
    assert b"bytewax" in metrics
