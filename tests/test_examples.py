"""Example flows run end-to-end (regression net for docs/examples)."""

import os
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent


def _run(args, timeout=240):
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    return subprocess.run(
        args, capture_output=True, timeout=timeout, cwd=str(REPO), env=env
    )


def test_wordcount_example():
    res = _run(
        [sys.executable, "-m", "bytewax_amd.run", "examples.wordcount:flow"]
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "('question', 1)" in res.stdout.decode()


@pytest.mark.timeout(300)
def test_benchmark_windowing_example():
    res = _run(
        [
            sys.executable,
            "-m",
            "bytewax_amd.run",
            "examples.benchmark_windowing:flow",
        ],
        timeout=280,
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]


def test_gpu_wordcount_example_cpu_twin():
    res = _run([sys.executable, "examples/gpu_wordcount.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "counted 20000000 events" in res.stdout.decode()


def test_onebrc_example_cpu_twin():
    res = _run(
        [
            sys.executable,
            "examples/onebrc_gpu.py",
            "--rows",
            "200000",
            "--rows-per-batch",
            "50000",
        ]
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "aggregated 200000 rows over 10000 stations" in res.stdout.decode()


def test_csv_input_example():
    res = _run([sys.executable, "examples/csv_input.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "'instance':" in res.stdout.decode()


def test_join_example():
    res = _run([sys.executable, "examples/join.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert (
        "('123', ('Bumble', 'bee@example.com', 'yellow', 'buzz'))"
        in res.stdout.decode()
    )


def test_anomaly_detector_example():
    res = _run([sys.executable, "examples/anomaly_detector.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "v_metric:" in out and "hz_metric:" in out


def test_periodic_input_example():
    res = _run([sys.executable, "examples/periodic_input.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert res.stdout.decode().count("delay (ms):") == 8  # 2 flows x 4


def test_orderbook_example():
    res = _run([sys.executable, "examples/orderbook.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "BTC-USD" in out or "ETH-USD" in out
    assert "Summary(" in out


def test_search_session_example():
    res = _run([sys.executable, "examples/search_session.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "('1', 1.0)" in out and "('2', 0.0)" in out


def test_events_to_parquet_example(tmp_path):
    res = _run(
        [sys.executable, str(REPO / "examples" / "events_to_parquet.py")],
    )
    # The example writes ./parquet_demo_out relative to the cwd (repo
    # root under _run); assert and clean up.
    out_dir = REPO / "parquet_demo_out"
    try:
        assert res.returncode == 0, res.stderr.decode()[-1500:]
        assert "wrote" in res.stdout.decode()
        assert list(out_dir.rglob("*.parquet"))
    finally:
        import shutil

        shutil.rmtree(out_dir, ignore_errors=True)


def test_basic_example():
    res = _run([sys.executable, "examples/basic.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "<dance>" in res.stdout.decode()


def test_batch_operator_example():
    res = _run([sys.executable, "examples/batch_operator.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "avg batch:" in res.stdout.decode()


def test_event_time_processing_example():
    res = _run([sys.executable, "examples/event_time_processing.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "('LATE'" in out
    assert "('s1', (1, 19.0))" in out


def test_poll_and_split_example():
    res = _run([sys.executable, "examples/poll_and_split.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "'type': 'story'" in out and "'type': 'comment'" in out


def test_wikistream_example():
    res = _run([sys.executable, "examples/wikistream.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "edits" in res.stdout.decode()


def test_custom_metrics_example():
    res = _run([sys.executable, "examples/custom_metrics.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "next_batch_delay_seconds{" in res.stdout.decode()


def test_tracing_otlp_example():
    res = _run([sys.executable, "examples/tracing_otlp.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "bytes of spans" in res.stdout.decode()


def test_apriori_example():
    res = _run([sys.executable, "examples/apriori.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    out = res.stdout.decode()
    assert "('milk', 4)" in out and "('bread,milk', 3)" in out


def test_partials_example():
    res = _run([sys.executable, "examples/partials.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    # range(5) + 5 chained add_one steps -> 5..9
    assert "partials.insp: 9" in res.stdout.decode()


def test_split_demo_example():
    res = _run([sys.executable, "examples/split_demo.py"])
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    assert "_value" in res.stdout.decode()


def test_simple_kafka_example_builds():
    pytest.importorskip("confluent_kafka")
    res = _run(
        [
            sys.executable,
            "-c",
            "import examples.simple_kafka_in_and_out as m; print(m.flow.flow_id)",
        ]
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]


def test_serde_examples_compile():
    """The serde examples need a live broker + registry to IMPORT
    (module-level clients), so CI checks they at least compile."""
    import py_compile

    for name in ("confluent_serde.py", "redpanda_serde.py"):
        py_compile.compile(
            str(REPO / "examples" / name), doraise=True
        )
