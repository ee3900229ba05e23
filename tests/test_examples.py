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
