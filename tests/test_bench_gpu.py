"""bench.py contract smoke on a real GPU (the driver's entry point)."""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


@pytest.mark.parametrize("engine", ["native", "python"])
def test_bench_single_gpu_json_contract(engine):
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    res = subprocess.run(
        [
            sys.executable,
            "bench.py",
            "--engine",
            engine,
            "--steps",
            "12",
            "--warmup",
            "3",
            "--events-per-batch",
            "4000000",
            "--batches-per-poll",
            "4",
            "--vocab",
            "20000",
        ],
        capture_output=True,
        timeout=420,
        cwd=str(REPO),
        env=env,
    )
    assert res.returncode == 0, res.stderr.decode()[-1500:]
    line = res.stdout.decode().strip().splitlines()[-1]
    d = json.loads(line)
    assert d["n_gpus"] == 1
    assert d["steps"] == 12
    assert d["higher_is_better"] is True
    assert d["value"] > 0
    assert d["ms_per_step"] > 0
    assert d["data"] == "synthetic"
    got = d["config"]["engine"]
    if engine == "python":
        assert got.startswith("dataflow")
    else:
        assert got == engine
