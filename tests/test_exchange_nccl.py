"""RCCL-backed exchange on real hardware.

Runs the `exchange_by_key` collective path over the actual NCCL/RCCL
backend with 2 ranks mapped to one physical GPU — functional
validation of the multi-GPU wire format (bucketing kernel, split-size
exchange, int32 timestamp compression) without an 8-GPU node.
"""

import os
import subprocess
import sys
from pathlib import Path

import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent


def test_exchange_2rank_rccl_one_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    res = subprocess.run(
        [sys.executable, str(REPO / "scripts" / "probe_nccl_2rank.py"), "2"],
        capture_output=True,
        timeout=300,
        cwd=str(REPO),
        env=env,
    )
    out = res.stdout.decode()
    assert res.returncode == 0, (out + res.stderr.decode())[-2000:]
    assert "PROBE_2RANK_RESULT: PASS" in out
