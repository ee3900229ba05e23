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


def test_str_exchange_1rank_rccl_self_roundtrip():
    """The string-byte exchange over an actual NCCL(RCCL) group:
    1-rank force self-exchange — device pack kernels + the four
    all-to-allv collectives + offset rebuild return the same string
    multiset with its ts/vals attached."""
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    prog = r"""
import os, random, sys
os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29733")
import torch
import torch.distributed as dist
from bytewax_amd.gpu.strings import exchange_str_by_key, pack_strings

dist.init_process_group("nccl", rank=0, world_size=1)
rng = random.Random(3)
words = [f"word-{rng.randrange(2000)}" for _ in range(50_000)]
data, offs = pack_strings(words)
dev = torch.device("cuda:0")
d = torch.from_numpy(data).to(dev)
o = torch.from_numpy(offs).to(dev)
ts = torch.arange(len(words), dtype=torch.int64, device=dev)
vals = ts * 7
rb, ro, rt, rv = exchange_str_by_key(d, o, ts, vals, force=True)
rb_h, ro_h = rb.cpu().numpy(), ro.cpu().numpy()
got = sorted(
    (bytes(rb_h[ro_h[i]:ro_h[i+1]]).decode(), int(t), int(v))
    for i, (t, v) in enumerate(zip(rt.cpu().tolist(), rv.cpu().tolist()))
)
sent = sorted((w, i, i * 7) for i, w in enumerate(words))
assert got == sent, (len(got), len(sent))
print("STR_NCCL_SELF OK", flush=True)
dist.destroy_process_group()
"""
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    res = subprocess.run(
        [sys.executable, "-c", prog],
        capture_output=True,
        timeout=300,
        cwd=str(REPO),
        env=env,
    )
    out = res.stdout.decode()
    assert res.returncode == 0, (out + res.stderr.decode())[-2000:]
    assert "STR_NCCL_SELF OK" in out
