"""Direct unit tests of the exchange wire format on 2 gloo ranks:
the ts_base fold and the lazy wire batch must reconstruct exact
absolute timestamps regardless of producer base."""

import os
import subprocess
import sys
import textwrap
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

_PROG = """
import os

import torch
import torch.distributed as dist

from bytewax_amd.gpu import RecordBatch, exchange_by_key

dist.init_process_group("gloo")
rank = dist.get_rank()

torch.manual_seed(100 + rank)
n = 5000
keys = torch.randint(0, 1000, (n,), dtype=torch.int32)
base = 1_000_000 * (rank + 1)
ts0 = torch.randint(0, 10_000, (n,), dtype=torch.int64)

# Path A: zero-based template with ts_base.
a = exchange_by_key(
    RecordBatch(keys, ts0, None, max_ts=base + 10_000, ts_base=base)
)
# Path B: identical data with absolute timestamps.
b = exchange_by_key(
    RecordBatch(keys, ts0 + base, None, max_ts=base + 10_000)
)
ra = sorted(zip(a.keys.tolist(), a.ts.tolist()))
rb = sorted(zip(b.keys.tolist(), b.ts.tolist()))
assert ra == rb, "ts_base fold mismatch"
# Every received key must hash-route to this rank.
from bytewax_amd.gpu import _mix64_torch
owners = torch.remainder(_mix64_torch(a.keys.to(torch.int64)), 2)
assert bool((owners == rank).all()), "routing violated"
t = torch.tensor([len(ra)], dtype=torch.int64)
dist.all_reduce(t)
assert int(t.item()) == 2 * n  # nothing lost or duplicated
if rank == 0:
    print("EXCHANGE OK")
dist.destroy_process_group()
"""


@pytest.mark.timeout(180)
def test_exchange_ts_base_fold_two_ranks(tmp_path: Path):
    prog = tmp_path / "ex.py"
    prog.write_text(textwrap.dedent(_PROG))
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(29820 + os.getpid() % 60)
    procs = []
    for rank in range(2):
        e = dict(env, RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK=str(rank))
        procs.append(
            subprocess.Popen(
                [sys.executable, str(prog)],
                env=e,
                stdout=subprocess.PIPE,
                stderr=subprocess.PIPE,
            )
        )
    outs = [p.communicate(timeout=150) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se.decode()[-1500:]
    assert "EXCHANGE OK" in outs[0][0].decode()
