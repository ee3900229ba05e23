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


@pytest.mark.timeout(240)
def test_sliding_window_exchange_2rank_gloo(tmp_path):
    """Sliding windows + the wire-format exchange at world 2 (gloo
    CPU twins): per-key sums across ranks must equal the single-rank
    brute force (the scatter expands events into overlapped windows
    AFTER the int32-segment exchange)."""
    import json
    import subprocess
    import sys
    import textwrap
    from pathlib import Path

    repo = Path(__file__).resolve().parent.parent
    script = tmp_path / "slide2.py"
    script.write_text(
        textwrap.dedent(
            """
            import os, json, torch
            import torch.distributed as dist
            from datetime import datetime, timedelta, timezone
            import bytewax_amd.operators as op
            from bytewax_amd.dataflow import Dataflow
            from bytewax_amd.gpu import RecordBatch
            from bytewax_amd.gpu.operators import (
                CollectCountsSink, keyed_window_agg,
            )
            from bytewax_amd.inputs import (
                DynamicSource, StatelessSourcePartition,
            )
            from bytewax_amd.testing import run_main

            dist.init_process_group("gloo")
            rank = dist.get_rank()
            ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
            ALIGN_MS = int(ALIGN.timestamp() * 1000)

            class Part(StatelessSourcePartition):
                def __init__(self):
                    self.i = 0
                def next_batch(self):
                    if self.i >= 4:
                        raise StopIteration()
                    g = torch.Generator().manual_seed(100 * rank + self.i)
                    n = 500
                    keys = torch.randint(0, 50, (n,), dtype=torch.int32,
                                         generator=g)
                    ts = torch.randint(0, 5000, (n,), dtype=torch.int32,
                                       generator=g)
                    self.i += 1
                    return [RecordBatch(
                        keys, ts, keys.to(torch.int64),
                        max_ts=ALIGN_MS + self.i * 5000 - 1,
                        ts_base=ALIGN_MS + (self.i - 1) * 5000,
                    )]

            class Src(DynamicSource):
                def build(self, *_a):
                    return Part()

            out = []
            flow = Dataflow("slide2")
            s = op.input("inp", flow, Src())
            agg = keyed_window_agg(
                "agg", s, align_to=ALIGN,
                length=timedelta(seconds=6),
                offset=timedelta(seconds=2),
                mode="sum", device="cpu", exchange=True,
            )
            op.output("out", agg, CollectCountsSink(out))
            run_main(flow, epoch_interval=timedelta(days=1))
            rows = {}
            for b in out:
                for k, t, v in zip(b.keys.tolist(), b.ts.tolist(),
                                   b.vals.tolist()):
                    rows[(k, t)] = rows.get((k, t), 0) + v
            gathered = [None, None]
            dist.all_gather_object(gathered, rows)
            if rank == 0:
                merged = {}
                for g2 in gathered:
                    for k, v in g2.items():
                        merged[k] = merged.get(k, 0) + v
                # Brute force over all ranks' events.
                expect = {}
                for r in range(2):
                    for i in range(4):
                        g = torch.Generator().manual_seed(100 * r + i)
                        n = 500
                        keys = torch.randint(0, 50, (n,),
                                             dtype=torch.int32, generator=g)
                        ts = torch.randint(0, 5000, (n,),
                                           dtype=torch.int32, generator=g)
                        base = ALIGN_MS + i * 5000
                        for k, t in zip(keys.tolist(), ts.tolist()):
                            at = base + t
                            hi = (at - ALIGN_MS) // 2000
                            lo = (at - ALIGN_MS - 6000) // 2000 + 1
                            for w in range(lo, hi + 1):
                                key = (k, ALIGN_MS + w * 2000)
                                expect[key] = expect.get(key, 0) + k
                assert merged == expect, "sliding exchange mismatch"
                print("SLIDE2_OK")
            dist.destroy_process_group()
            """
        )
    )
    env = dict(os.environ)
    env["PYTHONPATH"] = str(repo)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(29870 + os.getpid() % 100)
    procs = []
    for rank in range(2):
        e = dict(env)
        e.update(RANK=str(rank), WORLD_SIZE="2", LOCAL_RANK=str(rank))
        procs.append(
            subprocess.Popen(
                [sys.executable, str(script)],
                env=e, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
            )
        )
    outs = [p.communicate(timeout=200) for p in procs]
    for p, (so, se) in zip(procs, outs):
        assert p.returncode == 0, se.decode()[-2000:]
    assert "SLIDE2_OK" in outs[0][0].decode()


_STR_PROG = """
import os

import torch
import torch.distributed as dist

from bytewax_amd.gpu.strings import (
    exchange_str_by_key,
    pack_strings,
    str_owner_cpu,
)

dist.init_process_group("gloo")
rank = dist.get_rank()

import random

rng = random.Random(31 + rank)
words = [f"word-{rng.randrange(400)}" for _ in range(3000)]
data, offs = pack_strings(words)
ts = torch.arange(3000, dtype=torch.int64) + rank * 1_000_000
vals = torch.arange(3000, dtype=torch.int64) * (rank + 1)

rb, ro, rt, rv = exchange_str_by_key(
    data, offs, ts, vals
)
got = []
rb_np = rb.numpy()
ro_np = ro.numpy()
for i in range(len(ro_np) - 1):
    s = bytes(rb_np[ro_np[i] : ro_np[i + 1]]).decode()
    got.append((s, int(rt[i]), int(rv[i])))
# Routing: every received string owned by this rank.
for s, _t, _v in got:
    assert str_owner_cpu(s.encode(), 2) == rank, s
# Conservation: the union of received triples == union of sent.
sent = [
    (w, int(t), int(v)) for w, t, v in zip(words, ts.tolist(), vals.tolist())
]
gathered = [None, None]
dist.all_gather_object(gathered, (sent, got))
if rank == 0:
    all_sent = sorted(gathered[0][0] + gathered[1][0])
    all_got = sorted(gathered[0][1] + gathered[1][1])
    assert all_sent == all_got, (len(all_sent), len(all_got))
    print("STR EXCHANGE OK")
dist.destroy_process_group()
"""


@pytest.mark.timeout(180)
def test_str_exchange_two_ranks_gloo(tmp_path: Path):
    """String-key exchange at world 2 (CPU twin): raw bytes routed by
    content hash, events conserved, every string wholly owned by one
    rank (the multi-GPU str-keyed design in gpu/strings.py)."""
    prog = tmp_path / "strex.py"
    prog.write_text(textwrap.dedent(_STR_PROG))
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(29700 + os.getpid() % 60)
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
    assert "STR EXCHANGE OK" in outs[0][0].decode()


_STR_E2E_PROG = """
import os
import random

import torch
import torch.distributed as dist

from bytewax_amd.gpu import AGG_COUNT, WindowAggState
from bytewax_amd.gpu.operators import _StrWindowLogic
from bytewax_amd.gpu.strings import StringDict

dist.init_process_group("gloo")
rank = dist.get_rank()

dev = torch.device("cpu")
sdict = StringDict(dev)
state = WindowAggState(
    dev, 0, 60_000, AGG_COUNT, slots_pow=16, out_cap=1 << 16, radix=False
)
logic = _StrWindowLogic(sdict, state, 0, None, exchange=True)

rng = random.Random(7 + rank)
B, N = 4, 800
sent = []
out = []
for b in range(B):
    words = [f"user-{rng.randrange(50)}" for _ in range(N)]
    ts = [b * 30_000 + i % 1000 for i in range(N)]
    sent.extend(zip(words, ts))
    rows, _ = logic.on_batch([(words, ts)])
    out.extend(rows)
rows, _ = logic.on_eof()
out.extend(rows)

gathered = [None, None]
dist.all_gather_object(gathered, (sent, out))
if rank == 0:
    from collections import Counter

    truth = Counter()
    for sent_r, _o in gathered:
        for w, t in sent_r:
            truth[(w, (t // 60_000) * 60_000)] += 1
    got = Counter()
    for _s, out_r in gathered:
        for key, win_ms, v in out_r:
            got[(key, win_ms)] += v
    assert got == truth, (len(got), len(truth))
    print("STR E2E OK")
dist.destroy_process_group()
"""


@pytest.mark.timeout(180)
def test_str_windowing_exchange_two_ranks_gloo(tmp_path: Path):
    """keyed_window_agg_str's exchange path at world 2 (CPU twins):
    windowed counts over the union of both ranks' str events equal
    the brute force, with each string counted at its owning rank."""
    prog = tmp_path / "stre2e.py"
    prog.write_text(textwrap.dedent(_STR_E2E_PROG))
    env = dict(os.environ)
    env["PYTHONPATH"] = str(REPO)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = str(29640 + os.getpid() % 60)
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
    assert "STR E2E OK" in outs[0][0].decode()
