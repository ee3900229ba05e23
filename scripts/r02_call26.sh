#!/bin/bash
# Round-2 call 26: pinned-staging str path — measure + validate.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call26.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== str end-to-end (pinned staging) ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import time, random
import numpy as np, torch
from datetime import datetime, timedelta, timezone
import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.gpu.operators import keyed_window_agg_str
from bytewax_amd.gpu.strings import StringDict, pack_strings
from bytewax_amd.inputs import DynamicSource, StatelessSourcePartition
from bytewax_amd.testing import TestingSink, run_main

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ALIGN_MS = int(ALIGN.timestamp() * 1000)
rng = random.Random(7)
vocab = [f"word-{i}" for i in range(200_000)]
N, B = 4_000_000, 10
packed = pack_strings([vocab[rng.randrange(len(vocab))] for _ in range(N)])
ts_tmpl = (np.arange(N, dtype=np.int64) % 1000)

# 1) raw dict encode rate (packed input, warm)
d = StringDict(torch.device("cuda"), slots_pow=20)
for _ in range(3):
    d.encode(packed)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    d.encode(packed)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"dict encode (pinned): {N*10/dt/1e9:.2f} G strings/s")

def run():
    class Part(StatelessSourcePartition):
        def __init__(self):
            self.i = 0
        def next_batch(self):
            if self.i >= B:
                raise StopIteration()
            ts = ts_tmpl + (ALIGN_MS + self.i * 1000)
            self.i += 1
            return [(packed, ts)]
    class Src(DynamicSource):
        def build(self, *_a):
            return Part()
    out = []
    flow = Dataflow("strwc")
    s = op.input("inp", flow, Src())
    agg = keyed_window_agg_str("agg", s, align_to=ALIGN,
                               length=timedelta(seconds=60),
                               dict_slots_pow=20, device="cuda")
    op.output("out", agg, TestingSink(out))
    t0 = time.perf_counter()
    run_main(flow, epoch_interval=timedelta(days=365))
    torch.cuda.synchronize()
    return time.perf_counter() - t0, out

dt, out = run()  # cold
print(f"cold run: {N*B/dt/1e6:.0f} M events/s")
for tag in ("warm1", "warm2"):
    dt, out = run()
    print(f"{tag} run: {N*B/dt/1e6:.0f} M events/s")
rows = sum(len(v) if isinstance(v, list) else 1 for v in out)
print(f"out windows: {len(out)}")
PYEOF

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L
tail -40 $L
