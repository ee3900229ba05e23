#!/bin/bash
# Round-2 call 32: full gpu suite + join defaults confirm (1024-thread
# merge kernel, coarse 1, radix on) + str pack kernel test.
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
L=$REPO/gpurun_out/r02_call32.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1000 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== join microbench (defaults) ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import time, torch
from bytewax_amd.gpu.state import HashJoinState
dev = torch.device("cuda:0")
N, B, VOCAB = 25_000_000, 20, 1_000_000
g = torch.Generator(device="cuda").manual_seed(5)
sides = []
for s in (0, 1):
    ks, vs = [], []
    for b in range(4):
        ks.append(torch.randint(0, VOCAB, (N,), dtype=torch.int32,
                                device=dev, generator=g))
        vs.append(torch.randint(0, 1 << 30, (N,), dtype=torch.int64,
                                device=dev, generator=g))
    sides.append((ks, vs))
st = HashJoinState(dev, slots_pow=21, out_cap=1 << 24)
st.insert(0, sides[0][0][0], sides[0][1][0])
st.insert(1, sides[1][0][0], sides[1][1][0])
st.take_joined()
torch.cuda.synchronize()
t0 = time.perf_counter()
pairs = 0
for b in range(B):
    st.insert(0, sides[0][0][b % 4], sides[0][1][b % 4])
    st.insert(1, sides[1][0][b % 4], sides[1][1][b % 4])
    out = st.take_joined()
    if out is not None:
        pairs += out[0].numel()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"join defaults: {2*N*B/dt/1e9:.2f} Ge/s ({pairs} pairs)")
PYEOF

echo "=== stream join example (defaults) ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1
echo "=== str-keyed wordcount warm (regression check) ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import time, random
import numpy as np, torch
from datetime import datetime, timedelta, timezone
import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.gpu.operators import keyed_window_agg_str
from bytewax_amd.gpu.strings import pack_strings
from bytewax_amd.inputs import DynamicSource, StatelessSourcePartition
from bytewax_amd.testing import TestingSink, run_main
ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ALIGN_MS = int(ALIGN.timestamp() * 1000)
rng = random.Random(7)
vocab = [f"word-{i}" for i in range(200_000)]
N, B = 4_000_000, 10
packed = pack_strings([vocab[rng.randrange(len(vocab))] for _ in range(N)])
ts_tmpl = (np.arange(N, dtype=np.int64) % 1000)
def run():
    class Part(StatelessSourcePartition):
        def __init__(self): self.i = 0
        def next_batch(self):
            if self.i >= B: raise StopIteration()
            ts = ts_tmpl + (ALIGN_MS + self.i * 1000)
            self.i += 1
            return [(packed, ts)]
    class Src(DynamicSource):
        def build(self, *_a): return Part()
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
    return time.perf_counter() - t0
run()
for tag in ("warm1", "warm2"):
    dt = run()
    print(f"{tag}: {N*B/dt/1e6:.0f} M events/s")
PYEOF
tail -25 $L
