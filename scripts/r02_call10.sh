#!/bin/bash
# Round-2 GPU call 10: BASELINE config-3 shape (2-key dedup), session
# + 1BRC PMC for the record, str-keyed end-to-end number.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call10.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== config-3 shape: 2-key benchmark_windowing, dedup path ===" >> $L
timeout 240 python bench.py --engine native --steps 20 --warmup 5 --batches-per-poll 10 --vocab 2 --dedup --events-per-batch 32000000 --latency-probes 0 >> $L 2>&1
echo "=== same on the dataflow engine ===" >> $L
timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 20 --vocab 2 --dedup --events-per-batch 32000000 >> $L 2>&1

echo "=== str-keyed wordcount end-to-end (packed columns) ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import time, torch
from datetime import datetime, timedelta, timezone
import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.gpu.operators import keyed_window_agg_str
from bytewax_amd.gpu.strings import pack_strings
from bytewax_amd.inputs import DynamicSource, StatelessSourcePartition
from bytewax_amd.testing import TestingSink, run_main

ALIGN = datetime(2024, 1, 1, tzinfo=timezone.utc)
ALIGN_MS = int(ALIGN.timestamp() * 1000)
import random
rng = random.Random(7)
vocab = [f"word-{i}" for i in range(200_000)]
N, B = 4_000_000, 10
packed = pack_strings([vocab[rng.randrange(len(vocab))] for _ in range(N)])

import numpy as np
ts_tmpl = (np.arange(N, dtype=np.int64) % 1000)

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
dt = time.perf_counter() - t0
total = sum(v for _k, _w, v in out)
print(f"str-keyed wordcount: {N*B} events in {dt:.2f}s = "
      f"{N*B/dt/1e9:.2f}e9 events/s; counted {total} events over "
      f"{len(set(k for k,_w,_v in out))} distinct words")
PYEOF

echo "=== PMC: session + stats kernels ===" >> $L
export TMPDIR=/tmp; cd /tmp
timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_sess -o sess -- \
  python $REPO/examples/sessions_gpu.py >> $L 2>&1
timeout 600 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof_brc -o brc -- \
  python $REPO/examples/onebrc_gpu.py >> $L 2>&1
cd $REPO
tail -4 $L
