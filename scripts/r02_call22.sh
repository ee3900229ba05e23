#!/bin/bash
# Round-2 call 22: extra coverage datapoints.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call22.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO
echo "=== 1BRC 100k stations ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import subprocess, sys, re
# onebrc example has N_STATIONS const; run via env-free variant inline
PYEOF
sed 's/N_STATIONS = 10_000/N_STATIONS = 100_000/' examples/onebrc_gpu.py > /tmp/onebrc_100k.py
timeout 300 python /tmp/onebrc_100k.py >> $L 2>&1
echo "=== join 1B events ===" >> $L
sed 's/events = 10_000_000/events = 25_000_000/; s/n_batches = 20/n_batches = 20/' examples/stream_join_gpu.py > /tmp/join_big.py
timeout 300 python /tmp/join_big.py >> $L 2>&1
echo "=== sliding soak (60s/20s, 200 batches x 32M) ===" >> $L
timeout 600 python - >> $L 2>&1 <<'PYEOF'
import time, torch
from bytewax_amd.gpu import RecordBatch, WindowAggState, AGG_COUNT, _ms
from datetime import datetime, timezone
align = _ms(datetime(2024, 1, 1, tzinfo=timezone.utc))
dev = torch.device("cuda:0")
n = 32_000_000
g = torch.Generator(device="cuda").manual_seed(3)
keys = [torch.randint(0, 1_000_000, (n,), dtype=torch.int32, generator=g, device=dev) for _ in range(4)]
tmpl = ((torch.arange(n, dtype=torch.int64, device=dev) * 5000) // n).to(torch.int32)
st = WindowAggState(dev, align, 60_000, AGG_COUNT, slots_pow=24, out_cap=1 << 24,
                    radix=True, off_ms=20_000, max_batch=n)
rows = 0
def run(k0, kN):
    global rows
    for i in range(k0, kN):
        st.insert(RecordBatch(keys[i % 4], tmpl, max_ts=align + (i + 1) * 5000 - 1, ts_base=align + i * 5000))
        out = st.close_due()
        if out is not None:
            rows += len(out)
run(0, 5)
torch.cuda.synchronize()
t0 = time.perf_counter()
run(5, 205)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"sliding soak: 200x32M = {200*n/1e9:.1f}e9 events at {200*n/dt/1e9:.1f}e9 events/s; closed rows {rows}")
PYEOF
echo "=== headline rep ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
tail -4 $L
