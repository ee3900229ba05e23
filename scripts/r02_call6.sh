#!/bin/bash
# Round-2 GPU call 6: sliding-radix validation + perf, regression
# sweep of the headline paths after the off_ms threading.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call6.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== headline confirm ===" >> $L
timeout 240 python bench.py --steps 10 --warmup 3 --batches-per-poll 20 >> $L 2>&1

echo "=== sliding-window bench (radix expansion, 60s len / 20s off) ===" >> $L
timeout 240 python - >> $L 2>&1 <<'PYEOF'
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
for i in range(3):
    st.insert(RecordBatch(keys[i % 4], tmpl, max_ts=align + (i + 1) * 5000 - 1, ts_base=align + i * 5000))
    st.close_due()
torch.cuda.synchronize()
t0 = time.perf_counter()
K = 40
for i in range(3, 3 + K):
    st.insert(RecordBatch(keys[i % 4], tmpl, max_ts=align + (i + 1) * 5000 - 1, ts_base=align + i * 5000))
    st.close_due()
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(f"sliding radix (3x expansion): {K * n / dt / 1e9:.1f}e9 events/s ({dt / K * 1000:.2f} ms/batch)")
PYEOF

echo "=== 1BRC + sessions confirm ===" >> $L
timeout 300 python examples/onebrc_gpu.py >> $L 2>&1
timeout 300 python examples/sessions_gpu.py >> $L 2>&1
tail -4 $L
