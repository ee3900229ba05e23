#!/bin/bash
# Round-2 GPU call 9: post-revert validation + long soaks + string
# dict throughput for the record.
set -x
REPO=/root/repo
L=$REPO/gpurun_out/r02_call9.log
mkdir -p $REPO/gpurun_out
: > $L
cd $REPO

echo "=== pytest -m gpu (full) ===" >> $L
timeout 1500 python -m pytest tests -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== smoke ===" >> $L
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $L 2>&1

echo "=== headline soak: dataflow defaults (20x200 batches) ===" >> $L
timeout 420 python bench.py --steps 20 --warmup 5 >> $L 2>&1
echo "=== native soak: 2000 batches ===" >> $L
timeout 420 python bench.py --engine native --steps 200 --warmup 5 --batches-per-poll 10 --latency-probes 20 >> $L 2>&1

echo "=== string dict encode throughput (device kernels, packed input) ===" >> $L
timeout 300 python - >> $L 2>&1 <<'PYEOF'
import time, torch
from bytewax_amd.gpu.strings import StringDict, pack_strings
vocab = [f"key-{i:07d}" for i in range(1_000_000)]
import random
rng = random.Random(5)
strings = [vocab[rng.randrange(len(vocab))] for _ in range(4_000_000)]
t0 = time.perf_counter()
data, offs = pack_strings(strings)
t_pack = time.perf_counter() - t0
d = StringDict(torch.device("cuda:0"), slots_pow=22)
ids = d.encode((data, offs))  # cold: creates 1M ids
torch.cuda.synchronize()
t0 = time.perf_counter()
reps = 5
for _ in range(reps):
    ids = d.encode((data, offs))
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / reps
n = len(strings)
print(f"host pack: {n/t_pack/1e6:.1f} M strings/s; device encode "
      f"(H2D + hash + dict lookup): {n/dt/1e6:.1f} M strings/s "
      f"({dt*1000:.1f} ms / 4M strings, 1M distinct)")
PYEOF
tail -4 $L
