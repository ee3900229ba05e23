#!/bin/bash
# Round-2 call 31: join merge-kernel thread sweep (256/512/1024).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
export TMPDIR=/tmp
L=$REPO/gpurun_out/r02_call31.log
mkdir -p $REPO/gpurun_out $REPO/gpurun_out/prof31
: > $L
cd $REPO

echo "=== join gpu tests (radix now default) ===" >> $L
timeout 400 python -m pytest tests/test_gpu_stats_join.py -m gpu -q >> $L 2>&1
echo "pytest rc=$?" >> $L

echo "=== microbench sweep: coarse 0/1/2 + direct baseline ===" >> $L
for CFG in "BYTEWAX_JOIN_THREADS=256" \
           "BYTEWAX_JOIN_THREADS=512" \
           "BYTEWAX_JOIN_THREADS=1024" \
           "BYTEWAX_JOIN_THREADS=512 AGAIN=1" \
           "BYTEWAX_JOIN_THREADS=1024 AGAIN=1"; do
  echo "--- $CFG ---" >> $L
  env $CFG timeout 300 python - >> $L 2>&1 <<'PYEOF'
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
print(f"join: {2*N*B/dt/1e9:.2f} Ge/s ({dt*1000/B:.2f} ms/round, "
      f"{pairs} pairs)")
PYEOF
done

echo "=== example (defaults, radix now on) ===" >> $L
timeout 300 python examples/stream_join_gpu.py >> $L 2>&1

echo "=== rocprof (best config defaults) ===" >> $L
timeout 300 rocprofv3 --kernel-trace --stats -d $REPO/gpurun_out/prof31 \
  -o join31 -- python - >> $L 2>&1 <<'PYEOF'
import torch
from bytewax_amd.gpu.state import HashJoinState
dev = torch.device("cuda:0")
N, VOCAB = 25_000_000, 1_000_000
g = torch.Generator(device="cuda").manual_seed(5)
k0 = torch.randint(0, VOCAB, (N,), dtype=torch.int32, device=dev, generator=g)
v0 = torch.randint(0, 1 << 30, (N,), dtype=torch.int64, device=dev, generator=g)
k1 = torch.randint(0, VOCAB, (N,), dtype=torch.int32, device=dev, generator=g)
v1 = torch.randint(0, 1 << 30, (N,), dtype=torch.int64, device=dev, generator=g)
st = HashJoinState(dev, slots_pow=21, out_cap=1 << 24)
for _ in range(6):
    st.insert(0, k0, v0)
    st.insert(1, k1, v1)
    st.take_joined()
torch.cuda.synchronize()
print("profiled 6 rounds")
PYEOF
grep -v "simple_timer\|tool.cpp\|generateRocpd" $L | tail -30
