#!/bin/bash
# Round-2 call 40: PMC evidence for the LDS-deduped join
# (write-amplification of the SUM scatter + LDS/wait mix of the merge).
set -x
REPO=/root/repo
export PYTHONPATH=$REPO
export TMPDIR=/tmp
L=$REPO/gpurun_out/r02_call40.log
mkdir -p $REPO/gpurun_out $REPO/gpurun_out/pmc40
: > $L
cd $REPO

PROG=$(cat <<'PYEOF'
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
print("done")
PYEOF
)

echo "=== PMC: TCC write/read reqs (join scatter + merge) ===" >> $L
timeout 420 rocprofv3 --pmc TCC_EA0_WRREQ TCC_EA0_WRREQ_64B TCC_EA0_RDREQ \
  -d $REPO/gpurun_out/pmc40 -o join_tcc -- python -c "$PROG" >> $L 2>&1
echo "rc=$?" >> $L
echo "=== PMC: SQ LDS/wait mix ===" >> $L
timeout 420 rocprofv3 --pmc SQ_INSTS_LDS SQ_INSTS_VALU SQ_WAIT_ANY SQ_BUSY_CYCLES \
  -d $REPO/gpurun_out/pmc40 -o join_sq -- python -c "$PROG" >> $L 2>&1
echo "rc=$?" >> $L
ls $REPO/gpurun_out/pmc40 >> $L
tail -8 $L
