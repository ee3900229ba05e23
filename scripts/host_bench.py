import time, sys
import bytewax_amd.operators as op
from bytewax_amd.dataflow import Dataflow
from bytewax_amd.testing import TestingSink, TestingSource, run_main
from datetime import timedelta
N = 200_000
words = [f"w{i%1000}" for i in range(N)]
best = 0
for rep in range(7):
    out = []
    flow = Dataflow("wc")
    s = op.input("inp", flow, TestingSource(words, batch_size=500))
    counts = op.count_final("count", s, lambda w: w)
    op.output("out", counts, TestingSink(out))
    t0 = time.perf_counter()
    run_main(flow, epoch_interval=timedelta(days=365))
    dt = time.perf_counter() - t0
    best = max(best, N/dt/1e6)
print(f"best {best:.2f} M events/s")
