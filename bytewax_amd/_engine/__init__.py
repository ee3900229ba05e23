"""The MI355X-native execution engine.

Replaces the reference's Rust/Timely runtime (reference src/worker.rs,
src/run.rs, src/timely.rs) with:

- a *flattening compiler* from the Python operator tree to the nine
  core steps (:mod:`bytewax_amd._engine.compile`);
- an epoch-aligned BSP worker loop (:mod:`bytewax_amd._engine.worker`)
  — sources are polled cooperatively, items propagate through the core
  graph in topological order, keyed exchanges route between workers,
  and every epoch close snapshots state for recovery;
- pluggable exchange transports: in-process (threads, CPU),
  ``torch.distributed`` (gloo for CPU processes, RCCL/xGMI for GPUs).

The Timely progress-tracking frontier is deliberately *not* replicated:
with epoch-aligned scheduling the frontier is a single integer per
cluster, synchronized at epoch boundaries — the natural fit for GPU
batch execution where work is launched per-epoch on HIP streams.
"""

from .worker import cluster_main, run_main  # noqa: F401
