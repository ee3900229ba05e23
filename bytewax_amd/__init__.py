"""bytewax_amd: an MI355X-native stateful stream-processing framework.

A from-scratch AMD-GPU-first framework with the capabilities and Python
API of Bytewax: a `Dataflow` operator-graph builder, stateless and
keyed-stateful operators, event/system-time windowing, joins,
partitioned sources/sinks, and epoch-coordinated recovery snapshots —
executed by an epoch-aligned BSP engine whose keyed hot path runs as
hand-written CDNA4 HIP kernels over columnar record batches, with
RCCL-over-xGMI key exchange across workers-as-GPUs.

Start with :mod:`bytewax_amd.dataflow` and
:mod:`bytewax_amd.operators`.
"""

__version__ = "0.1.0"
