"""Multi-process cluster transport over ``torch.distributed``.

Role parity: the reference's ``CommunicationConfig::Cluster`` TCP mesh
(reference src/run.rs:259-271) — but instead of a bespoke socket mesh,
the control plane and Python-object data plane ride gloo collectives,
and (on GPU workers) columnar RecordBatch exchange inside device
operators rides RCCL over xGMI (see :mod:`bytewax_amd.gpu`).

One process = one worker.  Message payloads are pickled per
destination and exchanged with ``all_to_all_single`` on byte tensors;
the epoch close/EOF/abort vote is a tiny all_reduce + broadcast pair.
"""

import pickle
import time
from datetime import timedelta
from typing import Any, Dict, List, Optional, Tuple

from ..dataflow import Dataflow
from ..inputs import AbortExecution
from ..recovery import RecoveryConfig, RecoveryStore

__all__ = ["dist_cluster_main"]


class _DistCtx:
    """`_WorkerCtx`-compatible transport over a torch.distributed
    process group."""

    def __init__(self, group, rank: int, world: int):
        import torch.distributed as dist

        self.dist = dist
        self.group = group
        self.worker_index = rank
        self.worker_count = world
        self._failure: Optional[BaseException] = None

    def barrier(self) -> None:
        self.dist.barrier(group=self.group)

    def fail(self, ex: BaseException) -> None:
        self._failure = ex

    def exchange_round(
        self, outbox: Dict[int, List[Tuple[int, int, List[Any]]]]
    ) -> List[Tuple[int, int, List[Any]]]:
        import torch

        world = self.worker_count
        blobs = []
        for dst in range(world):
            msgs = outbox.get(dst, [])
            blobs.append(pickle.dumps(msgs) if msgs else b"")
        sizes = torch.tensor([len(b) for b in blobs], dtype=torch.int64)
        recv_sizes = torch.empty_like(sizes)
        self.dist.all_to_all_single(recv_sizes, sizes, group=self.group)
        joined = b"".join(blobs)
        if joined:
            send_buf = torch.frombuffer(bytearray(joined), dtype=torch.uint8)
        else:
            send_buf = torch.empty(0, dtype=torch.uint8)
        out_splits = sizes.tolist()
        in_splits = recv_sizes.tolist()
        recv_buf = torch.empty(int(sum(in_splits)), dtype=torch.uint8)
        self.dist.all_to_all_single(
            recv_buf, send_buf, in_splits, out_splits, group=self.group
        )
        msgs: List[Tuple[int, int, List[Any]]] = []
        raw = recv_buf.numpy().tobytes()
        off = 0
        for sz in in_splits:
            if sz:
                msgs.extend(pickle.loads(raw[off : off + sz]))
            off += sz
        return msgs

    def vote_close(
        self, local_eof: bool, deadline: float, aborting: bool
    ) -> Tuple[bool, bool, bool]:
        import torch

        # One MIN all_reduce carries both votes: [eof, 1 - abort]
        # (abort-any == not min(1 - abort)).
        flags = torch.tensor(
            [1 if local_eof else 0, 0 if aborting else 1], dtype=torch.int64
        )
        self.dist.all_reduce(
            flags, op=self.dist.ReduceOp.MIN, group=self.group
        )
        all_eof = bool(int(flags[0].item()))
        abort = not bool(int(flags[1].item()))
        # Single clock: rank 0 decides closing on wall time.
        decision = torch.tensor(
            [
                1
                if (all_eof or abort or time.monotonic() >= deadline)
                else 0
            ],
            dtype=torch.int64,
        )
        self.dist.broadcast(decision, src=0, group=self.group)
        return (bool(int(decision.item())), all_eof, abort)


def dist_cluster_main(
    flow: Dataflow,
    addresses: List[str],
    proc_id: int,
    *,
    epoch_interval: Optional[timedelta] = None,
    recovery_config: Optional[RecoveryConfig] = None,
    worker_count_per_proc: int = 1,
) -> None:
    """Run this process's worker of a multi-process cluster.

    All processes must share the recovery directory filesystem if
    recovery is enabled.
    """
    import torch.distributed as dist

    from .worker import (
        EPOCH_INTERVAL_DEFAULT,
        _Interrupted,
        _load_resume_snaps,
        _Worker,
    )
    from .compile import compile_graph

    if worker_count_per_proc != 1:
        msg = (
            "the torch.distributed transport runs exactly one worker per "
            "process; scale with more processes"
        )
        raise ValueError(msg)
    if epoch_interval is None:
        epoch_interval = EPOCH_INTERVAL_DEFAULT

    world = len(addresses)
    owns_group = not (dist.is_available() and dist.is_initialized())
    if owns_group:
        master = addresses[0]
        dist.init_process_group(
            backend="gloo",
            init_method=f"tcp://{master}",
            rank=proc_id,
            world_size=world,
            timeout=timedelta(seconds=120),
        )
    ctx = _DistCtx(None, proc_id, world)

    graph = compile_graph(flow)
    store = None
    ex_num, resume_epoch = 0, 1
    if recovery_config is not None:
        interval = (
            epoch_interval
            if epoch_interval > timedelta(0)
            else timedelta(microseconds=1)
        )
        store = RecoveryStore(recovery_config, interval)
        # All ranks must compute resume_from before anyone writes the
        # new execution row.
        ex_num, resume_epoch = store.resume_from()
        ctx.barrier()
        if proc_id == 0:
            store.write_ex(ex_num, world, resume_epoch)
        ctx.barrier()
    resume_snaps = _load_resume_snaps(store, resume_epoch)

    worker = _Worker(
        graph, ctx, store, ex_num, resume_epoch, epoch_interval, resume_snaps
    )
    try:
        worker.run()
    except _Interrupted:
        pass
    finally:
        if store is not None:
            store.close()
        if owns_group:
            dist.destroy_process_group()
    if ctx._failure is not None and not isinstance(
        ctx._failure, AbortExecution
    ):
        raise ctx._failure
