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


class _HybridShared:
    """Per-process shared state for W worker THREADS x P processes
    (reference `Cluster{threads, process, addresses}`,
    src/run.rs:259-271).

    Thread 0 of each process is the gloo LEADER: inter-process
    exchanges and votes funnel through it (one collective per round
    per process — gloo groups are not thread-collective-safe), fenced
    by the local thread barrier on both sides.
    """

    def __init__(self, dist, local_n: int, proc_id: int, nprocs: int):
        import threading

        self.dist = dist
        self.local_n = local_n
        self.proc_id = proc_id
        self.nprocs = nprocs
        self.world = local_n * nprocs
        self.barrier = threading.Barrier(local_n)
        self.lock = threading.Lock()
        # Indexed by GLOBAL worker; only this process's slots are read.
        self.inboxes: List[List[Tuple[int, int, List[Any]]]] = [
            [] for _ in range(self.world)
        ]
        # Cross-process staging: dst global worker -> msgs.
        self.cross: Dict[int, List[Tuple[int, int, List[Any]]]] = {}
        self.eof_votes = [False] * local_n
        self.abort_votes = [False] * local_n
        self.close = False
        self.all_eof = False
        self.abort_res = False
        self.failure: Optional[BaseException] = None

    def owner_proc(self, worker: int) -> int:
        return worker // self.local_n


class _HybridCtx:
    """`_WorkerCtx`-compatible context for one worker thread of a
    threads-x-processes cluster."""

    def __init__(self, shared: _HybridShared, thread_idx: int):
        self.s = shared
        self.thread_idx = thread_idx
        self.worker_index = shared.proc_id * shared.local_n + thread_idx
        self.worker_count = shared.world

    def _tbarrier(self) -> None:
        from .worker import _Interrupted

        import threading

        try:
            self.s.barrier.wait()
        except threading.BrokenBarrierError:
            raise _Interrupted() from None

    def barrier(self) -> None:
        # Global: local threads align, the leader aligns processes.
        self._tbarrier()
        if self.thread_idx == 0:
            self.s.dist.barrier()
        self._tbarrier()

    def fail(self, ex: BaseException) -> None:
        with self.s.lock:
            if self.s.failure is None:
                self.s.failure = ex

    def exchange_round(
        self, outbox: Dict[int, List[Tuple[int, int, List[Any]]]]
    ) -> List[Tuple[int, int, List[Any]]]:
        import torch

        s = self.s
        with s.lock:
            for dst, msgs in outbox.items():
                if s.owner_proc(dst) == s.proc_id:
                    s.inboxes[dst].extend(msgs)
                else:
                    s.cross.setdefault(dst, []).extend(msgs)
        self._tbarrier()
        if self.thread_idx == 0:
            # One gloo round at process granularity: ship
            # {dst_worker: msgs} dicts per destination process.
            per_proc: List[Dict[int, List]] = [
                {} for _ in range(s.nprocs)
            ]
            for dst, msgs in s.cross.items():
                per_proc[s.owner_proc(dst)][dst] = msgs
            s.cross = {}
            blobs = [
                pickle.dumps(d) if d else b"" for d in per_proc
            ]
            sizes = torch.tensor(
                [len(b) for b in blobs], dtype=torch.int64
            )
            recv_sizes = torch.empty_like(sizes)
            s.dist.all_to_all_single(recv_sizes, sizes)
            joined = b"".join(blobs)
            send_buf = (
                torch.frombuffer(bytearray(joined), dtype=torch.uint8)
                if joined
                else torch.empty(0, dtype=torch.uint8)
            )
            out_splits = sizes.tolist()
            in_splits = recv_sizes.tolist()
            recv_buf = torch.empty(
                int(sum(in_splits)), dtype=torch.uint8
            )
            s.dist.all_to_all_single(
                recv_buf, send_buf, in_splits, out_splits
            )
            raw = recv_buf.numpy().tobytes()
            off = 0
            for sz in in_splits:
                if sz:
                    for dst, msgs in pickle.loads(
                        raw[off : off + sz]
                    ).items():
                        s.inboxes[dst].extend(msgs)
                off += sz
        self._tbarrier()
        with s.lock:
            msgs = s.inboxes[self.worker_index]
            s.inboxes[self.worker_index] = []
        return msgs

    def vote_close(
        self, local_eof: bool, deadline: float, aborting: bool
    ) -> Tuple[bool, bool, bool]:
        import torch

        s = self.s
        s.eof_votes[self.thread_idx] = local_eof
        s.abort_votes[self.thread_idx] = aborting
        self._tbarrier()
        if self.thread_idx == 0:
            flags = torch.tensor(
                [
                    1 if all(s.eof_votes) else 0,
                    0 if any(s.abort_votes) else 1,
                ],
                dtype=torch.int64,
            )
            s.dist.all_reduce(flags, op=s.dist.ReduceOp.MIN)
            s.all_eof = bool(int(flags[0].item()))
            s.abort_res = not bool(int(flags[1].item()))
            decision = torch.tensor(
                [
                    1
                    if (
                        s.all_eof
                        or s.abort_res
                        or time.monotonic() >= deadline
                    )
                    else 0
                ],
                dtype=torch.int64,
            )
            s.dist.broadcast(decision, src=0)
            s.close = bool(int(decision.item()))
        self._tbarrier()
        return (s.close, s.all_eof, s.abort_res)


def dist_cluster_main(
    flow: Dataflow,
    addresses: List[str],
    proc_id: int,
    *,
    epoch_interval: Optional[timedelta] = None,
    recovery_config: Optional[RecoveryConfig] = None,
    worker_count_per_proc: int = 1,
) -> None:
    """Run this process's worker(s) of a multi-process cluster.

    With ``worker_count_per_proc > 1`` each process runs that many
    worker THREADS (reference `Cluster{threads, process, addresses}`,
    src/run.rs:259-271); inter-process traffic funnels through thread
    0's gloo collectives.  All processes must share the recovery
    directory filesystem if recovery is enabled.
    """
    import threading

    import torch.distributed as dist

    from .worker import (
        EPOCH_INTERVAL_DEFAULT,
        _Interrupted,
        _load_resume_snaps,
        _Worker,
    )
    from .compile import compile_graph

    if worker_count_per_proc < 1:
        msg = "worker_count_per_proc must be >= 1"
        raise ValueError(msg)
    if epoch_interval is None:
        epoch_interval = EPOCH_INTERVAL_DEFAULT

    nprocs = len(addresses)
    local_n = worker_count_per_proc
    world = nprocs * local_n
    owns_group = not (dist.is_available() and dist.is_initialized())
    if owns_group:
        master = addresses[0]
        dist.init_process_group(
            backend="gloo",
            init_method=f"tcp://{master}",
            rank=proc_id,
            world_size=nprocs,
            timeout=timedelta(seconds=120),
        )

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
        dist.barrier()
        if proc_id == 0:
            store.write_ex(ex_num, world, resume_epoch)
        dist.barrier()
    resume_snaps = _load_resume_snaps(store, resume_epoch)

    if local_n == 1:
        ctx = _DistCtx(None, proc_id, nprocs)
        worker = _Worker(
            graph, ctx, store, ex_num, resume_epoch, epoch_interval,
            resume_snaps,
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
        return

    shared = _HybridShared(dist, local_n, proc_id, nprocs)
    workers = [
        _Worker(
            graph,
            _HybridCtx(shared, t),
            store,
            ex_num,
            resume_epoch,
            epoch_interval,
            resume_snaps,
        )
        for t in range(local_n)
    ]

    def run_worker(wk: _Worker) -> None:
        try:
            wk.run()
        except _Interrupted:
            pass
        except BaseException as ex:  # noqa: BLE001
            with shared.lock:
                if shared.failure is None:
                    shared.failure = ex
            shared.barrier.abort()

    threads = [
        threading.Thread(
            target=run_worker,
            args=(wk,),
            name=f"bytewax-amd-worker-{proc_id}-{t}",
            daemon=True,
        )
        for t, wk in enumerate(workers[1:], start=1)
    ]
    for t in threads:
        t.start()
    try:
        run_worker(workers[0])
    finally:
        for t in threads:
            t.join()
        if store is not None:
            store.close()
        if owns_group:
            dist.destroy_process_group()
    if shared.failure is not None and not isinstance(
        shared.failure, AbortExecution
    ):
        raise shared.failure
