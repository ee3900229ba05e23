"""The MI355X columnar fast path.

Streams between operators on this path carry :class:`RecordBatch`
items — columnar event batches resident in HBM — instead of Python
objects.  The keyed/windowed hot aggregations run as hand-written
CDNA4 HIP kernels (see ``bytewax_amd/_native/stream_kernels.hip``):

- :class:`WindowAggState`: HBM-resident open-address keyed window
  state with fused insert+watermark kernels, wave-compacted window
  close extraction, and pinned-host snapshot spill;
- :func:`exchange_by_key`: key-hash bucketing kernel + RCCL
  all-to-allv over xGMI across workers-as-GPUs;
- :func:`keyed_window_agg`: the dataflow operator tying these into
  the engine's epoch/snapshot machinery (it is a `stateful_batch`
  under the hood, so recovery and EOF flushes work like any other
  stateful operator).

This is the engine counterpart of the semantics defined in
:mod:`bytewax_amd.operators.windowing` (tumbling windows, count/sum
folds) for device-resolvable aggregations; arbitrary Python logic
stays on the host path.
"""

from dataclasses import dataclass
from datetime import datetime, timezone
from typing import Any, Dict, Optional, Tuple

from ._ext import build as build_ext  # noqa: F401
from ._ext import ext

EPOCH_UTC = datetime(1970, 1, 1, tzinfo=timezone.utc)

AGG_COUNT = 0
AGG_SUM = 1

_INT64_MAX = (1 << 63) - 1


def _ms(dt: datetime) -> int:
    return int((dt - EPOCH_UTC).total_seconds() * 1000)


@dataclass
class RecordBatch:
    """A columnar batch of keyed events resident on one device.

    :arg keys: int32 key ids.
    :arg ts: int64 timestamps (ms since epoch).
    :arg vals: optional int64 payload values.
    :arg max_ts: max timestamp in the batch if known host-side
        (avoids a device sync for watermark tracking).
    """

    keys: Any  # torch.Tensor int32
    #: int64 absolute ms, or int32 deltas relative to `ts_base` (the
    #: compact wire/template form — halves the timestamp read traffic
    #: of the insert kernels).
    ts: Any  # torch.Tensor int64 | int32
    vals: Optional[Any] = None  # torch.Tensor int64
    max_ts: Optional[int] = None
    #: Scalar added to every `ts` on the fly by consuming kernels —
    #: lets sources reuse one timestamp-template tensor across steps.
    ts_base: int = 0

    def __len__(self) -> int:
        return int(self.keys.numel())


def _mix64_torch(x):
    """splitmix64 finalizer on int64 tensors, bit-identical to the
    device `mix64` (logical shifts emulated by masking the arithmetic
    shift's sign extension; multiplication wraps in both)."""

    def shr(v, s):
        return (v >> s) & ((1 << (64 - s)) - 1)

    x = x ^ shr(x, 33)
    x = x * -48064316817838067  # 0xff51afd7ed558ccd as int64
    x = x ^ shr(x, 33)
    x = x * -4265267296055464877  # 0xc4ceb9fe1a85ec53 as int64
    x = x ^ shr(x, 33)
    return x


def _bucket_cpu(batch: RecordBatch, world: int):
    """Host twin of the bucketing kernels (power-of-two worlds match
    the device hash exactly)."""
    import torch

    dst = torch.remainder(
        _mix64_torch(batch.keys.to(torch.int64)), world
    )
    order = torch.argsort(dst, stable=True)
    counts = torch.bincount(dst, minlength=world).to(torch.int32)
    send_keys = batch.keys[order]
    send_ts = batch.ts[order]
    send_vals = batch.vals[order] if batch.vals is not None else None
    return counts, send_keys, send_ts, send_vals


def exchange_by_key(
    batch: RecordBatch, group=None, async_op: bool = False,
    force: bool = False,
):
    """Exchange a batch across all workers so each key lands on its
    owning worker: bucketing kernel → RCCL all-to-allv over xGMI (gloo
    when the batch lives on CPU).

    Collective: every rank must call this once per scheduling step.

    With ``async_op`` the value collectives are issued asynchronously
    (the split-size exchange still synchronizes) and the result is
    ``(works, batch)`` — call ``w.wait()`` on each work handle before
    consuming the batch.  This lets the exchange of step N overlap the
    aggregation compute of step N-1 on the compute stream.
    """
    import torch
    import torch.distributed as dist

    world = dist.get_world_size(group)
    if world == 1 and not force:
        # `force` runs the full bucket+collective path even at world 1
        # (self-copy) — used to validate the RCCL wiring on a 1-GPU
        # lease where 2 ranks per device are refused.
        return ([], batch) if async_op else batch
    dev = batch.keys.device
    n = len(batch)
    has_vals = batch.vals is not None
    if dev.type == "cpu":
        counts, send_keys, send_ts, send_vals = _bucket_cpu(batch, world)
    else:
        k = ext()
        counts = torch.zeros(world, dtype=torch.int32, device=dev)
        k.bucket_hist(batch.keys, world, counts)
        offsets = torch.cumsum(counts, 0, dtype=torch.int32) - counts
        cursors = offsets.clone()
        send_keys = torch.empty(n, dtype=torch.int32, device=dev)
        send_ts = torch.empty(n, dtype=batch.ts.dtype, device=dev)
        send_vals = torch.empty(
            n if has_vals else 0, dtype=torch.int64, device=dev
        )
        k.bucket_scatter(
            batch.keys,
            batch.ts,
            batch.vals if has_vals else None,
            world,
            cursors,
            send_keys,
            send_ts,
            send_vals,
        )
    # Exchange split sizes (host sync of 2*world ints — control plane).
    recv_counts = torch.empty_like(counts)
    dist.all_to_all_single(recv_counts, counts, group=group)
    in_splits = counts.tolist()
    out_splits = recv_counts.tolist()
    m = int(sum(out_splits))

    # Timestamp compression: ship int32 deltas from this rank's base
    # (a batch spans far less than 2^31 ms) — cuts exchanged bytes by
    # a third on the biggest column.  Receivers rebuild absolute
    # timestamps from the per-source-rank bases.
    # `send_ts` may itself be relative to batch.ts_base (producers
    # ship zero-based templates); fold that producer base into the
    # wire base so absolute timestamps are never materialized here.
    if batch.ts.dtype == torch.int32:
        # Already wire format: int32 deltas relative to ts_base.
        ts_base = batch.ts_base
        send_ts32 = send_ts
    elif batch.max_ts is not None:
        # Host-known watermark avoids a device sync; any base within
        # 2^31 ms of every timestamp works.
        ts_base = batch.max_ts - (1 << 30)
        send_ts32 = (send_ts - (ts_base - batch.ts_base)).to(torch.int32)
    else:
        ts_base = (
            int(send_ts.min().item()) + batch.ts_base if n > 0 else 0
        )
        send_ts32 = (send_ts - (ts_base - batch.ts_base)).to(torch.int32)
    bases = torch.zeros(world, dtype=torch.int64, device=dev)
    my_base = torch.full((world,), ts_base, dtype=torch.int64, device=dev)
    dist.all_to_all_single(bases, my_base, group=group)

    recv_keys = torch.empty(m, dtype=torch.int32, device=dev)
    recv_ts32 = torch.empty(m, dtype=torch.int32, device=dev)
    works = []
    w = dist.all_to_all_single(
        recv_keys, send_keys, out_splits, in_splits, group=group,
        async_op=async_op,
    )
    if w is not None:
        works.append(w)
    w = dist.all_to_all_single(
        recv_ts32, send_ts32, out_splits, in_splits, group=group,
        async_op=async_op,
    )
    if w is not None:
        works.append(w)
    recv_vals = None
    if has_vals:
        recv_vals = torch.empty(m, dtype=torch.int64, device=dev)
        w = dist.all_to_all_single(
            recv_vals, send_vals, out_splits, in_splits, group=group,
            async_op=async_op,
        )
        if w is not None:
            works.append(w)
    # Bases were exchanged synchronously above; reading them to host is
    # a tiny D2H on the control plane (we already synced on the split
    # sizes).  Keeping them as scalars lets the radix insert consume
    # the int32 segments directly — no per-event base tensor at all.
    out = _LazyTsBatch(
        recv_keys, recv_ts32, out_splits, bases.tolist(), recv_vals,
        batch.max_ts,
    )
    if async_op:
        return (works, out)
    return out.materialize()


class _LazyTsBatch:
    """Exchange output in wire format: int32 timestamp deltas laid out
    as contiguous per-source-rank segments with scalar int64 bases.

    The radix insert path consumes this directly
    (:meth:`WindowAggState.insert_lazy` — one scatter launch per
    segment, no int64 rebuild); other consumers ``materialize()`` the
    absolute-timestamp :class:`RecordBatch` after the async exchange
    completes."""

    def __init__(self, keys, ts32, seg_counts, seg_bases, vals, max_ts):
        self.keys = keys
        self.ts32 = ts32
        self.seg_counts = list(seg_counts)
        self.seg_bases = [int(b) for b in seg_bases]
        self.vals = vals
        self.max_ts = max_ts

    def __len__(self) -> int:
        return self.keys.numel()

    def materialize(self) -> RecordBatch:
        import torch

        ts = self.ts32.to(torch.int64)
        off = 0
        for cnt, base in zip(self.seg_counts, self.seg_bases):
            if cnt and base:
                ts[off : off + cnt] += base
            off += cnt
        return RecordBatch(
            self.keys, ts, self.vals, max_ts=self.max_ts
        )


class WindowAggState:
    """HBM-resident keyed tumbling-window aggregation state.

    Slots are (window_id << 32 | key) → int64 accumulator in an
    open-address hash table sized for the live key×window working set.
    """

    def __init__(
        self,
        device,
        align_ms: int,
        len_ms: int,
        mode: int = AGG_COUNT,
        slots_pow: int = 20,
        dedup: bool = False,
        out_cap: int = 1 << 20,
        radix: bool = False,
        region_bits: int = 11,
        max_batch: int = 0,
        off_ms: int = 0,
        radix_v2: bool = False,
    ):
        """:arg radix: Use the radix-partitioned LDS-staged insert path
        (fastest for high-cardinality keys; requires `max_batch`, the
        largest batch size that will be inserted).
        :arg region_bits: log2 of the table-region size when `radix`.
        :arg off_ms: Window stride; 0 or == len_ms for tumbling,
            < len_ms for sliding (single-pass insert path only).
        """
        import torch

        self.device = device
        self.align_ms = align_ms
        self.len_ms = len_ms
        self.off_ms = off_ms if off_ms > 0 else len_ms
        if self.off_ms > len_ms:
            msg = "window offset must be <= length"
            raise ValueError(msg)
        if self.off_ms < len_ms and dedup:
            msg = "sliding windows are incompatible with the dedup path"
            raise ValueError(msg)
        # Sliding expansion factor: each event lands in this many
        # windows; radix scatter buffers scale by it.
        self.expand = (
            -(-len_ms // self.off_ms) if self.off_ms < len_ms else 1
        )
        self.mode = mode
        self.dedup = dedup
        self.max_ts_host = 0  # watermark if batches carry max_ts
        self.closed_horizon = -(1 << 62)  # window ids below are closed
        # Deferred close readback (close_launch/close_resolve).
        self._close_pending = False
        self._close_ev = None
        self._outn_pin = None
        self._carry_cpu = None
        if radix_v2 and (mode != AGG_COUNT or dedup):
            msg = "radix_v2 supports the COUNT mode only"
            raise ValueError(msg)
        self.radix_v2 = radix_v2 and device.type != "cpu"
        if self.radix_v2:
            radix = True  # shares the region table layout + buffers
        self.radix = radix
        self.region_bits = region_bits if radix else 0
        self.cpu = device.type == "cpu"
        if radix and not self.cpu:
            self.n_regions = (1 << slots_pow) >> region_bits
            if self.n_regions < 1:
                msg = "slots_pow must exceed region_bits"
                raise ValueError(msg)
            self.rx_gcursors = torch.zeros(
                self.n_regions, dtype=torch.int32, device=device
            )
            self.rx_ov_cursor = torch.zeros(
                1, dtype=torch.int32, device=device
            )
            self._alloc_rx(max(max_batch, 1))
        if self.cpu:
            # Host twin of the device table: used for CPU-only test
            # runs of the columnar path (gloo, no GPU).  Not a
            # production fallback — `ext()` still hard-fails on GPU
            # machines if the extension is missing.
            self._table: Dict[Tuple[int, int], int] = {}
            return
        self.k = ext()
        self.nslots = 1 << slots_pow
        self.tkeys = torch.full(
            (self.nslots,), -1, dtype=torch.int64, device=device
        )
        self.tvals = torch.zeros(self.nslots, dtype=torch.int64, device=device)
        # Double-buffered: window close migrates live cells into the
        # alternate table (in-place slot deletion would break probe
        # chains) and swaps.
        self.tkeys_alt = torch.full(
            (self.nslots,), -1, dtype=torch.int64, device=device
        )
        self.tvals_alt = torch.zeros(
            self.nslots, dtype=torch.int64, device=device
        )
        self.max_ts_dev = torch.zeros(1, dtype=torch.int64, device=device)
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)
        self.out_cap = out_cap
        self.out_keys = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out_wins = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out_vals = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_n = torch.zeros(1, dtype=torch.int32, device=device)
        # Pinned staging for snapshot spill to host DRAM.
        self._pin_keys = None

    def _insert_cpu(self, batch: RecordBatch) -> None:
        import numpy as np

        keys = batch.keys.numpy()
        ts64 = batch.ts.numpy().astype("int64", copy=False)
        wins = (
            (ts64 + batch.ts_base - self.align_ms) // self.off_ms
        ).astype("int64")
        if self.mode == AGG_COUNT:
            vals = np.ones(len(keys), dtype="int64")
        else:
            vals = batch.vals.numpy()
        packed = (wins << 32) | keys.astype("uint32")
        uniq, inv = np.unique(packed, return_inverse=True)
        sums = np.bincount(inv, weights=vals).astype("int64")
        for p, s in zip(uniq.tolist(), sums.tolist()):
            kw = (int(np.uint32(p & 0xFFFFFFFF)), int(p >> 32))
            self._table[kw] = self._table.get(kw, 0) + s
        if self.off_ms < self.len_ms:
            # Sliding: also fold into the earlier overlapped windows.
            t_arr = batch.ts.numpy().astype("int64", copy=False) + batch.ts_base
            lo = (t_arr - self.align_ms - self.len_ms) // self.off_ms + 1
            for k, w_hi, w_lo, v in zip(
                keys.tolist(), wins.tolist(), lo.tolist(), vals.tolist()
            ):
                for wn in range(int(w_lo), int(w_hi)):
                    kw = (int(k), int(wn))
                    self._table[kw] = self._table.get(kw, 0) + int(v)
        mx = int(batch.ts.max().item()) + batch.ts_base if len(batch) else 0
        if mx > self.max_ts_host:
            self.max_ts_host = mx

    def _alloc_rx(self, max_batch: int) -> None:
        """Fixed-capacity per-region scatter buffers (~2.5x the batch)
        plus the overflow spill."""
        import torch

        # Floor of 4 granule lines per region so the staged scatter's
        # 8-event reservation granularity always has working capacity
        # even for tiny states/batches.
        per_region = max(
            32,
            -(-max_batch * self.expand * 5 // 2) // self.n_regions + 1,
        )
        total = per_region * self.n_regions
        self.rx_max_batch = max_batch
        self.rx_packed = torch.empty(
            total, dtype=torch.int64, device=self.device
        )
        if getattr(self, "radix_v2", False):
            cap_c = (-(-max_batch * 5 // 2) // 256 + 8) // 8 * 8
            res_cap = max(1 << 14, cap_c // 4)
            self.v2_gcur = torch.zeros(256, dtype=torch.int32, device=self.device)
            self.v2_gres = torch.zeros(256, dtype=torch.int32, device=self.device)
            self.v2_ev = torch.empty(
                256 * cap_c, dtype=torch.int64, device=self.device
            )
            self.v2_ev_res = torch.empty(
                256 * res_cap, dtype=torch.int64, device=self.device
            )
        self.rx_vals = torch.empty(
            total if self.mode == AGG_SUM else 0,
            dtype=torch.int64,
            device=self.device,
        )
        # Sized for the degenerate case where most events hash into a
        # few regions (tiny key cardinality): the whole batch may
        # spill and aggregate through the direct path — slow but
        # correct.  Use `dedup=True` for such workloads instead.
        ov = max(1 << 20, max_batch)
        self.rx_ov_packed = torch.empty(
            ov, dtype=torch.int64, device=self.device
        )
        self.rx_ov_vals = torch.empty(
            ov if self.mode == AGG_SUM else 0,
            dtype=torch.int64,
            device=self.device,
        )
        if hasattr(self, "rx_packed2"):
            # Keep the pipelined second buffer set in sync on growth.
            self.rx_gcursors2 = torch.zeros_like(self.rx_gcursors)
            self.rx_packed2 = torch.empty_like(self.rx_packed)
            self.rx_vals2 = torch.empty_like(self.rx_vals)
            self.rx_ov_cursor2 = torch.zeros_like(self.rx_ov_cursor)
            self.rx_ov_packed2 = torch.empty_like(self.rx_ov_packed)
            self.rx_ov_vals2 = torch.empty_like(self.rx_ov_vals)

    def insert(self, batch: RecordBatch) -> None:
        if self.cpu:
            self._insert_cpu(batch)
            # Same contract as the device path: a host-known max_ts
            # advances the watermark even past the event timestamps.
            if (
                batch.max_ts is not None
                and batch.max_ts > self.max_ts_host
            ):
                self.max_ts_host = batch.max_ts
            return
        if self.radix and len(batch) > self.rx_max_batch:
            # Scatter buffers grow to fit the largest batch seen
            # (exchange-received batches vary in size; sliding
            # expansion is applied inside _alloc_rx).
            self._alloc_rx(int(len(batch) * 5 // 4))
        import torch

        if self.radix_v2:
            self.k.radix_v2_window_insert(
                batch.keys,
                batch.ts.to(torch.int64),
                self.tkeys,
                self.tvals,
                self.max_ts_dev,
                self.error_flag,
                self.v2_gcur,
                self.v2_gres,
                self.v2_ev,
                self.v2_ev_res,
                self.align_ms,
                self.len_ms,
                batch.ts_base,
                self.region_bits,
            )
        elif self.radix:
            self.k.radix_window_insert(
                batch.keys,
                batch.ts,
                batch.vals,
                self.tkeys,
                self.tvals,
                self.max_ts_dev,
                self.error_flag,
                self.rx_gcursors,
                self.rx_packed,
                self.rx_vals,
                self.rx_ov_cursor,
                self.rx_ov_packed,
                self.rx_ov_vals,
                self.align_ms,
                self.len_ms,
                self.mode,
                batch.ts_base,
                self.region_bits,
                self.off_ms,
                [],
                [],
            )
        else:
            self.k.window_agg_insert(
                batch.keys,
                batch.ts,
                batch.vals,
                self.tkeys,
                self.tvals,
                self.max_ts_dev,
                self.error_flag,
                self.align_ms,
                self.len_ms,
                self.mode,
                self.dedup,
                batch.ts_base,
                self.region_bits,
                self.off_ms,
            )
        if batch.max_ts is not None and batch.max_ts > self.max_ts_host:
            self.max_ts_host = batch.max_ts

    def _ensure_pipe(self) -> None:
        import torch

        if getattr(self, "_pipe_ready", False):
            return
        self._pipe_ready = True
        self._sc_stream = torch.cuda.Stream(self.device)
        self._pipe_parity = 0
        self._ev_sc = [torch.cuda.Event(), torch.cuda.Event()]
        self._ev_ag = [torch.cuda.Event(), torch.cuda.Event()]
        self._ag_recorded = [False, False]
        if not hasattr(self, "rx_packed2"):
            self.rx_gcursors2 = torch.zeros_like(self.rx_gcursors)
            self.rx_packed2 = torch.empty_like(self.rx_packed)
            self.rx_vals2 = torch.empty_like(self.rx_vals)
            self.rx_ov_cursor2 = torch.zeros_like(self.rx_ov_cursor)
            self.rx_ov_packed2 = torch.empty_like(self.rx_ov_packed)
            self.rx_ov_vals2 = torch.empty_like(self.rx_ov_vals)

    def _pipe_bufs(self, par: int):
        if par == 0:
            return (
                self.rx_gcursors,
                self.rx_packed,
                self.rx_vals,
                self.rx_ov_cursor,
                self.rx_ov_packed,
                self.rx_ov_vals,
            )
        return (
            self.rx_gcursors2,
            self.rx_packed2,
            self.rx_vals2,
            self.rx_ov_cursor2,
            self.rx_ov_packed2,
            self.rx_ov_vals2,
        )

    def insert_pipelined(
        self,
        keys,
        ts,
        vals,
        ts_base: int,
        seg_counts,
        seg_bases,
        max_ts: Optional[int],
        works=None,
    ) -> None:
        """Radix insert with the scatter on a side stream (the per-step
        engine's twin of the native loop's two-stream pipeline).

        The scatter of this call overlaps the aggregation of the
        PREVIOUS call (alternating buffer parity; events fence buffer
        reuse).  ``works`` (async exchange handles) are waited on the
        side stream, so the NCCL completion gates only the scatter —
        not the previous step's aggregation."""
        import torch

        n = int(keys.numel())
        if n > self.rx_max_batch:
            if getattr(self, "_pipe_ready", False):
                # Retire in-flight side-stream work before freeing the
                # old buffers it may still be reading.
                torch.cuda.synchronize(self.device)
            self._alloc_rx(int(n * 5 // 4))
        self._ensure_pipe()
        par = self._pipe_parity
        self._pipe_parity ^= 1
        bufs = self._pipe_bufs(par)
        cur = torch.cuda.current_stream(self.device)
        with torch.cuda.stream(self._sc_stream):
            if works:
                for w in works:
                    w.wait()
            if self._ag_recorded[par]:
                self._sc_stream.wait_event(self._ev_ag[par])
            # Consumed on a non-allocating stream: pin lifetimes.
            keys.record_stream(self._sc_stream)
            ts.record_stream(self._sc_stream)
            if vals is not None:
                vals.record_stream(self._sc_stream)
            self.k.radix_scatter_only(
                keys,
                ts,
                vals,
                self.max_ts_dev,
                self.error_flag,
                *bufs,
                self.nslots,
                self.align_ms,
                self.len_ms,
                self.mode,
                ts_base,
                self.region_bits,
                self.off_ms,
                list(seg_counts),
                [int(b) for b in seg_bases],
            )
            self._ev_sc[par].record(self._sc_stream)
        cur.wait_event(self._ev_sc[par])
        self.k.radix_agg_only(
            self.tkeys,
            self.tvals,
            self.error_flag,
            *bufs,
            self.mode,
            self.region_bits,
        )
        self._ev_ag[par].record(cur)
        self._ag_recorded[par] = True
        if max_ts is not None and max_ts > self.max_ts_host:
            self.max_ts_host = max_ts

    def insert_lazy(self, lz: "_LazyTsBatch") -> None:
        """Insert exchange output in wire format (int32 timestamp
        deltas in per-source-rank segments).

        On the radix path the scatter kernel consumes the segments
        directly with their scalar bases, skipping the int64 timestamp
        rebuild (~3 HBM passes over the biggest column per step at
        world > 1).  Other paths materialize first.
        """
        if self.cpu or not self.radix or self.radix_v2:
            self.insert(lz.materialize())
            return
        n = len(lz)
        if n > self.rx_max_batch:
            self._alloc_rx(int(n * 5 // 4))
        self.k.radix_window_insert(
            lz.keys,
            lz.ts32,
            lz.vals,
            self.tkeys,
            self.tvals,
            self.max_ts_dev,
            self.error_flag,
            self.rx_gcursors,
            self.rx_packed,
            self.rx_vals,
            self.rx_ov_cursor,
            self.rx_ov_packed,
            self.rx_ov_vals,
            self.align_ms,
            self.len_ms,
            self.mode,
            0,
            self.region_bits,
            self.off_ms,
            lz.seg_counts,
            lz.seg_bases,
        )
        if lz.max_ts is not None and lz.max_ts > self.max_ts_host:
            self.max_ts_host = lz.max_ts

    def watermark_ms(self, sync: bool = False) -> int:
        if sync and not self.cpu:
            dev = int(self.max_ts_dev.item())
            if dev > self.max_ts_host:
                self.max_ts_host = dev
        return self.max_ts_host

    def _extract_cpu(
        self, win_lo: int, win_hi: int, delete: bool
    ) -> Optional[RecordBatch]:
        import torch

        hit = [
            (k, w, v)
            for (k, w), v in self._table.items()
            if win_lo <= w < win_hi
        ]
        if not hit:
            return None
        if delete:
            for k, w, _v in hit:
                del self._table[(k, w)]
        keys = torch.tensor([k for k, _w, _v in hit], dtype=torch.int32)
        ts = torch.tensor(
            [w * self.off_ms + self.align_ms for _k, w, _v in hit],
            dtype=torch.int64,
        )
        vals = torch.tensor([v for _k, _w, v in hit], dtype=torch.int64)
        return RecordBatch(keys, ts, vals)

    def _read_out(self) -> Optional[RecordBatch]:
        import torch

        n = int(self.out_n.item())  # syncs; amortized over window period
        if n > self.out_cap:
            msg = (
                f"window close produced {n} rows > out_cap {self.out_cap}; "
                "increase out_cap"
            )
            raise RuntimeError(msg)
        if int(self.error_flag.item()) != 0:
            msg = "keyed window state table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        if n == 0:
            return None
        return RecordBatch(
            self.out_keys[:n].clone(),
            self.out_wins[:n].to(torch.int64) * self.off_ms + self.align_ms,
            self.out_vals[:n].clone(),
        )

    def _extract_range(self, win_lo: int, win_hi: int) -> Optional[RecordBatch]:
        """Read cells in [win_lo, win_hi) without mutating the table."""
        if self.cpu:
            return self._extract_cpu(win_lo, win_hi, delete=False)
        self.out_n.zero_()
        self.k.close_extract(
            self.tkeys,
            self.tvals,
            win_lo,
            win_hi,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
        )
        return self._read_out()

    def close_due(self, wait_ms: int = 0) -> Optional[RecordBatch]:
        """Extract all windows fully below the current watermark and
        reclaim their space (migrate live cells to a fresh table)."""
        wm = self.watermark_ms()
        horizon = (
            wm - wait_ms - self.align_ms - self.len_ms
        ) // self.off_ms + 1
        if horizon <= self.closed_horizon:
            return None
        if self.cpu:
            out = self._extract_cpu(-(1 << 40), horizon, delete=True)
            self.closed_horizon = horizon
            return out
        self.out_n.zero_()
        self.k.close_migrate(
            self.tkeys,
            self.tvals,
            self.tkeys_alt,
            self.tvals_alt,
            horizon,
            self.region_bits,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
            self.error_flag,
        )
        self.tkeys, self.tkeys_alt = self.tkeys_alt, self.tkeys
        self.tvals, self.tvals_alt = self.tvals_alt, self.tvals
        # Reset the retired table asynchronously for the next close.
        self.tkeys_alt.fill_(-1)
        self.tvals_alt.zero_()
        out = self._read_out()
        self.closed_horizon = horizon
        return out

    def close_launch(self, wait_ms: int = 0) -> None:
        """Launch the due-window close without reading it back.

        The migrate kernel and the row-count D2H land on the stream
        asynchronously; :meth:`close_resolve` (typically called at the
        START of the next scheduling step) picks up the emitted rows.
        This keeps the close latency hidden behind one step of
        pipeline instead of stalling the stream at every watermark
        advance — under the per-step Python engine the synchronous
        readback was the p99 outlier.
        """
        import torch

        if self._close_pending:
            # One in flight at a time; resolve first.
            return
        wm = self.watermark_ms()
        horizon = (
            wm - wait_ms - self.align_ms - self.len_ms
        ) // self.off_ms + 1
        if horizon <= self.closed_horizon:
            return
        if self.cpu:
            self._carry_cpu = self._extract_cpu(
                -(1 << 40), horizon, delete=True
            )
            self.closed_horizon = horizon
            self._close_pending = self._carry_cpu is not None
            return
        self.out_n.zero_()
        self.k.close_migrate(
            self.tkeys,
            self.tvals,
            self.tkeys_alt,
            self.tvals_alt,
            horizon,
            self.region_bits,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
            self.error_flag,
        )
        self.tkeys, self.tkeys_alt = self.tkeys_alt, self.tkeys
        self.tvals, self.tvals_alt = self.tvals_alt, self.tvals
        self.tkeys_alt.fill_(-1)
        self.tvals_alt.zero_()
        self.closed_horizon = horizon
        if self._outn_pin is None:
            self._outn_pin = torch.zeros(
                2,
                dtype=torch.int32,
                pin_memory=torch.cuda.is_available(),
            )
        self._outn_pin[0:1].copy_(self.out_n, non_blocking=True)
        self._outn_pin[1:2].copy_(self.error_flag, non_blocking=True)
        self._close_ev = torch.cuda.Event()
        self._close_ev.record()
        self._close_pending = True

    def close_resolve(self) -> Optional[RecordBatch]:
        """Read back the rows of the last :meth:`close_launch`.

        Must run before the NEXT close_launch / close_all /
        snapshot_to_host (those reuse the out staging buffers)."""
        import torch

        if not self._close_pending:
            return None
        self._close_pending = False
        if self.cpu:
            out = self._carry_cpu
            self._carry_cpu = None
            return out
        self._close_ev.synchronize()
        n = int(self._outn_pin[0].item())
        if int(self._outn_pin[1].item()) != 0:
            msg = "keyed window state table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        if n > self.out_cap:
            msg = (
                f"window close produced {n} rows > out_cap "
                f"{self.out_cap}; increase out_cap"
            )
            raise RuntimeError(msg)
        if n == 0:
            return None
        return RecordBatch(
            self.out_keys[:n].clone(),
            self.out_wins[:n].to(torch.int64) * self.off_ms + self.align_ms,
            self.out_vals[:n].clone(),
        )

    def close_all(self) -> Optional[RecordBatch]:
        """EOF: close every window at or above the closed horizon."""
        if self._close_pending:
            msg = "close_resolve() the pending close before close_all()"
            raise RuntimeError(msg)
        if self.cpu:
            return self._extract_cpu(self.closed_horizon, 1 << 40, delete=True)
        out = self._extract_range(self.closed_horizon, 1 << 40)
        self.closed_horizon = 1 << 40
        return out

    def snapshot_to_host(self) -> Dict[str, Any]:
        """Spill the live table to pinned host memory (recovery)."""
        import torch

        if self.cpu:
            import numpy as np

            items = list(self._table.items())
            return {
                "keys": np.array([k for (k, _w), _v in items], dtype="int32"),
                "wins": np.array([w for (_k, w), _v in items], dtype="int32"),
                "vals": np.array([v for _kw, v in items], dtype="int64"),
                "max_ts": self.watermark_ms(),
                "closed_horizon": self.closed_horizon,
            }
        self.out_n.zero_()
        self.k.close_extract(
            self.tkeys,
            self.tvals,
            self.closed_horizon,
            1 << 40,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
        )
        n = int(self.out_n.item())
        if self._pin_keys is None or self._pin_keys.numel() < n:
            cap = max(n, 1)
            pin = torch.cuda.is_available()
            self._pin_keys = torch.empty(cap, dtype=torch.int32, pin_memory=pin)
            self._pin_wins = torch.empty(cap, dtype=torch.int32, pin_memory=pin)
            self._pin_vals = torch.empty(cap, dtype=torch.int64, pin_memory=pin)
        self._pin_keys[:n].copy_(self.out_keys[:n], non_blocking=True)
        self._pin_wins[:n].copy_(self.out_wins[:n], non_blocking=True)
        self._pin_vals[:n].copy_(self.out_vals[:n], non_blocking=True)
        if torch.cuda.is_available():
            torch.cuda.synchronize(self.device)
        return {
            "keys": self._pin_keys[:n].numpy().copy(),
            "wins": self._pin_wins[:n].numpy().copy(),
            "vals": self._pin_vals[:n].numpy().copy(),
            "max_ts": self.watermark_ms(),
            "closed_horizon": self.closed_horizon,
        }

    def native_run(
        self,
        key_pool,
        ts_pool,
        start_step: int,
        n_steps: int,
        sim_ms_per_batch: int,
        wait_ms: int = 0,
        pipelined: bool = True,
    ):
        """Run `n_steps` of the window pipeline through the native C++
        step loop (no Python between steps).  Single-worker path; the
        multi-GPU exchange path uses the per-step Python loop.

        With ``pipelined`` (radix only), the scatter of step N+1 runs
        on a second HIP stream into a second region-buffer set while
        the aggregation of step N drains the first — the agg stage
        hides behind the scatter.

        Returns (closed_rows, step_launch_ns: torch int64 tensor).
        """
        import torch

        if self.cpu:
            msg = "native_run requires a device table"
            raise RuntimeError(msg)
        pipe = pipelined and self.radix and not self.radix_v2
        if pipe and not hasattr(self, "rx_gcursors2"):
            self.rx_gcursors2 = torch.zeros_like(self.rx_gcursors)
            self.rx_packed2 = torch.empty_like(self.rx_packed)
            self.rx_ov_cursor2 = torch.zeros_like(self.rx_ov_cursor)
            self.rx_ov_packed2 = torch.empty_like(self.rx_ov_packed)
        step_ns = torch.zeros(n_steps, dtype=torch.int64)
        state_out = torch.zeros(3, dtype=torch.int64)
        rows = self.k.native_run_window_steps(
            list(key_pool),
            list(ts_pool),
            start_step,
            n_steps,
            sim_ms_per_batch,
            self.tkeys,
            self.tvals,
            self.max_ts_dev,
            self.error_flag,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
            self.align_ms,
            self.len_ms,
            wait_ms,
            self.mode,
            self.dedup,
            self.closed_horizon,
            step_ns,
            state_out,
            self.region_bits,
            self.radix,
            self.rx_gcursors if self.radix else None,
            self.rx_packed if self.radix else None,
            self.rx_vals if self.radix else None,
            self.rx_ov_cursor if self.radix else None,
            self.rx_ov_packed if self.radix else None,
            self.rx_ov_vals if self.radix else None,
            self.tkeys_alt,
            self.tvals_alt,
            self.radix_v2,
            self.v2_gcur if self.radix_v2 else None,
            self.v2_gres if self.radix_v2 else None,
            self.v2_ev if self.radix_v2 else None,
            self.v2_ev_res if self.radix_v2 else None,
            pipe,
            self.rx_gcursors2 if pipe else None,
            self.rx_packed2 if pipe else None,
            self.rx_ov_cursor2 if pipe else None,
            self.rx_ov_packed2 if pipe else None,
        )
        if int(state_out[2].item()) == 1:
            self.tkeys, self.tkeys_alt = self.tkeys_alt, self.tkeys
            self.tvals, self.tvals_alt = self.tvals_alt, self.tvals
        self.closed_horizon = int(state_out[0].item())
        self.max_ts_host = max(
            self.max_ts_host,
            self.align_ms + (start_step + n_steps) * sim_ms_per_batch - 1,
        )
        if int(self.error_flag.item()) != 0:
            msg = "keyed window state table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        return rows, step_ns

    def native_run_graph(
        self,
        key_pool,
        ts_pool,
        start_step: int,
        n_steps: int,
        sim_ms_per_batch: int,
        wait_ms: int = 0,
    ) -> int:
        """Run `n_steps` through the hipGraph-captured native loop: one
        pool cycle of {insert, timestamp-bump} kernels is captured once
        and each replay costs a single hipGraphLaunch (latency mode;
        COUNT, single-pass).  Returns closed rows."""
        import torch

        if self.cpu or self.radix or self.dedup or self.mode != AGG_COUNT:
            msg = (
                "graph mode supports the single-pass COUNT path on GPU"
            )
            raise RuntimeError(msg)
        if not hasattr(self, "_ts_base_dev"):
            self._ts_base_dev = torch.zeros(
                1, dtype=torch.int64, device=self.device
            )
        state_out = torch.zeros(3, dtype=torch.int64)
        rows = self.k.native_run_window_steps_graph(
            list(key_pool),
            list(ts_pool),
            start_step,
            n_steps,
            sim_ms_per_batch,
            self.tkeys,
            self.tvals,
            self.max_ts_dev,
            self.error_flag,
            self.out_keys,
            self.out_wins,
            self.out_vals,
            self.out_n,
            self.align_ms,
            self.len_ms,
            wait_ms,
            self.closed_horizon,
            state_out,
            self.region_bits,
            self.tkeys_alt,
            self.tvals_alt,
            self._ts_base_dev,
        )
        self.closed_horizon = int(state_out[0].item())
        self.max_ts_host = max(
            self.max_ts_host,
            self.align_ms + (start_step + n_steps) * sim_ms_per_batch - 1,
        )
        if int(self.error_flag.item()) != 0:
            msg = "keyed window state table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        return rows

    def restore_from_host(self, snap: Dict[str, Any]) -> None:
        import torch

        n = len(snap["keys"])
        if self.cpu:
            for k, w, v in zip(
                snap["keys"].tolist(),
                snap["wins"].tolist(),
                snap["vals"].tolist(),
            ):
                kw = (int(k), int(w))
                self._table[kw] = self._table.get(kw, 0) + int(v)
            self.max_ts_host = snap["max_ts"]
            self.closed_horizon = snap["closed_horizon"]
            return
        if n:
            keys = torch.as_tensor(snap["keys"]).to(self.device)
            wins = torch.as_tensor(snap["wins"]).to(torch.int64)
            # Timestamps chosen so the kernel's `(t - align) / stride`
            # reproduces the exact window ids (stride == len disables
            # sliding expansion during restore).
            ts = (wins * self.len_ms + self.align_ms).to(self.device)
            vals = torch.as_tensor(snap["vals"]).to(self.device)
            self.k.window_agg_insert(
                keys,
                ts,
                vals,
                self.tkeys,
                self.tvals,
                self.max_ts_dev,
                self.error_flag,
                self.align_ms,
                self.len_ms,
                AGG_SUM,  # re-add saved accumulators regardless of mode
                False,
                0,
                self.region_bits,
                self.len_ms,
            )
        self.max_ts_host = snap["max_ts"]
        self.closed_horizon = snap["closed_horizon"]

    def reinsert_labeled(self, batch: RecordBatch) -> None:
        """Re-add spilled accumulator rows (rescale re-exchange).

        ``batch.ts`` carries window LABELS (win * len + align, the
        snapshot/emission format), and ``vals`` are saved accumulators
        — added with SUM semantics regardless of mode, exactly like
        :meth:`restore_from_host`.  Does not advance the watermark or
        the closed horizon (the caller merges those)."""
        if len(batch) == 0:
            return
        if self.cpu:
            wins = (
                (batch.ts + batch.ts_base - self.align_ms) // self.len_ms
            ).tolist()
            for k, w, v in zip(
                batch.keys.tolist(), wins, batch.vals.tolist()
            ):
                kw = (int(k), int(w))
                self._table[kw] = self._table.get(kw, 0) + int(v)
            return
        self.k.window_agg_insert(
            batch.keys,
            batch.ts,
            batch.vals,
            self.tkeys,
            self.tvals,
            self.max_ts_dev,
            self.error_flag,
            self.align_ms,
            self.len_ms,
            AGG_SUM,  # re-add saved accumulators regardless of mode
            False,
            batch.ts_base,
            self.region_bits,
            self.len_ms,
        )
