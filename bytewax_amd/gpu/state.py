"""Additional HBM-resident keyed state: running stats and hash join.

Device twins of `reduce_final`-style keyed aggregations and the
`join` operator for the columnar path:

- :class:`StatsAggState` — 1BRC-style count/sum/min/max per
  (key, window) in one fused pass (BASELINE config 5);
- :class:`HashJoinState` — stream-stream join with HBM-resident
  open-address state, "last" insert / "complete" emit (config 4).

Both support pinned-host snapshot spill for recovery and have CPU
twins for GPU-less test runs.
"""

from typing import Any, Dict, List, Optional, Tuple

from . import AGG_COUNT, AGG_SUM, RecordBatch, _ms  # noqa: F401
from ._ext import ext

_I64_MAX = (1 << 63) - 1
_I64_MIN = -(1 << 63)


class StatsAggState:
    """Keyed (windowed) running count/sum/min/max on device."""

    def __init__(
        self,
        device,
        align_ms: int,
        len_ms: int,
        slots_pow: int = 20,
        out_cap: int = 1 << 20,
        radix: bool = True,
        region_bits: int = 10,
    ):
        import torch

        self.device = device
        self.align_ms = align_ms
        self.len_ms = len_ms
        self.max_ts_host = 0
        self.closed_horizon = -(1 << 62)
        self.cpu = device.type == "cpu"
        if self.cpu:
            self._table: Dict[Tuple[int, int], Tuple[int, int, int, int]] = {}
            return
        self.k = ext()
        # The radix path needs enough regions for parallelism; keep at
        # least 1024 workgroups' worth.
        if radix:
            slots_pow = max(slots_pow, region_bits + 10)
        self.radix = radix
        self.region_bits = region_bits
        self.nslots = 1 << slots_pow
        if radix:
            self.n_regions = self.nslots >> region_bits
            self.rx_gcursors = torch.zeros(
                self.n_regions, dtype=torch.int32, device=device
            )
            self.rx_ov_cursor = torch.zeros(
                1, dtype=torch.int32, device=device
            )
            self.rx_max_batch = 0
            self.rx_packed = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_vals = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_ov_packed = torch.empty(
                1, dtype=torch.int64, device=device
            )
            self.rx_ov_vals = torch.empty(1, dtype=torch.int64, device=device)
        self.tkeys = torch.full(
            (self.nslots,), -1, dtype=torch.int64, device=device
        )
        self.tcnt = torch.zeros(self.nslots, dtype=torch.int64, device=device)
        self.tsum = torch.zeros(self.nslots, dtype=torch.int64, device=device)
        self.tmin = torch.full(
            (self.nslots,), _I64_MAX, dtype=torch.int64, device=device
        )
        self.tmax = torch.full(
            (self.nslots,), _I64_MIN, dtype=torch.int64, device=device
        )
        self.max_ts_dev = torch.zeros(1, dtype=torch.int64, device=device)
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)
        self.out_cap = out_cap
        self.out_keys = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out_wins = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out = {
            name: torch.empty(out_cap, dtype=torch.int64, device=device)
            for name in ("cnt", "sum", "min", "max")
        }
        self.out_n = torch.zeros(1, dtype=torch.int32, device=device)

    def insert(self, batch: RecordBatch) -> None:
        import torch

        if batch.vals is None:
            msg = "stats aggregation requires a `vals` column"
            raise ValueError(msg)
        if (
            not self.cpu
            and not self.radix
            and batch.ts.dtype != torch.int64
        ):
            # The plain (non-radix) stats kernel takes absolute int64
            # timestamps; the radix scatter consumes int32 templates
            # natively.
            batch = RecordBatch(
                batch.keys,
                batch.ts.to(torch.int64),
                batch.vals,
                max_ts=batch.max_ts,
                ts_base=batch.ts_base,
            )
        if self.cpu:
            self._insert_cpu(batch)
        elif self.radix:
            import torch

            if len(batch) > self.rx_max_batch:
                mb = int(len(batch) * 5 // 4)
                self.rx_max_batch = mb
                per_region = -(-mb * 5 // 2) // self.n_regions + 1
                total = per_region * self.n_regions
                self.rx_packed = torch.empty(
                    total, dtype=torch.int64, device=self.device
                )
                self.rx_vals = torch.empty(
                    total, dtype=torch.int64, device=self.device
                )
                ov = max(1 << 20, mb)
                self.rx_ov_packed = torch.empty(
                    ov, dtype=torch.int64, device=self.device
                )
                self.rx_ov_vals = torch.empty(
                    ov, dtype=torch.int64, device=self.device
                )
            self.k.radix_stats_insert(
                batch.keys,
                batch.ts,
                batch.vals,
                self.tkeys,
                self.tcnt,
                self.tsum,
                self.tmin,
                self.tmax,
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
                batch.ts_base,
                self.region_bits,
            )
        else:
            self.k.stats_insert(
                batch.keys,
                batch.ts,
                batch.vals,
                self.tkeys,
                self.tcnt,
                self.tsum,
                self.tmin,
                self.tmax,
                self.max_ts_dev,
                self.error_flag,
                self.align_ms,
                self.len_ms,
                batch.ts_base,
            )
        if batch.max_ts is not None and batch.max_ts > self.max_ts_host:
            self.max_ts_host = batch.max_ts

    def _insert_cpu(self, batch: RecordBatch) -> None:
        keys = batch.keys.tolist()
        import torch

        wins = (
            (batch.ts.to(torch.int64) + batch.ts_base - self.align_ms)
            // self.len_ms
        ).tolist()
        vals = batch.vals.tolist()
        for k, w, v in zip(keys, wins, vals):
            cnt, s, mn, mx = self._table.get(
                (k, w), (0, 0, _I64_MAX, _I64_MIN)
            )
            self._table[(k, w)] = (
                cnt + 1, s + v, min(mn, v), max(mx, v)
            )
        if len(batch):
            mx_ts = int(batch.ts.max().item()) + batch.ts_base
            if mx_ts > self.max_ts_host:
                self.max_ts_host = mx_ts

    def extract(
        self, horizon: Optional[int] = None, clear: bool = True
    ) -> Optional[Dict[str, Any]]:
        """Extract (key, window) stats in [closed_horizon, horizon)
        (up to +inf if None).  On device the table is never mutated —
        already-emitted cells are excluded by the horizon range (no
        in-place deletion: it would corrupt probe chains); callers
        advance `closed_horizon` after a closing extract.  Size
        `slots_pow` for the stream's total distinct (key, window)
        cells (1BRC-style whole-stream aggregation has exactly one
        window)."""
        import torch

        if horizon is None:
            horizon = 1 << 40
        win_lo = self.closed_horizon
        if win_lo < -(1 << 39):
            win_lo = -(1 << 39)
        if self.cpu:
            hit = [
                (k, w, v)
                for (k, w), v in self._table.items()
                if win_lo <= w < horizon
            ]
            if not hit:
                return None
            if clear:
                for k, w, _v in hit:
                    del self._table[(k, w)]
            return {
                "keys": torch.tensor([k for k, _w, _v in hit], dtype=torch.int32),
                "wins": torch.tensor([w for _k, w, _v in hit], dtype=torch.int32),
                "cnt": torch.tensor([v[0] for *_x, v in hit], dtype=torch.int64),
                "sum": torch.tensor([v[1] for *_x, v in hit], dtype=torch.int64),
                "min": torch.tensor([v[2] for *_x, v in hit], dtype=torch.int64),
                "max": torch.tensor([v[3] for *_x, v in hit], dtype=torch.int64),
            }
        self.out_n.zero_()
        self.k.stats_extract(
            self.tkeys,
            self.tcnt,
            self.tsum,
            self.tmin,
            self.tmax,
            win_lo,
            horizon,
            self.out_keys,
            self.out_wins,
            self.out["cnt"],
            self.out["sum"],
            self.out["min"],
            self.out["max"],
            self.out_n,
        )
        n = int(self.out_n.item())
        if n == 0:
            return None
        if n > self.out_cap:
            msg = f"stats extract produced {n} rows > out_cap"
            raise RuntimeError(msg)
        if int(self.error_flag.item()) != 0:
            msg = "stats table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        return {
            "keys": self.out_keys[:n].clone(),
            "wins": self.out_wins[:n].clone(),
            "cnt": self.out["cnt"][:n].clone(),
            "sum": self.out["sum"][:n].clone(),
            "min": self.out["min"][:n].clone(),
            "max": self.out["max"][:n].clone(),
        }

    def snapshot_to_host(self) -> Dict[str, Any]:
        import torch

        snap = self.extract(None, clear=False)
        out: Dict[str, Any] = {
            "max_ts": self.max_ts_host,
            "closed_horizon": self.closed_horizon,
        }
        if snap is None:
            out["n"] = 0
            return out
        for name, t in snap.items():
            out[name] = t.cpu().numpy().copy()
        out["n"] = len(out["keys"])
        return out

    def restore_from_host(self, snap: Dict[str, Any]) -> None:
        self.max_ts_host = snap["max_ts"]
        self.closed_horizon = snap["closed_horizon"]
        if snap["n"] == 0:
            return
        self.merge_rows(snap)

    def merge_rows(self, snap: Dict[str, Any]) -> None:
        """Additively merge saved stats rows into the live table
        (count/sum add, min/max observe).  Correct for occupied cells
        and duplicate rows, so rescale can concatenate multiple donor
        shards' spills before merging."""
        import torch

        if len(snap.get("keys", ())) == 0:
            return
        if self.cpu:
            for k, w, c, s, mn, mx in zip(
                snap["keys"].tolist(),
                snap["wins"].tolist(),
                snap["cnt"].tolist(),
                snap["sum"].tolist(),
                snap["min"].tolist(),
                snap["max"].tolist(),
            ):
                kw = (int(k), int(w))
                cur = self._table.get(kw)
                if cur is None:
                    self._table[kw] = (c, s, mn, mx)
                else:
                    self._table[kw] = (
                        cur[0] + c,
                        cur[1] + s,
                        min(cur[2], mn),
                        max(cur[3], mx),
                    )
            return
        wins = torch.as_tensor(snap["wins"]).to(torch.int64)
        ts = (wins * self.len_ms + self.align_ms).to(self.device)
        keys = torch.as_tensor(snap["keys"]).to(self.device)
        # Rebuild slots: count/sum re-add; min/max re-observe.  Use the
        # insert kernel once per stat via synthetic value columns would
        # be wrong for cnt/sum, so write slots directly through a
        # one-event-per-row insert of each stat:
        #   cnt: add (cnt-1) extra via sum of ones is wasteful; instead
        #   insert value=min (sets min), value=max (sets max), then fix
        #   cnt/sum arithmetically with a second pass.
        mn = torch.as_tensor(snap["min"]).to(self.device)
        mx = torch.as_tensor(snap["max"]).to(self.device)
        self.k.stats_insert(
            keys, ts, mn, self.tkeys, self.tcnt, self.tsum, self.tmin,
            self.tmax, self.max_ts_dev, self.error_flag,
            self.align_ms, self.len_ms, 0,
        )
        self.k.stats_insert(
            keys, ts, mx, self.tkeys, self.tcnt, self.tsum, self.tmin,
            self.tmax, self.max_ts_dev, self.error_flag,
            self.align_ms, self.len_ms, 0,
        )
        # Correct cnt/sum: current slots have cnt=2, sum=min+max; the
        # delta columns below restore the snapshotted values exactly.
        cnt_fix = torch.as_tensor(snap["cnt"]).to(self.device) - 2
        sum_fix = (
            torch.as_tensor(snap["sum"]).to(self.device)
            - mn
            - mx
        )
        self.k.stats_fixup(
            keys, ts, cnt_fix, sum_fix, self.tkeys, self.tcnt, self.tsum,
            self.align_ms, self.len_ms,
        )


class HashJoinState:
    """Two-sided stream join state on device ("last"/"complete").

    With ``radix=True`` (default) each side's batch is partitioned
    into table-region segments first, so the exchange of flags/values
    stays L2-local per workgroup instead of random across the table.
    """

    N_SIDES = 2

    def __init__(
        self,
        device,
        slots_pow: int = 20,
        out_cap: int = 1 << 20,
        radix: Optional[bool] = None,
        region_bits: int = 11,
    ):
        import os

        import torch

        if radix is None:
            # Radix default: the LDS-deduped region join measures
            # 1.6x the direct atomics at 1M keys (r02 call 29/30,
            # profiles/); BYTEWAX_JOIN_RADIX=0 restores direct.
            radix = os.environ.get("BYTEWAX_JOIN_RADIX", "1") == "1"
        self.device = device
        self.cpu = device.type == "cpu"
        if self.cpu:
            self._table: Dict[int, list] = {}
            return
        self.k = ext()
        if radix:
            slots_pow = max(slots_pow, region_bits + 10)
        self.radix = radix
        self.region_bits = region_bits
        self.nslots = 1 << slots_pow
        if radix:
            self.n_regions = self.nslots >> region_bits
            self.rx_gcursors = torch.zeros(
                self.n_regions, dtype=torch.int32, device=device
            )
            self.rx_ov_cursor = torch.zeros(1, dtype=torch.int32, device=device)
            self.rx_max_batch = 0
            self.rx_packed = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_vals = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_ov_packed = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_ov_vals = torch.empty(1, dtype=torch.int64, device=device)
            self.rx_zeros = torch.zeros(1, dtype=torch.int64, device=device)
            self.rx_maxts = torch.zeros(1, dtype=torch.int64, device=device)
        self.tkeys = torch.full(
            (self.nslots,), -1, dtype=torch.int64, device=device
        )
        self.tval0 = torch.zeros(self.nslots, dtype=torch.int64, device=device)
        self.tval1 = torch.zeros(self.nslots, dtype=torch.int64, device=device)
        self.tflags = torch.zeros(self.nslots, dtype=torch.int32, device=device)
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)
        self.out_cap = out_cap
        self.out_keys = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out_v0 = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_v1 = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_n = torch.zeros(1, dtype=torch.int32, device=device)
        self._pending = 0

    def insert(self, side: int, keys, vals) -> None:
        """Insert one side's (keys, vals) columns; completed pairs
        accumulate in the output buffer until `take_joined`."""
        if self.cpu:
            for k, v in zip(keys.tolist(), vals.tolist()):
                ent = self._table.setdefault(int(k), [None, None, 0])
                ent[side] = int(v)
                ent[2] |= 1 << side
            return
        if self.radix:
            import torch

            n = int(keys.numel())
            if n > self.rx_max_batch:
                mb = int(n * 5 // 4)
                self.rx_max_batch = mb
                per_region = -(-mb * 5 // 2) // self.n_regions + 1
                total = per_region * self.n_regions
                self.rx_packed = torch.empty(
                    total, dtype=torch.int64, device=self.device
                )
                self.rx_vals = torch.empty(
                    total, dtype=torch.int64, device=self.device
                )
                ov = max(1 << 20, mb)
                self.rx_ov_packed = torch.empty(
                    ov, dtype=torch.int64, device=self.device
                )
                self.rx_ov_vals = torch.empty(
                    ov, dtype=torch.int64, device=self.device
                )
                self.rx_zeros = torch.zeros(
                    mb, dtype=torch.int64, device=self.device
                )
            self.k.radix_join_insert(
                keys,
                self.rx_zeros,
                vals,
                side,
                self.N_SIDES,
                self.tkeys,
                self.tval0,
                self.tval1,
                self.tflags,
                self.rx_gcursors,
                self.rx_packed,
                self.rx_vals,
                self.rx_ov_cursor,
                self.rx_ov_packed,
                self.rx_ov_vals,
                self.out_keys,
                self.out_v0,
                self.out_v1,
                self.out_n,
                self.rx_maxts,
                self.error_flag,
                self.region_bits,
            )
            return
        self.k.join_insert(
            keys,
            vals,
            side,
            self.N_SIDES,
            self.tkeys,
            self.tval0,
            self.tval1,
            self.tflags,
            self.out_keys,
            self.out_v0,
            self.out_v1,
            self.out_n,
            self.error_flag,
        )

    def snapshot_to_host(self) -> Dict[str, Any]:
        """Spill live (single-side) cells to host for recovery.

        Completed pairs reset their flags on emission
        (k_join_insert), so the live state is exactly the cells with
        a nonzero side mask.  Reuses the (drained) output buffers as
        extraction scratch."""
        import numpy as np
        import torch

        if self.cpu:
            rows = [
                (k, e[0] or 0, e[1] or 0, e[2])
                for k, e in self._table.items()
                if e[2]
            ]
            return {
                "keys": np.array([r[0] for r in rows], dtype="int32"),
                "v0": np.array([r[1] for r in rows], dtype="int64"),
                "v1": np.array([r[2] for r in rows], dtype="int64"),
                "flags": np.array([r[3] for r in rows], dtype="int32"),
            }
        if not hasattr(self, "_snap_flags"):
            self._snap_flags = torch.empty(
                self.out_cap, dtype=torch.int32, device=self.device
            )
            self._snap_n = torch.zeros(
                1, dtype=torch.int32, device=self.device
            )
        self._snap_n.zero_()
        self.k.join_extract(
            self.tkeys,
            self.tval0,
            self.tval1,
            self.tflags,
            self.out_keys,
            self.out_v0,
            self.out_v1,
            self._snap_flags,
            self._snap_n,
        )
        n = int(self._snap_n.item())
        if n > self.out_cap:
            msg = f"join snapshot produced {n} rows > out_cap"
            raise RuntimeError(msg)
        return {
            "keys": self.out_keys[:n].cpu().numpy().copy(),
            "v0": self.out_v0[:n].cpu().numpy().copy(),
            "v1": self.out_v1[:n].cpu().numpy().copy(),
            "flags": self._snap_flags[:n].cpu().numpy().copy(),
        }

    def restore_from_host(self, snap: Dict[str, Any]) -> None:
        """Re-insert each spilled side presence.  Single-side cells
        cannot re-complete, so nothing re-emits."""
        import torch

        flags = snap["flags"]
        for side, col in ((0, "v0"), (1, "v1")):
            m = (flags & (1 << side)) != 0
            if m.any():
                keys = torch.as_tensor(snap["keys"][m]).to(self.device)
                vals = torch.as_tensor(snap[col][m]).to(self.device)
                self.insert(side, keys, vals)

    def take_joined(self):
        """Drain completed (key, v0, v1) rows."""
        import torch

        if self.cpu:
            out = []
            for k, ent in self._table.items():
                if ent[2] == 3:
                    out.append((k, ent[0], ent[1]))
                    ent[2] = 0
            if not out:
                return None
            return (
                torch.tensor([k for k, *_v in out], dtype=torch.int32),
                torch.tensor([v0 for _k, v0, _v1 in out], dtype=torch.int64),
                torch.tensor([v1 for *_kv, v1 in out], dtype=torch.int64),
            )
        # One D2H fetch for (count, error) — two .item() calls would
        # sync the stream twice per drained batch.
        n, err = torch.cat((self.out_n, self.error_flag)).cpu().tolist()
        if err != 0:
            msg = "join table overflowed; increase slots_pow"
            raise RuntimeError(msg)
        if n == 0:
            return None
        if n > self.out_cap:
            msg = f"join produced {n} rows > out_cap"
            raise RuntimeError(msg)
        out = (
            self.out_keys[:n].clone(),
            self.out_v0[:n].clone(),
            self.out_v1[:n].clone(),
        )
        self.out_n.zero_()
        return out


class SessionAggState:
    """Gap-based session aggregation state on device.

    One cell per key: (session_start, last_ts, accumulator).  Batches
    are sorted by (key, ts) on insert and one thread walks each key's
    segment in order (parallelism = distinct keys — built for
    high-cardinality streams; low-cardinality session streams belong
    on the host path's `SessionWindower`).  A gap > ``gap_ms`` closes
    the running session; the watermark close emits sessions idle past
    ``watermark - gap``.  Role parity: reference
    windowing.py _SessionWindowerLogic under watermark-ordered input
    (in-order ingestion makes merges degenerate to extension).
    """

    def __init__(
        self,
        device,
        gap_ms: int,
        mode: int = AGG_COUNT,
        slots_pow: int = 20,
        out_cap: int = 1 << 20,
    ):
        import torch

        self.device = device
        self.cpu = device.type == "cpu"
        self.gap_ms = gap_ms
        self.mode = mode
        self.max_ts_host = 0
        if self.cpu:
            self._table: Dict[int, Tuple[int, int, int]] = {}
            self._closed: List[Tuple[int, int, int, int]] = []
            return
        self.k = ext()
        self.nslots = 1 << slots_pow
        mk = lambda fill: torch.full(  # noqa: E731
            (self.nslots,), fill, dtype=torch.int64, device=device
        )
        self.skeys, self.sstart = mk(-1), mk(0)
        self.slast, self.sacc = mk(-1), mk(0)
        self.skeys_alt, self.sstart_alt = mk(-1), mk(0)
        self.slast_alt, self.sacc_alt = mk(-1), mk(0)
        self.out_cap = out_cap
        self.out_keys = torch.empty(out_cap, dtype=torch.int32, device=device)
        self.out_start = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_end = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_vals = torch.empty(out_cap, dtype=torch.int64, device=device)
        self.out_n = torch.zeros(1, dtype=torch.int32, device=device)
        self.max_ts_dev = torch.zeros(1, dtype=torch.int64, device=device)
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)

    def _sort(self, batch: RecordBatch):
        import torch

        order = torch.argsort(batch.ts, stable=True)
        k1 = batch.keys[order]
        order2 = torch.argsort(k1.to(torch.int64), stable=True)
        perm = order[order2]
        keys = batch.keys[perm]
        ts = batch.ts[perm].to(torch.int64) + batch.ts_base
        vals = batch.vals[perm] if batch.vals is not None else None
        return keys, ts, vals

    def _insert_cpu(self, batch: RecordBatch) -> None:
        keys, ts, vals = self._sort(batch)
        it = zip(
            keys.tolist(),
            ts.tolist(),
            vals.tolist() if vals is not None else [1] * len(keys),
        )
        for k, t, v in it:
            if self.mode == AGG_COUNT:
                v = 1
            cur = self._table.get(k)
            if cur is not None and t - cur[1] > self.gap_ms:
                self._closed.append((k, cur[0], cur[1], cur[2]))
                cur = None
            if cur is None:
                self._table[k] = (t, t, v)
            else:
                self._table[k] = (cur[0], t, cur[2] + v)
            if t > self.max_ts_host:
                self.max_ts_host = t

    def _insert_batch_fast(self, batch: RecordBatch) -> None:
        """Batch-level fast path (COUNT, gap >= batch span): no
        session boundary can fall inside the batch, so each key's
        per-batch (count, min_ts, max_ts) merges as one unit — fused
        into ONE radix pass: AGG_TS scatter partitions (key, t) pairs
        into segments, one block LDS-aggregates each segment and
        merges its distinct keys straight into the session table
        (round-2 rework: replaces the stats-table round trip and its
        five per-batch table clears)."""
        import torch

        n = len(batch)
        if not hasattr(self, "rx_gcursors"):
            # Segment count ~nslots/2^seg_bits; LDS staging in the agg
            # covers up to 2^12 distinct keys per segment.
            self.seg_bits = max(self.nslots.bit_length() - 1 - 10, 1)
            nseg = self.nslots >> self.seg_bits
            self.rx_gcursors = torch.zeros(
                nseg, dtype=torch.int32, device=self.device
            )
            self.rx_ov_cursor = torch.zeros(
                1, dtype=torch.int32, device=self.device
            )
            self.rx_max_batch = 0
            self.rx_packed = torch.empty(1, dtype=torch.int64, device=self.device)
            self.rx_vals = torch.empty(1, dtype=torch.int64, device=self.device)
            self.rx_ov_packed = torch.empty(
                1, dtype=torch.int64, device=self.device
            )
            self.rx_ov_vals = torch.empty(
                1, dtype=torch.int64, device=self.device
            )
        if n > self.rx_max_batch:
            mb = int(n * 5 // 4)
            self.rx_max_batch = mb
            nseg = self.nslots >> self.seg_bits
            per_seg = max(64, -(-mb * 5 // 2) // nseg + 1)
            total = per_seg * nseg
            self.rx_packed = torch.empty(
                total, dtype=torch.int64, device=self.device
            )
            self.rx_vals = torch.empty(
                total, dtype=torch.int64, device=self.device
            )
            ov = max(1 << 20, mb)
            self.rx_ov_packed = torch.empty(
                ov, dtype=torch.int64, device=self.device
            )
            self.rx_ov_vals = torch.empty(
                ov, dtype=torch.int64, device=self.device
            )
        self.k.session_radix_insert(
            batch.keys,
            batch.ts,
            batch.ts_base,
            self.gap_ms,
            self.skeys, self.sstart, self.slast, self.sacc,
            self.rx_gcursors, self.rx_packed, self.rx_vals,
            self.rx_ov_cursor, self.rx_ov_packed, self.rx_ov_vals,
            self.out_keys, self.out_start, self.out_end, self.out_vals,
            self.out_n, self.max_ts_dev, self.error_flag,
            self.seg_bits,
        )
        if batch.max_ts is not None and batch.max_ts > self.max_ts_host:
            self.max_ts_host = batch.max_ts

    def insert(self, batch: RecordBatch) -> None:
        import torch

        if len(batch) == 0:
            return
        if self.cpu:
            self._insert_cpu(batch)
            if (
                batch.max_ts is not None
                and batch.max_ts > self.max_ts_host
            ):
                self.max_ts_host = batch.max_ts
            return
        if (
            self.mode == AGG_COUNT
            and batch.ts_base != 0
            and batch.max_ts is not None
            and self.gap_ms >= (batch.max_ts - batch.ts_base)
        ):
            # Producers shipping a zero-based template guarantee
            # ts in [0, max_ts - ts_base]; with the gap at least that
            # span, the batch-level merge is exact.
            self._insert_batch_fast(batch)
            return
        keys, ts, vals = self._sort(batch)
        neq = keys[1:] != keys[:-1]
        idx = torch.nonzero(neq).flatten() + 1
        zero = torch.zeros(1, dtype=torch.int64, device=self.device)
        seg_start = torch.cat([zero, idx])
        n = torch.full((1,), len(keys), dtype=torch.int64, device=self.device)
        seg_end = torch.cat([idx, n])
        self.k.session_insert(
            keys, ts, vals, seg_start, seg_end,
            self.skeys, self.sstart, self.slast, self.sacc,
            self.out_keys, self.out_start, self.out_end, self.out_vals,
            self.out_n, self.max_ts_dev, self.error_flag,
            self.gap_ms, self.mode,
        )
        if batch.max_ts is not None and batch.max_ts > self.max_ts_host:
            self.max_ts_host = batch.max_ts

    def _drain_out(self) -> Optional[Dict[str, Any]]:
        if self.cpu:
            if not self._closed:
                return None
            import torch

            rows = self._closed
            self._closed = []
            return {
                "keys": torch.tensor([r[0] for r in rows], dtype=torch.int32),
                "start": torch.tensor([r[1] for r in rows], dtype=torch.int64),
                "end": torch.tensor([r[2] for r in rows], dtype=torch.int64),
                "vals": torch.tensor([r[3] for r in rows], dtype=torch.int64),
            }
        n = int(self.out_n.item())
        if int(self.error_flag.item()) != 0:
            msg = "session state overflow; increase slots_pow/out_cap"
            raise RuntimeError(msg)
        if hasattr(self, "_bs") and int(self._bs.error_flag.item()) != 0:
            msg = "session batch-stats overflow; increase slots_pow"
            raise RuntimeError(msg)
        if n == 0:
            return None
        out = {
            "keys": self.out_keys[:n].clone(),
            "start": self.out_start[:n].clone(),
            "end": self.out_end[:n].clone(),
            "vals": self.out_vals[:n].clone(),
        }
        self.out_n.zero_()
        return out

    def close_due(self, wait_ms: int = 0) -> Optional[Dict[str, Any]]:
        """Emit sessions idle past watermark - gap (plus any closed by
        gaps during inserts since the last drain)."""
        horizon = self.max_ts_host - self.gap_ms - wait_ms
        if self.cpu:
            for k in list(self._table):
                s, last, acc = self._table[k]
                if last < horizon:
                    self._closed.append((k, s, last, acc))
                    del self._table[k]
            return self._drain_out()
        self.k.session_close_migrate(
            self.skeys, self.sstart, self.slast, self.sacc,
            self.skeys_alt, self.sstart_alt, self.slast_alt, self.sacc_alt,
            self.out_keys, self.out_start, self.out_end, self.out_vals,
            self.out_n, self.error_flag, horizon,
        )
        self._swap_reset()
        return self._drain_out()

    def _swap_reset(self) -> None:
        self.skeys, self.skeys_alt = self.skeys_alt, self.skeys
        self.sstart, self.sstart_alt = self.sstart_alt, self.sstart
        self.slast, self.slast_alt = self.slast_alt, self.slast
        self.sacc, self.sacc_alt = self.sacc_alt, self.sacc
        self.skeys_alt.fill_(-1)
        self.slast_alt.fill_(-1)
        self.sstart_alt.zero_()
        self.sacc_alt.zero_()

    def close_all(self) -> Optional[Dict[str, Any]]:
        if self.cpu:
            for k in list(self._table):
                s, last, acc = self._table[k]
                self._closed.append((k, s, last, acc))
            self._table.clear()
            return self._drain_out()
        self.k.session_close_migrate(
            self.skeys, self.sstart, self.slast, self.sacc,
            self.skeys_alt, self.sstart_alt, self.slast_alt, self.sacc_alt,
            self.out_keys, self.out_start, self.out_end, self.out_vals,
            self.out_n, self.error_flag, 1 << 60,
        )
        self._swap_reset()
        return self._drain_out()

    def snapshot_to_host(self) -> Dict[str, Any]:
        import numpy as np

        if self.cpu:
            items = list(self._table.items())
            return {
                "keys": np.array([k for k, _ in items], dtype="int32"),
                "start": np.array([v[0] for _, v in items], dtype="int64"),
                "last": np.array([v[1] for _, v in items], dtype="int64"),
                "acc": np.array([v[2] for _, v in items], dtype="int64"),
                "max_ts": self.max_ts_host,
                "pending": list(self._closed),
            }
        live = (self.skeys >= 0) & (self.slast >= 0)
        # Sessions closed by gaps during insert() but not yet drained
        # sit in the out buffer; persist them (without draining — a
        # snapshot must not mutate) so a restore re-emits them exactly
        # once, mirroring the CPU twin's `_closed` list.
        pend_n = int(self.out_n.item())
        pending = [
            (int(k), int(s), int(e), int(v))
            for k, s, e, v in zip(
                self.out_keys[:pend_n].cpu().tolist(),
                self.out_start[:pend_n].cpu().tolist(),
                self.out_end[:pend_n].cpu().tolist(),
                self.out_vals[:pend_n].cpu().tolist(),
            )
        ]
        return {
            "keys": self.skeys[live].to("cpu").numpy().astype("int32"),
            "start": self.sstart[live].cpu().numpy().copy(),
            "last": self.slast[live].cpu().numpy().copy(),
            "acc": self.sacc[live].cpu().numpy().copy(),
            "max_ts": self.max_ts_host,
            "pending": pending,
        }

    def restore_from_host(self, snap: Dict[str, Any]) -> None:
        import torch

        self.max_ts_host = snap["max_ts"]
        if self.cpu:
            for k, s, last, acc in zip(
                snap["keys"].tolist(), snap["start"].tolist(),
                snap["last"].tolist(), snap["acc"].tolist(),
            ):
                self._table[int(k)] = (int(s), int(last), int(acc))
            self._closed.extend(
                tuple(r) for r in snap.get("pending", [])
            )
            return
        pending = snap.get("pending", [])
        if pending:
            # Re-stage snapshotted closed-but-undrained sessions into
            # the out buffer; the next drain re-emits them.
            n0 = int(self.out_n.item())
            n1 = n0 + len(pending)
            if n1 > self.out_cap:
                msg = "session restore overflows out_cap"
                raise RuntimeError(msg)
            self.out_keys[n0:n1] = torch.tensor(
                [r[0] for r in pending], dtype=torch.int32
            ).to(self.device)
            self.out_start[n0:n1] = torch.tensor(
                [r[1] for r in pending], dtype=torch.int64
            ).to(self.device)
            self.out_end[n0:n1] = torch.tensor(
                [r[2] for r in pending], dtype=torch.int64
            ).to(self.device)
            self.out_vals[n0:n1] = torch.tensor(
                [r[3] for r in pending], dtype=torch.int64
            ).to(self.device)
            self.out_n.fill_(n1)
        if len(snap["keys"]) == 0:
            return
        self.k.session_restore(
            torch.as_tensor(snap["keys"]).to(self.device),
            torch.as_tensor(snap["start"]).to(self.device),
            torch.as_tensor(snap["last"]).to(self.device),
            torch.as_tensor(snap["acc"]).to(self.device),
            self.skeys, self.sstart, self.slast, self.sacc,
            self.error_flag,
        )
