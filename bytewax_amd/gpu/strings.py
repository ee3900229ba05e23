"""Device string dictionary: str keys -> dense int32 ids.

The reference's key contract is `str` (reference src/operators.rs:
363-439 `extract_key`); the columnar fast path keys RecordBatches by
int32 id.  :class:`StringDict` bridges them: string bytes are shipped
to the device once per batch and hashed/deduplicated by the
`dict_encode` HIP kernels into dense ids; the host keeps the
authoritative id -> string list (new ids come back as
``(id, batch_index)`` pairs, so string bytes never travel D2H).

Key identity is 128-bit hash equality (FNV-1a finalized twice with
independent seeds).  At 10^6 distinct keys the collision probability
is ~1e-26; table anomalies (full table, unpublished slot) fail loudly
via the kernel error flag instead of aggregating wrong.
"""

from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np

from ._ext import ext

StrBatch = Union[Sequence[str], Tuple[np.ndarray, np.ndarray]]


def pack_strings(strings: Sequence[str]) -> Tuple[np.ndarray, np.ndarray]:
    """Pack python strings into (uint8 bytes, int64 offsets[n+1])."""
    bs = [s.encode() for s in strings]
    offs = np.zeros(len(bs) + 1, dtype=np.int64)
    np.cumsum([len(b) for b in bs], out=offs[1:])
    data = np.frombuffer(b"".join(bs), dtype=np.uint8).copy()
    return data, offs


def str_owner_cpu(b: bytes, world: int) -> int:
    """CPU twin of the device content-hash owner (same FNV/mix
    pipeline as `str_hash128`, so CPU tests predict device routing)."""
    h = 0xCBF29CE484222325
    for c in b:
        h = ((h ^ c) * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    hi = _mix64((h + 0x2545F4914F6CDD1D) & 0xFFFFFFFFFFFFFFFF)
    return hi % world


def _mix64(x: int) -> int:
    m = 0xFFFFFFFFFFFFFFFF
    x = (x ^ (x >> 33)) & m
    x = (x * 0xFF51AFD7ED558CCD) & m
    x = (x ^ (x >> 33)) & m
    x = (x * 0xC4CEB9FE1A85EC53) & m
    return (x ^ (x >> 33)) & m


def exchange_str_by_key(
    data, offs, ts, vals=None, group=None, force: bool = False
):
    """Exchange a str-keyed batch so every string's events land on
    the rank that owns the string by CONTENT hash: the multi-GPU
    str-keyed design (dictionary ids are per-rank, so raw bytes are
    exchanged and encoded at the owner — never ids).

    Inputs are device tensors (uint8 bytes, int64 offs[n+1], int64
    ts[n], optional int64 vals[n]) or numpy/CPU tensors for the gloo
    twin.  Returns `(data, offs, ts, vals)` in the same form.
    Collective: every rank must call this once per scheduling step.
    """
    import torch
    import torch.distributed as dist

    from ._ext import ext as _ext_mod

    world = dist.get_world_size(group)
    if world == 1 and not force:
        return data, offs, ts, vals
    dev = (
        data.device
        if isinstance(data, torch.Tensor)
        else torch.device("cpu")
    )
    if dev.type == "cpu":
        # CPU twin: python bucketing + gloo all-to-all.
        data_np = np.asarray(data, dtype=np.uint8)
        offs_np = np.asarray(offs, dtype=np.int64)
        ts_np = np.asarray(ts.numpy() if hasattr(ts, "numpy") else ts)
        vals_np = None
        if vals is not None:
            vals_np = np.asarray(
                vals.numpy() if hasattr(vals, "numpy") else vals
            )
        n = len(offs_np) - 1
        buckets = [[] for _ in range(world)]
        for i in range(n):
            b = bytes(data_np[offs_np[i] : offs_np[i + 1]])
            buckets[str_owner_cpu(b, world)].append(i)
        send_lens, send_ts, send_vals, send_bytes = [], [], [], []
        counts = []
        for idxs in buckets:
            counts.append(len(idxs))
            for i in idxs:
                send_lens.append(int(offs_np[i + 1] - offs_np[i]))
                send_ts.append(int(ts_np[i]))
                if vals_np is not None:
                    send_vals.append(int(vals_np[i]))
                send_bytes.append(
                    data_np[offs_np[i] : offs_np[i + 1]].tobytes()
                )
        byte_counts = []
        k = 0
        for c in counts:
            byte_counts.append(sum(send_lens[k : k + c]))
            k += c
        t_counts = torch.tensor(counts, dtype=torch.int32)
        t_bcounts = torch.tensor(byte_counts, dtype=torch.int64)
        t_lens = torch.tensor(send_lens, dtype=torch.int32)
        t_ts = torch.tensor(send_ts, dtype=torch.int64)
        t_vals = (
            torch.tensor(send_vals, dtype=torch.int64)
            if vals_np is not None
            else None
        )
        joined = b"".join(send_bytes)
        t_bytes = (
            torch.frombuffer(bytearray(joined), dtype=torch.uint8)
            if joined
            else torch.empty(0, dtype=torch.uint8)
        )
    else:
        k = _ext_mod()
        n = int(offs.numel()) - 1
        t_counts = torch.zeros(world, dtype=torch.int32, device=dev)
        t_bcounts = torch.zeros(world, dtype=torch.int64, device=dev)
        t_lens = torch.empty(max(n, 1), dtype=torch.int32, device=dev)
        t_ts = torch.empty(max(n, 1), dtype=torch.int64, device=dev)
        t_vals = (
            torch.empty(max(n, 1), dtype=torch.int64, device=dev)
            if vals is not None
            else None
        )
        t_bytes = torch.empty(
            max(int(data.numel()), 1), dtype=torch.uint8, device=dev
        )
        k.str_exchange_pack(
            data, offs, ts, vals, world, t_counts, t_bcounts,
            t_lens,
            t_ts,
            t_vals if t_vals is not None
            else torch.empty(0, dtype=torch.int64, device=dev),
            t_bytes,
        )
        t_lens = t_lens[:n]
        t_ts = t_ts[:n]
        if t_vals is not None:
            t_vals = t_vals[:n]
        t_bytes = t_bytes[: int(data.numel())]

    # Control plane: exchange split sizes (strings and bytes).
    in_splits = t_counts.tolist()
    in_bsplits = (
        t_bcounts.tolist()
        if dev.type == "cpu"
        else t_bcounts.cpu().tolist()
    )
    recv_counts = torch.empty_like(t_counts)
    dist.all_to_all_single(recv_counts, t_counts.contiguous(), group=group)
    bc = t_bcounts.contiguous()
    recv_bcounts = torch.empty_like(bc)
    dist.all_to_all_single(recv_bcounts, bc, group=group)
    out_splits = recv_counts.tolist()
    out_bsplits = (
        recv_bcounts.tolist()
        if dev.type == "cpu"
        else recv_bcounts.cpu().tolist()
    )
    m = int(sum(out_splits))
    mb = int(sum(out_bsplits))

    recv_lens = torch.empty(m, dtype=torch.int32, device=dev)
    recv_ts = torch.empty(m, dtype=torch.int64, device=dev)
    recv_bytes = torch.empty(mb, dtype=torch.uint8, device=dev)
    dist.all_to_all_single(
        recv_lens, t_lens.contiguous(), out_splits, in_splits, group=group
    )
    dist.all_to_all_single(
        recv_ts, t_ts.contiguous(), out_splits, in_splits, group=group
    )
    dist.all_to_all_single(
        recv_bytes, t_bytes.contiguous(), out_bsplits, in_bsplits,
        group=group,
    )
    recv_vals = None
    if t_vals is not None:
        recv_vals = torch.empty(m, dtype=torch.int64, device=dev)
        dist.all_to_all_single(
            recv_vals, t_vals.contiguous(), out_splits, in_splits,
            group=group,
        )
    lens64 = recv_lens.to(torch.int64)
    recv_offs = torch.zeros(m + 1, dtype=torch.int64, device=dev)
    recv_offs[1:] = torch.cumsum(lens64, 0)
    return recv_bytes, recv_offs, recv_ts, recv_vals


class StringDict:
    """Keyed-stream string dictionary with a device-resident table.

    ``encode`` maps a batch of strings to device int32 ids (creating
    ids for unseen strings); ``decode`` maps ids back through the
    host-side list.  ``snapshot``/``restore`` preserve the exact
    id assignment across restarts (ids are re-pinned on restore).
    """

    def __init__(self, device, slots_pow: int = 21, new_cap: int = 1 << 20):
        import torch

        self.device = device
        self.cpu = device.type == "cpu"
        self.id2str: List[str] = []
        if self.cpu:
            self._map: Dict[str, int] = {}
            return
        self.k = ext()
        nslots = 1 << slots_pow
        self.dlo = torch.full((nslots,), -1, dtype=torch.int64, device=device)
        self.dhi = torch.zeros(nslots, dtype=torch.int64, device=device)
        self.dids = torch.zeros(nslots, dtype=torch.int32, device=device)
        self.counter = torch.zeros(1, dtype=torch.int32, device=device)
        self.new_cap = new_cap
        self.new_ids = torch.empty(new_cap, dtype=torch.int32, device=device)
        self.new_idx = torch.empty(new_cap, dtype=torch.int32, device=device)
        self.new_n = torch.zeros(1, dtype=torch.int32, device=device)
        self.error_flag = torch.zeros(1, dtype=torch.int32, device=device)
        # Optional pinned staging for the per-batch H2D of string
        # bytes (BYTEWAX_STR_PINNED=1).  Measured SLOWER than the
        # direct pageable copy on MI355X (the extra serial host
        # memcpy into the pinned buffer outweighs the DMA gain; see
        # profiles/r02_kernel_stats.md), so off by default.
        import os

        self._pinned = os.environ.get("BYTEWAX_STR_PINNED", "0") == "1"
        self._pin_bytes = None
        self._pin_offs = None

    def __len__(self) -> int:
        return len(self.id2str)

    @staticmethod
    def _as_packed(batch: StrBatch) -> Tuple[np.ndarray, np.ndarray, Optional[Sequence[str]]]:
        if isinstance(batch, tuple) and len(batch) == 2:
            data, offs = batch
            return np.asarray(data, dtype=np.uint8), np.asarray(
                offs, dtype=np.int64
            ), None
        return (*pack_strings(batch), batch)

    def _strings_of(
        self, data: np.ndarray, offs: np.ndarray, idx: int,
        strings: Optional[Sequence[str]],
    ) -> str:
        if strings is not None:
            return strings[idx]
        return bytes(data[offs[idx] : offs[idx + 1]]).decode()

    def encode(self, batch: StrBatch):
        """Encode a batch of strings -> int32 id tensor on the device
        (CPU twin returns a CPU tensor).

        Also accepts a ``(uint8 bytes, int64 offs)`` pair of DEVICE
        tensors (e.g. straight from :func:`exchange_str_by_key`): the
        kernels consume them in place and host bytes are fetched only
        when the batch introduced new ids.
        """
        import torch

        if (
            not self.cpu
            and isinstance(batch, tuple)
            and len(batch) == 2
            and isinstance(batch[0], torch.Tensor)
            and batch[0].device.type != "cpu"
        ):
            return self._encode_dev(batch[0], batch[1])
        data, offs, strings = self._as_packed(batch)
        n = len(offs) - 1
        if self.cpu:
            ids = np.empty(n, dtype=np.int32)
            for i in range(n):
                s = self._strings_of(data, offs, i, strings)
                j = self._map.get(s)
                if j is None:
                    j = len(self.id2str)
                    self._map[s] = j
                    self.id2str.append(s)
                ids[i] = j
            return torch.from_numpy(ids)
        d_bytes, d_offs = self._stage(data, offs)
        return self._run_encode(d_bytes, d_offs, n, data, offs, strings)

    def _encode_dev(self, d_bytes, d_offs):
        n = int(d_offs.numel()) - 1
        return self._run_encode(d_bytes, d_offs, n, None, None, None)

    def _run_encode(self, d_bytes, d_offs, n, data, offs, strings):
        import torch

        out_ids = torch.empty(n, dtype=torch.int32, device=self.device)
        if n == 0:
            return out_ids
        self.k.dict_encode(
            d_bytes, d_offs, self.dlo, self.dhi, self.dids, self.counter,
            self.new_ids, self.new_idx, self.new_n, out_ids,
            self.error_flag,
        )
        n_new = int(self.new_n.item())
        if int(self.error_flag.item()) != 0:
            code = int(self.error_flag.item())
            msg = {
                1: "string dictionary table full; increase slots_pow",
                2: "string dictionary lookup failed (128-bit hash "
                   "collision or table anomaly)",
                3: "too many new keys in one batch; increase new_cap",
            }.get(code, f"string dictionary error {code}")
            raise RuntimeError(msg)
        if n_new:
            if data is None:
                # Device-tensor input: fetch host bytes only when the
                # batch introduced new ids.
                data = d_bytes.cpu().numpy()
                offs = d_offs.cpu().numpy()
            ids_h = self.new_ids[:n_new].cpu().numpy()
            idx_h = self.new_idx[:n_new].cpu().numpy()
            need = len(self.id2str) + n_new
            self.id2str.extend([""] * n_new)
            for i, bi in zip(ids_h.tolist(), idx_h.tolist()):
                if i >= need:
                    msg = "string dictionary id out of range"
                    raise RuntimeError(msg)
                self.id2str[i] = self._strings_of(data, offs, bi, strings)
        return out_ids

    def _stage(self, data: np.ndarray, offs: np.ndarray):
        """Ship packed bytes/offsets to the device: direct pageable
        copies by default; grow-on-demand pinned staging under
        BYTEWAX_STR_PINNED=1 (async DMA on the current stream, so the
        following kernel launch orders after it)."""
        import torch

        if not self._pinned:
            return (
                torch.from_numpy(data).to(self.device),
                torch.from_numpy(offs).to(self.device),
            )
        nb, no = data.shape[0], offs.shape[0]
        if self._pin_bytes is None or self._pin_bytes.numel() < nb:
            cap = max(1 << 16, 1 << max(0, nb - 1).bit_length())
            self._pin_bytes = torch.empty(
                cap, dtype=torch.uint8, pin_memory=True
            )
        if self._pin_offs is None or self._pin_offs.numel() < no:
            cap = max(1 << 10, 1 << max(0, no - 1).bit_length())
            self._pin_offs = torch.empty(
                cap, dtype=torch.int64, pin_memory=True
            )
        if nb:
            self._pin_bytes[:nb].copy_(torch.from_numpy(data))
        self._pin_offs[:no].copy_(torch.from_numpy(offs))
        d_bytes = self._pin_bytes[:nb].to(self.device, non_blocking=True)
        d_offs = self._pin_offs[:no].to(self.device, non_blocking=True)
        return d_bytes, d_offs

    def decode(self, ids) -> List[str]:
        """Map an id tensor (any device) back to strings."""
        arr = ids.cpu().numpy() if hasattr(ids, "cpu") else np.asarray(ids)
        return [self.id2str[int(i)] for i in arr]

    def snapshot(self) -> Dict[str, Any]:
        return {"strings": list(self.id2str)}

    def restore(self, snap: Dict[str, Any]) -> None:
        import torch

        strings = snap["strings"]
        if not strings:
            return
        if self.id2str:
            msg = "restore() requires an empty dictionary"
            raise RuntimeError(msg)
        self.id2str = list(strings)
        if self.cpu:
            self._map = {s: i for i, s in enumerate(strings)}
            return
        data, offs = pack_strings(strings)
        ids = np.arange(len(strings), dtype=np.int32)
        d_bytes, d_offs = self._stage(data, offs)
        self.k.dict_restore(
            d_bytes,
            d_offs,
            torch.from_numpy(ids).to(self.device),
            self.dlo, self.dhi, self.dids, self.error_flag,
        )
        self.counter.fill_(len(strings))
        if int(self.error_flag.item()) != 0:
            msg = "string dictionary restore failed (table full?)"
            raise RuntimeError(msg)
