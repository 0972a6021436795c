"""Device-resident columnar batches.

Role-parity with the reference's frame package (frame/frame.go:82-92): a Frame
is a set of equal-length typed columns with a key prefix, supporting zero-copy
slicing, copy/append, per-row hashing of the prefix, and lexicographic
ordering by prefix.  Here each numeric column is a torch tensor living in
MI355X HBM3E (or host memory on the CPU path); OBJECT columns are Python lists
on the host and only flow through CPU pipelines.

The heavy per-row "ops" of the reference (frame/ops.go: hash/less/swap/encode)
become whole-column device kernels: hashing is the murmur3 HIP kernel
(bigslice_amd/csrc), ordering is torch/rocPRIM sort, and copies are
hipMemcpyAsync under torch.
"""

from __future__ import annotations

from typing import List, Sequence, Union

import torch

from . import hashing
from .schema import OBJECT, Schema, infer_dtype, is_object


class SegmentedColumn:
    """A device-resident list-valued column: row i is
    values[starts[i]:ends[i]].  This is the MI355X representation of the
    reference's slice-valued Cogroup output columns ([]T, cogroup.go:94):
    the segments stay in HBM as one values tensor + per-row offset
    tensors, and only materialize as Python lists at scan time."""

    __slots__ = ("values", "starts", "ends")

    def __init__(self, values: torch.Tensor, starts: torch.Tensor,
                 ends: torch.Tensor):
        self.values = values
        self.starts = starts
        self.ends = ends

    def __len__(self):
        return self.starts.shape[0]

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            return SegmentedColumn(self.values, self.starts[idx],
                                   self.ends[idx])
        raise TypeError("SegmentedColumn supports slice indexing only")

    def select(self, indices: torch.Tensor) -> "SegmentedColumn":
        idx = indices.to(self.starts.device)
        return SegmentedColumn(self.values, self.starts[idx],
                               self.ends[idx])

    def to(self, device, non_blocking=False) -> "SegmentedColumn":
        return SegmentedColumn(
            self.values.to(device, non_blocking=non_blocking),
            self.starts.to(device, non_blocking=non_blocking),
            self.ends.to(device, non_blocking=non_blocking))

    def nbytes(self) -> int:
        return (self.values.numel() * self.values.element_size() +
                self.starts.numel() * 16)

    def tolists(self) -> list:
        v = self.values.cpu()
        s = self.starts.cpu().tolist()
        e = self.ends.cpu().tolist()
        return [v[a:b].tolist() for a, b in zip(s, e)]


class BytesColumn:
    """Device-resident variable-length byte rows: row i is
    data[offsets[i]:offsets[i+1]] of one uint8 tensor.  The first-class
    string/bytes column of the reference (frame/ops_builtin.go:143-164),
    MI355X-native: hashing runs the K17 murmur3 kernel, partitioning and
    shuffling move (lengths, bytes) tensor pairs, and ordering uses
    64-bit two-seed dictionary ids (strings.string_ids) — grouping is by
    id (a 2^-64 per-pair collision is accepted, like any dictionary
    encoding)."""

    __slots__ = ("data", "offsets")

    def __init__(self, data: torch.Tensor, offsets: torch.Tensor):
        self.data = data          # uint8 [total]
        self.offsets = offsets    # int64 [n+1], absolute into data

    @staticmethod
    def from_list(rows, device: str = "cpu") -> "BytesColumn":
        """Build from a list of bytes/str."""
        import numpy as np
        enc = [r.encode("utf-8") if isinstance(r, str) else bytes(r)
               for r in rows]
        offsets = np.zeros(len(enc) + 1, dtype=np.int64)
        np.cumsum(np.fromiter((len(b) for b in enc), np.int64, len(enc)),
                  out=offsets[1:])
        data = np.frombuffer(b"".join(enc), dtype=np.uint8).copy()
        return BytesColumn(
            torch.from_numpy(data).to(device),
            torch.from_numpy(offsets).to(device))

    def __len__(self):
        return self.offsets.shape[0] - 1

    def __getitem__(self, idx):
        if isinstance(idx, slice):
            start, stop, step = idx.indices(len(self))
            if step != 1:
                raise TypeError("BytesColumn slicing requires step 1")
            # zero-copy: offsets subrange stays absolute into data
            return BytesColumn(self.data, self.offsets[start:stop + 1])
        raise TypeError("BytesColumn supports slice indexing only")

    @property
    def device(self):
        return self.data.device

    def nbytes(self) -> int:
        base = int(self.offsets[0]) if len(self) else 0
        end = int(self.offsets[-1]) if len(self) else 0
        return (end - base) + self.offsets.numel() * 8

    def lengths(self) -> torch.Tensor:
        return self.offsets[1:] - self.offsets[:-1]

    def compacted(self) -> "BytesColumn":
        """Materialize a zero-based copy (drops slack from slicing)."""
        if len(self) == 0:
            return BytesColumn(
                torch.empty(0, dtype=torch.uint8,
                            device=self.data.device),
                torch.zeros(1, dtype=torch.int64,
                            device=self.offsets.device))
        base = self.offsets[0]
        data = self.data[base:self.offsets[-1]].contiguous()
        return BytesColumn(data, (self.offsets - base).contiguous())

    def select(self, indices: torch.Tensor) -> "BytesColumn":
        idx = indices.to(self.offsets.device)
        lens = self.lengths()[idx]
        n_out = idx.shape[0]
        offs = torch.zeros(n_out + 1, dtype=torch.int64,
                           device=self.offsets.device)
        torch.cumsum(lens, 0, out=offs[1:])
        total = int(offs[-1]) if n_out else 0
        if total == 0:
            return BytesColumn(
                torch.empty(0, dtype=torch.uint8,
                            device=self.data.device), offs)
        # gather: absolute source index for every output byte
        row_id = torch.repeat_interleave(
            torch.arange(n_out, device=idx.device), lens)
        within = torch.arange(total, device=idx.device) - offs[row_id]
        src = self.offsets[idx][row_id] + within
        return BytesColumn(self.data[src], offs)

    def to(self, device, non_blocking=False) -> "BytesColumn":
        c = self.compacted()
        return BytesColumn(
            c.data.to(device, non_blocking=non_blocking),
            c.offsets.to(device, non_blocking=non_blocking))

    def clone(self) -> "BytesColumn":
        c = self.compacted()
        return BytesColumn(c.data.clone(), c.offsets.clone())

    def ids64(self) -> torch.Tensor:
        """64-bit two-seed murmur3 dictionary ids per row (device
        kernel on GPU, bit-identical host fallback): the sort/group
        key for BYTES columns."""
        from . import strings
        c = self.compacted()
        if c.data.is_cuda:
            from . import kernels
            return kernels._C.hash_bytes64(
                c.data.contiguous(), c.offsets.contiguous(),
                strings._ID_SEED_HI, strings._ID_SEED_LO)
        from . import hashing
        data = c.data.numpy().tobytes()
        offs = c.offsets.tolist()
        mm = hashing.murmur3_bytes
        vals = []
        for i in range(len(c)):
            e = data[offs[i]:offs[i + 1]]
            v = (mm(e, strings._ID_SEED_HI) << 32) | \
                mm(e, strings._ID_SEED_LO)
            vals.append(v - (1 << 64) if v >= (1 << 63) else v)
        return torch.tensor(vals, dtype=torch.int64)

    def hash32(self, seed: int) -> torch.Tensor:
        """Row murmur3-32 (the K3/K17 partition hash), bit-identical to
        the host OBJECT string path so placement is reproducible."""
        c = self.compacted()
        if c.data.is_cuda:
            from . import kernels
            return kernels._C.hash_bytes(c.data.contiguous(),
                                         c.offsets.contiguous(), seed)
        from . import hashing
        data = c.data.numpy().tobytes()
        offs = c.offsets.tolist()
        import numpy as np
        out = np.empty(len(c), dtype=np.uint32)
        for i in range(len(c)):
            out[i] = hashing.murmur3_bytes(data[offs[i]:offs[i + 1]],
                                           seed)
        return torch.from_numpy(out.astype(np.int64))

    def tolists(self) -> list:
        c = self.compacted()
        data = c.data.cpu().numpy().tobytes()
        offs = c.offsets.cpu().tolist()
        return [data[offs[i]:offs[i + 1]] for i in range(len(c))]


Column = Union[torch.Tensor, list, SegmentedColumn, BytesColumn]


def _col_len(col: Column) -> int:
    if isinstance(col, torch.Tensor):
        return col.shape[0]
    return len(col)


def _col_dtype(col: Column):
    if isinstance(col, torch.Tensor):
        return col.dtype
    if isinstance(col, BytesColumn):
        from .schema import BYTES
        return BYTES
    return OBJECT


class Frame:
    """Equal-length typed columns with a key prefix.

    combined_id: non-None marks a frame whose prefix keys are UNIQUE,
    produced by one combiner instance (the id).  A keyed reduce that
    receives frames of a single combiner instance may pass them through
    without re-aggregating (the machine-combiners single-stream case).
    Transforms that preserve row identity propagate it; merges drop it.
    """

    __slots__ = ("columns", "prefix", "combined_id")

    def __init__(self, columns: Sequence[Column], prefix: int = None,
                 combined_id=None):
        self.columns: List[Column] = list(columns)
        n = None
        for c in self.columns:
            cl = _col_len(c)
            if n is None:
                n = cl
            elif cl != n:
                raise ValueError(f"ragged frame: column lengths {n} vs {cl}")
        if prefix is None:
            prefix = 1 if self.columns else 0
        self.prefix = prefix
        self.combined_id = combined_id

    # -- construction ----------------------------------------------------

    @staticmethod
    def from_lists(cols: Sequence[list], prefix: int = None,
                   schema: Schema = None, device: str = "cpu") -> "Frame":
        """Build a frame from Python lists, inferring dtypes per column."""
        out: List[Column] = []
        for i, c in enumerate(cols):
            dt = schema.dtypes[i] if schema else (
                infer_dtype(c[0]) if c else OBJECT)
            if is_object(dt):
                out.append(list(c))
            elif dt == "bytes":
                out.append(c if isinstance(c, BytesColumn)
                           else BytesColumn.from_list(c, device))
            else:
                out.append(torch.tensor(c, dtype=dt, device=device))
        if schema is not None and prefix is None:
            prefix = schema.prefix
        return Frame(out, prefix)

    @staticmethod
    def empty(schema: Schema, device: str = "cpu") -> "Frame":
        cols: List[Column] = []
        for dt in schema.dtypes:
            if is_object(dt):
                cols.append([])
            elif dt == "bytes":
                cols.append(BytesColumn(
                    torch.empty(0, dtype=torch.uint8, device=device),
                    torch.zeros(1, dtype=torch.int64, device=device)))
            else:
                cols.append(torch.empty(0, dtype=dt, device=device))
        return Frame(cols, schema.prefix)

    # -- basic properties -------------------------------------------------

    def __len__(self) -> int:
        return _col_len(self.columns[0]) if self.columns else 0

    @property
    def num_columns(self) -> int:
        return len(self.columns)

    @property
    def schema(self) -> Schema:
        return Schema([_col_dtype(c) for c in self.columns], self.prefix)

    @property
    def device(self) -> str:
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                return str(c.device)
        return "cpu"

    @property
    def has_objects(self) -> bool:
        """True when a column is host-only (Python lists); segmented
        device columns are NOT host-only."""
        return any(not isinstance(c, (torch.Tensor, SegmentedColumn,
                                      BytesColumn))
                   for c in self.columns)

    def nbytes(self) -> int:
        total = 0
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                total += c.numel() * c.element_size()
            elif isinstance(c, (SegmentedColumn, BytesColumn)):
                total += c.nbytes()
            else:
                total += sum(len(str(x)) for x in c)  # rough
        return total

    # -- slicing / copying -------------------------------------------------

    def slice(self, start: int, stop: int) -> "Frame":
        """Zero-copy row range (reference frame.Slice, frame.go:244)."""
        return Frame(
            [c[start:stop] for c in self.columns], self.prefix,
            combined_id=self.combined_id)

    def select(self, indices: torch.Tensor) -> "Frame":
        """Gather rows by index tensor (device gather kernel)."""
        cols: List[Column] = []
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                cols.append(c[indices.to(c.device)])
            elif isinstance(c, (SegmentedColumn, BytesColumn)):
                cols.append(c.select(indices))
            else:
                idx = indices.cpu().tolist()
                cols.append([c[i] for i in idx])
        return Frame(cols, self.prefix, combined_id=self.combined_id)

    def mask(self, keep: torch.Tensor) -> "Frame":
        """Filter rows by boolean mask (compaction; reference Filter's
        frame.Copy loop, slice.go:688-722, as one device kernel)."""
        cols: List[Column] = []
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                cols.append(c[keep.to(c.device)])
            elif isinstance(c, (SegmentedColumn, BytesColumn)):
                cols.append(c.select(keep.nonzero().flatten()))
            else:
                km = keep.cpu().numpy()
                cols.append([x for x, k in zip(c, km) if k])
        return Frame(cols, self.prefix, combined_id=self.combined_id)

    @staticmethod
    def concat(frames: Sequence["Frame"]) -> "Frame":
        """Concatenate frames row-wise (reference AppendFrame)."""
        frames = [f for f in frames if len(f) > 0]
        if not frames:
            raise ValueError("concat of no rows; use Frame.empty")
        first = frames[0]
        if len(frames) == 1:
            return first
        cols: List[Column] = []
        for i in range(first.num_columns):
            parts = [f.columns[i] for f in frames]
            if isinstance(parts[0], torch.Tensor):
                cols.append(torch.cat(parts))
            elif isinstance(parts[0], SegmentedColumn):
                vals, starts, ends, off = [], [], [], 0
                for p in parts:
                    vals.append(p.values)
                    starts.append(p.starts + off)
                    ends.append(p.ends + off)
                    off += p.values.shape[0]
                cols.append(SegmentedColumn(torch.cat(vals),
                                            torch.cat(starts),
                                            torch.cat(ends)))
            elif isinstance(parts[0], BytesColumn):
                datas, offs, off = [], [], 0
                for p in parts:
                    pc = p.compacted()
                    datas.append(pc.data)
                    offs.append(pc.offsets[:-1] + off)
                    off += pc.data.shape[0]
                offs.append(torch.tensor([off], dtype=torch.int64,
                                         device=datas[0].device))
                cols.append(BytesColumn(torch.cat(datas),
                                        torch.cat(offs)))
            else:
                merged: list = []
                for p in parts:
                    merged.extend(p)
                cols.append(merged)
        return Frame(cols, first.prefix)

    def to(self, device: str, non_blocking: bool = False) -> "Frame":
        cols: List[Column] = []
        for c in self.columns:
            if isinstance(c, (torch.Tensor, SegmentedColumn,
                              BytesColumn)):
                cols.append(c.to(device, non_blocking=non_blocking))
            else:
                if device != "cpu":
                    raise ValueError(
                        "object columns cannot move to device memory")
                cols.append(c)
        return Frame(cols, self.prefix, combined_id=self.combined_id)

    def to_pinned_host(self) -> "Frame":
        """Copy device columns into pinned host DRAM (fast H2D
        readback; the store's high-water tiering uses this).  Falls
        back to pageable memory when the pinned allocator is
        exhausted."""
        cols: List[Column] = []
        for c in self.columns:
            if isinstance(c, torch.Tensor) and c.is_cuda:
                try:
                    dst = torch.empty(c.shape, dtype=c.dtype,
                                      device="cpu", pin_memory=True)
                except RuntimeError:
                    dst = torch.empty(c.shape, dtype=c.dtype,
                                      device="cpu")
                dst.copy_(c)
                cols.append(dst)
            elif isinstance(c, (SegmentedColumn, BytesColumn)):
                cols.append(c.to("cpu"))
            else:
                cols.append(c)
        return Frame(cols, self.prefix, combined_id=self.combined_id)

    def clone(self) -> "Frame":
        cols = []
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                cols.append(c.clone())
            elif isinstance(c, BytesColumn):
                cols.append(c.clone())
            elif isinstance(c, SegmentedColumn):
                cols.append(SegmentedColumn(c.values.clone(),
                                            c.starts.clone(),
                                            c.ends.clone()))
            else:
                cols.append(list(c))
        return Frame(cols, self.prefix)

    def with_prefix(self, prefix: int) -> "Frame":
        if not (0 < prefix <= self.num_columns):
            raise ValueError(f"invalid prefix {prefix}")
        return Frame(self.columns, prefix)

    # -- key operations ----------------------------------------------------

    def hash(self, seed: int = 0) -> torch.Tensor:
        """32-bit murmur3 hash of the prefix columns, XOR-combined
        (bit-identical to reference frame.HashWithSeed, frame/frame.go:395-401
        + ops_builtin.go:140-164, so partition assignment is reproducible).

        Returns a uint32-valued tensor (as int64 on CPU path for numpy ease,
        uint32 on device).
        """
        return hashing.hash_columns(self.columns[: self.prefix], seed)

    def argsort_by_prefix(self) -> torch.Tensor:
        """Row permutation sorting lexicographically by prefix columns
        (reference frame.Less, frame/frame.go:375-385)."""
        n = len(self)
        cols = [c.ids64() if isinstance(c, BytesColumn) else c
                for c in self.columns[: self.prefix]]
        if any(not isinstance(c, torch.Tensor) for c in cols):
            from .hashing import custom_less_key
            keys = list(zip(*[list(c) if not isinstance(c, torch.Tensor)
                              else c.cpu().tolist() for c in cols]))
            order = sorted(range(n), key=lambda i: tuple(
                custom_less_key(v) for v in keys[i]))
            return torch.tensor(order, dtype=torch.int64)
        # Stable sorts applied from least- to most-significant column give
        # a lexicographic order.
        perm = torch.arange(n, device=cols[0].device)
        for c in reversed(cols):
            _, o = torch.sort(c[perm], stable=True)
            perm = perm[o]
        return perm

    def sort_by_prefix(self) -> "Frame":
        return self.select(self.argsort_by_prefix())

    # -- conversion ---------------------------------------------------------

    def column_lists(self) -> List[list]:
        out = []
        for c in self.columns:
            if isinstance(c, torch.Tensor):
                out.append(c.cpu().tolist())
            elif isinstance(c, (SegmentedColumn, BytesColumn)):
                out.append(c.tolists())
            else:
                out.append(list(c))
        return out

    def rows(self) -> list:
        """Rows as tuples (single-column frames yield scalars in scan())."""
        return list(zip(*self.column_lists())) if self.columns else []

    def __repr__(self):
        return (f"Frame({len(self)} rows, {self.schema}, "
                f"device={self.device})")


# -- HBM high-water check (store/writer tiering) ---------------------------

_HW_TOTAL = None
_HW_SKIP = 0  # stride-cache countdown while safely below the mark
_HW_FRAC = None  # the threshold the countdown was computed under


def over_high_water() -> bool:
    """True when HBM allocation passes BIGSLICE_STORE_HIGH_WATER
    (fraction of device total, default 0.85): accumulating writers and
    the memory store tier device frames to pinned host DRAM past this
    point so jobs larger than HBM keep running instead of OOMing.

    torch.cuda.memory_allocated() materializes the allocator's whole
    stats dict (~0.5 ms; profiled at ~16 ms/step on the flagship
    bench), so while allocation sits below 70% of the mark the check
    runs only every 16th call — tiering decisions don't need per-frame
    precision far from the boundary; near or above it every call
    re-reads."""
    global _HW_TOTAL, _HW_SKIP
    if _HW_TOTAL is None:
        # device_count/properties can fail when first called from a
        # worker thread on ROCm; LocalExecutor prewarms this from the
        # main thread, and failures leave the cache unset for retry
        try:
            _HW_TOTAL = (torch.cuda.get_device_properties(0).total_memory
                         if torch.cuda.is_available() else 0)
        except Exception:
            return False
    if _HW_TOTAL == 0:
        return False
    import os
    frac = float(os.environ.get("BIGSLICE_STORE_HIGH_WATER", "0.85"))
    global _HW_FRAC
    if _HW_SKIP > 0 and frac == _HW_FRAC:
        _HW_SKIP -= 1
        return False
    used = torch.cuda.memory_allocated()
    if used < 0.7 * frac * _HW_TOTAL:
        _HW_SKIP = 15
        _HW_FRAC = frac
    return used > frac * _HW_TOTAL
