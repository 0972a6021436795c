"""Spill tier: HBM -> pinned host DRAM -> disk.

Role-parity with the reference's disk spiller (sliceio/spiller.go:27-127),
redesigned for the MI355X memory hierarchy: a spilled device batch first moves
to pinned host DRAM over hipMemcpyAsync on a side stream (non_blocking copy),
and only overflows to disk past a host-memory budget.  Readback streams
batches back to the device asynchronously.
"""

from __future__ import annotations

import os
import tempfile
from typing import List, Optional

import torch

from ..frame import Frame
from . import codec


class _DiskBatch:
    __slots__ = ("path",)

    def __init__(self, path: str):
        self.path = path

    def load(self, device: str) -> Frame:
        with open(self.path, "rb") as fp:
            f = codec.decode_frame(fp, device)
        return f


class Spiller:
    """Accumulates spilled batches, tiering host-DRAM -> disk."""

    def __init__(self, host_budget_bytes: int = 64 << 30,
                 dir: Optional[str] = None, pin: bool = None):
        self.host_budget = host_budget_bytes
        self.host_used = 0
        self.batches: List[object] = []  # Frame (host) or _DiskBatch
        self.rows = 0
        self._dir = dir
        self._tmpdir = None
        self._pin = pin

    def _ensure_dir(self) -> str:
        if self._dir is None:
            self._tmpdir = tempfile.TemporaryDirectory(prefix="bigslice-spill-")
            self._dir = self._tmpdir.name
        return self._dir

    def spill(self, frame: Frame) -> int:
        """Spill one batch; returns bytes spilled."""
        nbytes = frame.nbytes()
        self.rows += len(frame)
        if self.host_used + nbytes <= self.host_budget:
            host = self._to_host(frame)
            self.batches.append(host)
            self.host_used += nbytes
            return nbytes
        d = self._ensure_dir()
        path = os.path.join(d, f"spill-{len(self.batches):06d}")
        with open(path, "wb") as fp:
            codec.encode_frame(frame.to("cpu"), fp)
        self.batches.append(_DiskBatch(path))
        return nbytes

    def _to_host(self, frame: Frame) -> Frame:
        if frame.device == "cpu":
            return frame
        cols = []
        pin = torch.cuda.is_available() if self._pin is None else self._pin
        for c in frame.columns:
            if isinstance(c, torch.Tensor) and c.is_cuda:
                if pin:
                    dst = torch.empty(c.shape, dtype=c.dtype, device="cpu",
                                      pin_memory=True)
                    dst.copy_(c, non_blocking=True)
                    cols.append(dst)
                else:
                    cols.append(c.cpu())
            else:
                cols.append(c)
        if frame.device != "cpu" and pin:
            torch.cuda.synchronize()
        return Frame(cols, frame.prefix)

    def num_batches(self) -> int:
        return len(self.batches)

    def readers(self, device: str = "cpu") -> List["SpillReader"]:
        """One reader per spilled batch (for k-way merging)."""
        return [SpillReader([b], device) for b in self.batches]

    def reader(self, device: str = "cpu") -> "SpillReader":
        return SpillReader(list(self.batches), device)

    def close(self) -> None:
        self.batches.clear()
        if self._tmpdir is not None:
            self._tmpdir.cleanup()
            self._tmpdir = None


class SpillReader:
    def __init__(self, batches: List[object], device: str):
        self.batches = batches
        self.device = device
        self.i = 0

    def read(self) -> Optional[Frame]:
        if self.i >= len(self.batches):
            return None
        b = self.batches[self.i]
        self.i += 1
        if isinstance(b, _DiskBatch):
            return b.load(self.device)
        f: Frame = b
        if self.device != "cpu":
            return f.to(self.device, non_blocking=True)
        return f

    def close(self) -> None:
        pass

    def __iter__(self):
        while True:
            f = self.read()
            if f is None:
                return
            yield f
