"""Spill tier: HBM -> pinned host DRAM -> disk.

Role-parity with the reference's disk spiller (sliceio/spiller.go:27-127),
redesigned for the MI355X memory hierarchy: a spilled device batch moves
to pinned host DRAM via hipMemcpyAsync on a dedicated copy stream that
overlaps with compute on the task's stream (events order the handoff);
past a host-memory budget batches overflow to disk.  Readback streams
batches back to the device asynchronously.
"""

from __future__ import annotations

import os
import tempfile
import threading
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


class _HostBatch:
    """A host-resident batch whose D2H copy may still be in flight."""

    __slots__ = ("frame", "event", "src_refs", "src_bytes", "owner")

    def __init__(self, frame: Frame, event, src_refs, src_bytes=0,
                 owner=None):
        self.frame = frame
        self.event = event  # torch.cuda.Event or None
        self.src_refs = src_refs  # keep device tensors alive until done
        self.src_bytes = src_bytes
        self.owner = owner  # Spiller outstanding-bytes accounting

    def _release(self):
        if self.src_refs is not None:
            if self.src_bytes:
                with _outstanding_lock:
                    _outstanding[0] -= self.src_bytes
            self.src_refs = None

    def ready(self, cpu_access: bool = False):
        if self.event is not None:
            if cpu_access:
                # CPU will read the pinned buffers directly
                self.event.synchronize()
            else:
                # device consumption: stream ordering suffices
                torch.cuda.current_stream().wait_event(self.event)
            if self.event.query():
                self._release()
        return self.frame


_copy_streams = {}
_copy_lock = threading.Lock()

# GLOBAL un-drained D2H backlog (device bytes alive only to feed
# in-flight spill copies): concurrent tasks' spillers share the HBM,
# so the backpressure cap must be process-wide.
_outstanding = [0]
_outstanding_lock = threading.Lock()


def _copy_stream(device, direction: str = "d2h"):
    """Per-(thread, direction) copy streams: concurrent tasks' spills
    (D2H) and readbacks (H2D) must not serialize on one stream."""
    key = (device, direction, threading.get_ident())
    with _copy_lock:
        s = _copy_streams.get(key)
        if s is None:
            s = torch.cuda.Stream(device=device)
            _copy_streams[key] = s
        return s


class Spiller:
    """Accumulates spilled batches, tiering host-DRAM -> disk."""

    def __init__(self, host_budget_bytes: int = 64 << 30,
                 dir: Optional[str] = None, pin: bool = None):
        self.host_budget = host_budget_bytes
        self.host_used = 0
        self._accounted = 0  # bytes charged to the global accountant
        self.batches: List[object] = []  # _HostBatch | Frame | _DiskBatch
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
        """Spill one batch; returns bytes spilled.  Device batches copy
        out asynchronously on the copy stream; past
        SPILL_BACKPRESSURE_BYTES of un-drained device memory the caller
        blocks on the oldest copy (producers outrun the host link on
        large jobs and the backlog would OOM HBM).  Host placement is
        charged against BOTH this spiller's budget and the process-wide
        accountant (utils.hostmem) shared with the store tier; either
        refusal sends the batch to the unbounded disk tier."""
        self._release_completed()
        from .. import config
        from ..utils import hostmem
        while _outstanding[0] > config.SPILL_BACKPRESSURE_BYTES:
            self._wait_oldest()
            self._release_completed()
        nbytes = frame.nbytes()
        self.rows += len(frame)
        if self.host_used + nbytes <= self.host_budget and \
                hostmem.reserve(nbytes):
            self.batches.append(self._to_host(frame))
            self.host_used += nbytes
            self._accounted += nbytes
            return nbytes
        d = self._ensure_dir()
        path = os.path.join(d, f"spill-{len(self.batches):06d}")
        with open(path, "wb") as fp:
            codec.encode_frame(frame.to("cpu"), fp)
        self.batches.append(_DiskBatch(path))
        return nbytes

    def _to_host(self, frame: Frame):
        if frame.device == "cpu":
            return frame
        if self._pin is None:
            from ..runtime.store import _cuda_ok
            self._pin = _cuda_ok()
        pin = self._pin
        if not pin:
            return frame.to("cpu")
        device = frame.device
        cs = _copy_stream(device, "d2h")
        # the copy stream must see the producer's writes
        ready = torch.cuda.Event()
        ready.record()
        cs.wait_event(ready)
        cols = []
        src_refs = []
        src_bytes = 0
        with torch.cuda.stream(cs):
            for c in frame.columns:
                if isinstance(c, torch.Tensor) and c.is_cuda:
                    try:
                        dst = torch.empty(c.shape, dtype=c.dtype,
                                          device="cpu", pin_memory=True)
                    except RuntimeError:  # pinned allocator exhausted
                        dst = torch.empty(c.shape, dtype=c.dtype,
                                          device="cpu")
                    dst.copy_(c, non_blocking=True)
                    cols.append(dst)
                    src_refs.append(c)
                    src_bytes += c.numel() * c.element_size()
                else:
                    cols.append(c)
            done = torch.cuda.Event()
            done.record()
        with _outstanding_lock:
            _outstanding[0] += src_bytes
        return _HostBatch(Frame(cols, frame.prefix), done, src_refs,
                          src_bytes, self)

    def _release_completed(self) -> None:
        """Drop device-source references of finished D2H copies so
        spilling actually relieves HBM pressure during the build phase
        (not only at readback)."""
        for b in self.batches:
            if isinstance(b, _HostBatch) and b.src_refs is not None:
                if b.event is None or b.event.query():
                    b._release()

    def _wait_oldest(self) -> None:
        for b in self.batches:
            if isinstance(b, _HostBatch) and b.src_refs is not None \
                    and b.event is not None:
                b.event.synchronize()
                return

    def num_batches(self) -> int:
        return len(self.batches)

    def readers(self, device: str = "cpu") -> List["SpillReader"]:
        """One reader per spilled batch (for k-way merging)."""
        return [SpillReader([b], device) for b in self.batches]

    def reader(self, device: str = "cpu") -> "SpillReader":
        return SpillReader(list(self.batches), device)

    def close(self) -> None:
        self.batches.clear()
        if self._accounted:
            from ..utils import hostmem
            hostmem.release(self._accounted)
            self._accounted = 0
        if self._tmpdir is not None:
            self._tmpdir.cleanup()
            self._tmpdir = None


def load_batch(b, device: str) -> Frame:
    if isinstance(b, _DiskBatch):
        return b.load(device)
    if isinstance(b, _HostBatch):
        f = b.ready(cpu_access=(device == "cpu"))
    else:
        f = b
    if device != "cpu" and f.device == "cpu":
        return f.to(device, non_blocking=True)
    return f


class SpillReader:
    """Reads spilled batches back, prefetching the NEXT batch's H2D
    copy on the copy stream while the caller consumes the current one
    (keeps the host link busy through a k-way merge)."""

    def __init__(self, batches: List[object], device: str):
        self.batches = batches
        self.device = device
        self.i = 0
        self._pre = None  # (frame_on_device, ready_event)

    def _start_prefetch(self):
        if self.i >= len(self.batches) or self.device == "cpu":
            self._pre = None
            return
        b = self.batches[self.i]
        self.i += 1
        if isinstance(b, _DiskBatch):
            # disk loads are synchronous; no async prefetch
            self._pre = (b.load(self.device), None)
            return
        cs = _copy_stream(self.device, "h2d")
        with torch.cuda.stream(cs):
            # order the H2D read after the batch's D2H write (waits on
            # cs, the stream doing the read)
            f = b.ready(cpu_access=False) if isinstance(b, _HostBatch) \
                else b
            dev = f.to(self.device, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
        self._pre = (dev, ev)

    def read(self) -> Optional[Frame]:
        if self.device == "cpu":
            if self.i >= len(self.batches):
                return None
            b = self.batches[self.i]
            self.i += 1
            return load_batch(b, self.device)
        if self._pre is None:
            if self.i >= len(self.batches):
                return None
            self._start_prefetch()
            if self._pre is None:
                return None
        f, ev = self._pre
        self._pre = None
        self._start_prefetch()  # overlap next H2D with consumption
        if ev is not None:
            torch.cuda.current_stream().wait_event(ev)
        return f

    def close(self) -> None:
        pass

    def __iter__(self):
        while True:
            f = self.read()
            if f is None:
                return
            yield f
