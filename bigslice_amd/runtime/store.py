"""Task-output stores keyed by (task name, partition).

Role-parity: exec/store.go — Store interface (:43-66), memoryStore (:70-168),
fileStore with `{prefix}/{op}/{shard}-of-{nshard}/p{partition}` layout and a
trailing 8-byte little-endian record-count footer (:173-261).  The memory
store here holds device-resident Frames directly in HBM (zero-copy within a
process); the file store keeps the reference's count-footer layout.
"""

from __future__ import annotations

import os
import struct
import threading
from typing import Dict, List, Tuple

import torch

# event-ordered executor mode (BIGSLICE_TASK_SYNC=0): cross-stream
# readers must record_stream so the allocator defers block reuse
_EVENT_MODE = os.environ.get("BIGSLICE_TASK_SYNC", "1") != "1"

_CUDA_OK = None


def _cuda_ok() -> bool:
    """torch.cuda.is_available() is UNCACHED on ROCm (a full
    hipGetDeviceCount per call, ~0.1 ms) and this path runs per frame
    read; availability cannot change after process start."""
    global _CUDA_OK
    if _CUDA_OK is None:
        _CUDA_OK = torch.cuda.is_available()
    return _CUDA_OK


from ..frame import Frame
from ..sliceio import Reader, codec


class Store:
    # True when contents survive a process restart (FileStore): gates
    # the distributed executor's per-phase checkpoint-discovery
    # collective so plain MemoryStore runs stay collective-free.
    persistent = False

    def put(self, task_name: str, partition: int, frames: List[Frame],
            rows: int) -> None:
        raise NotImplementedError

    def has(self, task_name: str, partition: int) -> bool:
        raise NotImplementedError

    def open(self, task_name: str, partition: int,
             device: str = "cpu") -> Reader:
        raise NotImplementedError

    def stat(self, task_name: str, partition: int) -> Tuple[int, int]:
        """Returns (bytes, records)."""
        raise NotImplementedError

    def discard_task(self, task_name: str) -> None:
        raise NotImplementedError


class _DiskEntry:
    """A stored partition that overflowed the host tier to disk."""

    __slots__ = ("path",)

    def __init__(self, path: str):
        self.path = path


class MemoryStore(Store):
    """In-memory partition buffers (frames stay on their device).

    Tiering: when HBM allocation passes BIGSLICE_STORE_HIGH_WATER
    (fraction of total, default 0.85) at put time, stored frames copy
    to PINNED host DRAM instead — jobs whose outputs no longer fit
    beside their inputs (e.g. a 10B-row single-GPU sort: 160 GB in +
    160 GB out > 288 GB) keep running at host-link speed instead of
    raising OOM.  Past BIGSLICE_STORE_HOST_BUDGET_BYTES of host-tiered
    frames (default 128 GB) the entry overflows to disk (CRC32 wire
    codec).  open() moves tiered frames back on demand."""

    def __init__(self):
        self._data: Dict[Tuple[str, int], object] = {}
        self._lock = threading.Lock()
        self._host_bytes = 0
        self._host_keys: Dict[Tuple[str, int], int] = {}
        self._tmpdir = None

    def _tier_mode(self, frames) -> str:
        from ..frame import over_high_water
        if not frames or not over_high_water():
            return "none"
        budget = int(os.environ.get("BIGSLICE_STORE_HOST_BUDGET_BYTES",
                                    str(128 << 30)))
        if self._host_bytes >= budget:
            return "disk"
        # also charge the process-wide accountant shared with the
        # spiller: the sum of all host tiers must fit the box's DRAM
        from ..utils import hostmem
        nb = sum(f.nbytes() for f in frames)
        return "host" if hostmem.reserve(nb) else "disk"

    def _disk_path(self, task_name, partition) -> str:
        import tempfile
        with self._lock:
            if self._tmpdir is None:
                self._tmpdir = tempfile.TemporaryDirectory(
                    prefix="bigslice-store-")
        safe = task_name.replace("/", "_")
        return os.path.join(self._tmpdir.name,
                            f"{safe}-p{partition:03d}")

    def put(self, task_name, partition, frames, rows):
        mode = self._tier_mode(frames)
        if mode == "host":
            frames = [f.to_pinned_host() if f.device != "cpu" else f
                      for f in frames]
            nb = sum(f.nbytes() for f in frames)
            with self._lock:
                self._host_bytes += nb
                self._host_keys[(task_name, partition)] = nb
        elif mode == "disk":
            path = self._disk_path(task_name, partition)
            with open(path, "wb") as fp:
                for f in frames:
                    codec.encode_frame(f.to("cpu"), fp)
            with self._lock:
                self._data[(task_name, partition)] = (
                    _DiskEntry(path), rows)
            return
        with self._lock:
            self._data[(task_name, partition)] = (frames, rows)

    def has(self, task_name, partition):
        with self._lock:
            return (task_name, partition) in self._data

    def open(self, task_name, partition, device="cpu"):
        with self._lock:
            entry = self._data.get((task_name, partition))
        if entry is None:
            raise KeyError(f"no output for {task_name} p{partition}")
        frames, _ = entry
        if isinstance(frames, _DiskEntry):
            return _DiskReader(frames.path, device)
        return _FrameListReader(frames, device)

    def stat(self, task_name, partition):
        with self._lock:
            entry = self._data.get((task_name, partition))
        if entry is None:
            return (0, 0)
        frames, rows = entry
        if isinstance(frames, _DiskEntry):
            return (os.path.getsize(frames.path), rows)
        return (sum(f.nbytes() for f in frames), rows)

    def discard_task(self, task_name):
        from ..utils import hostmem
        with self._lock:
            for k in [k for k in self._data if k[0] == task_name]:
                frames, _ = self._data.pop(k)
                nb = self._host_keys.pop(k, 0)
                if nb:
                    self._host_bytes -= nb
                    hostmem.release(nb)
                if isinstance(frames, _DiskEntry):
                    try:
                        os.unlink(frames.path)
                    except OSError:
                        pass


class _DiskReader(Reader):
    def __init__(self, path, device):
        self._fp = open(path, "rb")
        self._size = os.path.getsize(path)
        self._device = device

    def read(self):
        if self._fp.tell() >= self._size:
            return None
        return codec.decode_frame(self._fp, self._device)

    def close(self):
        self._fp.close()


class _FrameListReader(Reader):
    def __init__(self, frames, device):
        self.frames = frames
        self.device = device
        self.i = 0

    def read(self):
        if self.i >= len(self.frames):
            return None
        f = self.frames[self.i]
        self.i += 1
        if f.device != "cpu" and _EVENT_MODE and _cuda_ok():
            # cross-stream consumers: keep the caching allocator from
            # reusing these blocks until the reading stream passes
            # this point (the producer recorded its event; ordering is
            # via wait_event in the executor)
            cur = torch.cuda.current_stream()
            for c in f.columns:
                if isinstance(c, torch.Tensor) and c.is_cuda:
                    c.record_stream(cur)
        if self.device != "cpu" and f.device != self.device and \
                not f.has_objects:
            f = f.to(self.device, non_blocking=True)
        return f


class FileStore(Store):
    """File-backed store with the reference's layout and count footer
    (exec/store.go:173-261): {prefix}/{task}/p{partition} with 8-byte LE
    record count appended after the encoded stream.

    ``peers_dir`` (the checkpoint root holding every rank's store)
    enables the pull-path role of the reference's Worker.Read
    (exec/bigmachine.go:1332-1402): reads fall back to SIBLING ranks'
    directories, so after a rank loss or a restart at a different
    world size the survivors ADOPT the orphaned rank's completed
    partitions instead of recomputing them.  Writes are atomic
    (tmp+rename) and footer-validated, so only complete partitions are
    ever visible — the resumability the reference gets from byte
    offsets, provided here at partition granularity."""

    persistent = True

    def __init__(self, prefix: str, peers_dir: str = None):
        self.prefix = prefix
        self.peers_dir = peers_dir

    def _path(self, task_name: str, partition: int) -> str:
        safe = task_name.replace("/", "_")
        return os.path.join(self.prefix, safe, f"p{partition:03d}")

    def _resolve(self, task_name: str, partition: int):
        """Own path if present, else the first sibling rank's copy."""
        p = self._path(task_name, partition)
        if os.path.exists(p):
            return p
        if self.peers_dir:
            import glob
            safe = task_name.replace("/", "_")
            for cand in sorted(glob.glob(os.path.join(
                    self.peers_dir, "rank*", safe,
                    f"p{partition:03d}"))):
                return cand
        return None

    def put(self, task_name, partition, frames, rows):
        path = self._path(task_name, partition)
        os.makedirs(os.path.dirname(path), exist_ok=True)
        tmp = path + ".tmp"
        with open(tmp, "wb") as fp:
            for f in frames:
                codec.encode_frame(f.to("cpu"), fp)
            fp.write(struct.pack("<Q", rows))
        os.replace(tmp, path)

    def has(self, task_name, partition):
        return self._resolve(task_name, partition) is not None

    def has_local(self, task_name, partition):
        """Present in THIS rank's own directory (checkpoint discovery
        prefers local owners; peer adoption is the fallback)."""
        return os.path.exists(self._path(task_name, partition))

    def open(self, task_name, partition, device="cpu"):
        path = self._resolve(task_name, partition)
        if path is None:
            raise KeyError(f"no output for {task_name} p{partition}")
        size = os.path.getsize(path)
        fp = open(path, "rb")

        class _R(Reader):
            def __init__(self):
                self.remaining = size - 8  # stop before footer

            def read(self):
                if self.remaining <= 0:
                    return None
                start = fp.tell()
                f = codec.decode_frame(fp, device)
                self.remaining -= fp.tell() - start
                return f

            def close(self):
                fp.close()
        return _R()

    def stat(self, task_name, partition):
        path = self._resolve(task_name, partition)
        if path is None:
            return (0, 0)
        size = os.path.getsize(path)
        with open(path, "rb") as fp:
            fp.seek(size - 8)
            (rows,) = struct.unpack("<Q", fp.read(8))
        return (size, rows)

    def discard_task(self, task_name):
        import shutil
        d = os.path.join(self.prefix, task_name.replace("/", "_"))
        if os.path.isdir(d):
            shutil.rmtree(d, ignore_errors=True)
