"""Vectorized record streams of Frames.

Role-parity with the reference's sliceio package (sliceio/reader.go:33-56:
``Read(ctx, Frame) (n, error)`` with sentinel EOF).  The MI355X-native
protocol is batch-granular: ``Reader.read() -> Frame | None`` where None is
EOF and each Frame is a device (or host) batch of up to the configured chunk
rows.  Batches are produced/consumed on HIP streams; the codec and spiller
(wire format + host-DRAM tier) live in .codec and .spiller.
"""

from __future__ import annotations

from typing import Callable, Iterable, List, Optional

from ..frame import Frame
from ..schema import Schema

from .codec import encode_frame, decode_frame  # noqa: F401
from .spiller import Spiller  # noqa: F401


class Reader:
    """A stream of Frames.  read() returns None at EOF.

    Readers are single-use.  close() releases resources early.
    """

    def read(self) -> Optional[Frame]:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def __iter__(self):
        while True:
            f = self.read()
            if f is None:
                return
            yield f


class FrameReader(Reader):
    """Stream a single frame in chunks (reference sliceio.FrameReader)."""

    def __init__(self, frame: Frame, chunk: int = None):
        self.frame = frame
        self.chunk = chunk or len(frame) or 1
        self.off = 0

    def read(self) -> Optional[Frame]:
        if self.off >= len(self.frame):
            return None
        out = self.frame.slice(self.off, min(self.off + self.chunk,
                                             len(self.frame)))
        self.off += len(out)
        return out


class IterReader(Reader):
    """Adapt an iterator/generator of Frames."""

    def __init__(self, it: Iterable[Frame]):
        self.it = iter(it)

    def read(self) -> Optional[Frame]:
        try:
            return next(self.it)
        except StopIteration:
            return None


class FuncReader(Reader):
    """Adapt a nullary callable returning Frame|None."""

    def __init__(self, fn: Callable[[], Optional[Frame]]):
        self.fn = fn

    def read(self) -> Optional[Frame]:
        return self.fn()


class MultiReader(Reader):
    """Concatenation of readers (reference sliceio.MultiReader)."""

    def __init__(self, readers: List[Reader]):
        self.readers = list(readers)
        self.i = 0

    def read(self) -> Optional[Frame]:
        while self.i < len(self.readers):
            f = self.readers[self.i].read()
            if f is not None:
                return f
            self.readers[self.i].close()
            self.i += 1
        return None

    def close(self) -> None:
        for r in self.readers[self.i:]:
            r.close()


class EmptyReader(Reader):
    def read(self) -> Optional[Frame]:
        return None


class ErrReader(Reader):
    """Reader that raises a stored error (reference sliceio.ErrReader)."""

    def __init__(self, err: Exception):
        self.err = err

    def read(self) -> Optional[Frame]:
        raise self.err


def read_all(reader: Reader) -> Optional[Frame]:
    """Drain a reader into one frame (reference sliceio.ReadAll).
    Returns None if the stream was empty."""
    frames = [f for f in reader]
    if not frames:
        return None
    return Frame.concat(frames)


def read_all_or_empty(reader: Reader, schema: Schema,
                      device: str = "cpu") -> Frame:
    f = read_all(reader)
    return f if f is not None else Frame.empty(schema, device)


class Scanner:
    """Row-oriented consumption of a frame stream (reference
    sliceio.Scanner, scanner.go:27-141)."""

    def __init__(self, reader: Reader):
        self.reader = reader

    def rows(self):
        """Iterate rows as tuples (or scalars for 1-column streams)."""
        for f in self.reader:
            cols = f.column_lists()
            if len(cols) == 1:
                for v in cols[0]:
                    yield v
            else:
                for row in zip(*cols):
                    yield row

    def frames(self):
        return iter(self.reader)
