"""Slice operator exports."""

from .slice_base import (Dep, Name, Pragma, Slice, TaskContext,  # noqa
                         exclusive, materialize, procs, unwrap)
from .sources import Const, ReaderFunc, ScanReader  # noqa
from .elementwise import (Filter, Flatmap, Head, Map, Prefixed, Scan,  # noqa
                          WriterFunc, schema_of)
from .shuffle import Repartition, Reshard, Reshuffle  # noqa
from .reduce import Fold, Reduce  # noqa
from .cogroup import Cogroup  # noqa
from .cache import Cache, ReadCache  # noqa
from .aggregate import Aggregation  # noqa
from .archive import TarReader  # noqa
