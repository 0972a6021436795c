"""bigslice_amd: an MI355X-native data-parallel slice engine.

A from-scratch reimplementation of the capabilities of grailbio/bigslice
(serverless cluster data processing) redesigned for one node of 8x AMD
Instinct MI355X: columnar batches live in HBM3E as torch tensors, the hot
compute primitives (hash, partition scatter, hash-aggregate, sort) are
hand-written HIP/CDNA4 kernels, and the shuffle between GPUs is an RCCL
all-to-all over xGMI (one process per GPU via torch.distributed).

Public API parity map (reference -> here):
    bigslice.Func            -> bigslice_amd.func
    exec.Start / Session.Run -> bigslice_amd.start / Session.run
    Const/ReaderFunc/Map/... -> same names, vectorized UDF convention
    slicetest.Run/...        -> bigslice_amd.slicetest
"""

__version__ = "0.1.0"

from .schema import OBJECT, Schema  # noqa
from .frame import Frame  # noqa
from .ops import (Aggregation, Cache, Cogroup, Const, Dep, Filter,  # noqa
                  Flatmap, Fold, Head, Map, Pragma, Prefixed, ReadCache,
                  ReaderFunc, Reduce, Repartition, Reshard, Reshuffle, Scan,
                  ScanReader, Slice, TarReader, WriterFunc, exclusive,
                  materialize,
                  procs, schema_of, unwrap)
from . import slicetest  # noqa
from . import sliceconfig  # noqa
from . import sortio  # noqa
from .utils import metrics  # noqa
from .runtime import (Result, Session, func, registry_digest, start)  # noqa
from . import sliceio  # noqa
from . import config  # noqa
from . import strings  # noqa  (K17 device string hashing / dict ids)
