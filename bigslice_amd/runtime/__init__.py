"""Execution runtime exports."""

from .compile import CompileEnv, Compiler, pipeline_slices  # noqa
from .dist import DistExecutor  # noqa
from .eval import Executor, TooManyTriesError, evaluate  # noqa
from .local import LocalExecutor, TaskLost  # noqa
from .session import (FuncValue, Invocation, Result, Session, func,  # noqa
                      func_locations, registry_digest, start)
from .store import FileStore, MemoryStore, Store  # noqa
from .task import Task, TaskDep, TaskState, graph_string  # noqa
