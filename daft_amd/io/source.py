"""Custom python DataSource / DataSourceTask connector API (ref:
/root/reference/daft/io/source.py — DataSource splits into independently
readable tasks, each yielding RecordBatches with a shared schema).

Here a task's `read()` is a plain (synchronous) iterator of RecordBatch;
`DataSource.read()` assembles the task outputs into partitions of a
DataFrame.  Pushdown hints (columns / limit) are offered to the source
via `get_tasks(pushdowns)`; a source may ignore them (the engine still
applies them afterwards).
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass, field
from typing import Iterator, List, Optional

from ..recordbatch import RecordBatch
from ..schema import Schema


@dataclass
class Pushdowns:
    """Scan pushdown hints (ref: daft-scan Pushdowns)."""
    columns: Optional[List[str]] = None
    limit: Optional[int] = None
    filters: Optional[list] = field(default=None)


class DataSourceTask(ABC):
    """One independently readable partition of a DataSource."""

    @property
    @abstractmethod
    def schema(self) -> Schema:
        ...

    @abstractmethod
    def read(self) -> Iterator[RecordBatch]:
        """Yield this task's record batches."""
        ...


class DataSource(ABC):
    """Low-level interface for reading custom data into DataFrames."""

    @property
    @abstractmethod
    def name(self) -> str:
        ...

    @property
    @abstractmethod
    def schema(self) -> Schema:
        ...

    @abstractmethod
    def get_tasks(self, pushdowns: Optional[Pushdowns] = None
                  ) -> Iterator[DataSourceTask]:
        ...

    def read(self) -> "DataFrame":  # noqa: F821
        """Materialize every task into one DataFrame (task order is
        preserved as partition order)."""
        from . import from_recordbatches
        from ..series import Series
        batches: List[RecordBatch] = []
        for task in self.get_tasks(Pushdowns()):
            for rb in task.read():
                batches.append(rb)
        if not batches:
            cols = [Series.from_pylist(f.name, [], f.dtype)
                    for f in self.schema]
            batches = [RecordBatch(cols, num_rows=0)]
        return from_recordbatches(batches)


def read_source(source: DataSource) -> "DataFrame":  # noqa: F821
    """daft.read_source(my_source) (ref: daft/io read_source)."""
    return source.read()
