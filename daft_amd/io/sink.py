"""Custom python DataSink connector API (ref:
/root/reference/daft/io/sink.py): start() once, write() per batch
(streamed), finalize(results) -> summary RecordBatch.  Consumed by
DataFrame.write_sink."""
from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass
from typing import Any, Generic, List, Optional, TypeVar

from ..recordbatch import RecordBatch
from ..schema import Schema

WriteResultType = TypeVar("WriteResultType")


@dataclass
class WriteResult(Generic[WriteResultType]):
    """Wrapper for one write() call's outcome."""
    result: WriteResultType
    bytes_written: Optional[int] = None
    rows_written: Optional[int] = None


class DataSink(ABC, Generic[WriteResultType]):
    """Interface for writing DataFrames to a non-built-in sink."""

    def name(self) -> str:
        return "User-defined Data Sink"

    def schema(self) -> Optional[Schema]:
        """Schema of the summary batch finalize() returns (None = any)."""
        return None

    def start(self) -> None:
        pass

    @abstractmethod
    def write(self, batch: RecordBatch) -> WriteResult[WriteResultType]:
        ...

    @abstractmethod
    def finalize(self, results: List[WriteResult[WriteResultType]]
                 ) -> Any:
        ...
