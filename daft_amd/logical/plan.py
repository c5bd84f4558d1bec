"""Logical plan nodes (ref: /root/reference/src/daft-logical-plan/src/
logical_plan.rs:35-66 — Source, Project, Filter, Limit, Offset, Explode,
Unpivot, Sort, Repartition, IntoPartitions, Distinct, Aggregate, Pivot,
Concat, Union/Intersect, Join, Sink, Sample, MonotonicallyIncreasingId,
Window, TopN, IntoBatches)."""
from __future__ import annotations

import itertools
from typing import Any, Dict, List, Optional, Sequence, Tuple

from ..schema import DataType, Field, Schema
from ..expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                       ExprNode)

_ids = itertools.count()


class LogicalPlan:
    """Base logical plan node."""

    def __init__(self, children: List["LogicalPlan"]):
        self.children = children
        self.id = next(_ids)
        self._schema: Optional[Schema] = None

    @property
    def schema(self) -> Schema:
        if self._schema is None:
            self._schema = self._compute_schema()
        return self._schema

    def _compute_schema(self) -> Schema:
        raise NotImplementedError(type(self))

    def with_children(self, children: List["LogicalPlan"]) -> "LogicalPlan":
        raise NotImplementedError(type(self))

    def name(self) -> str:
        return type(self).__name__

    def describe(self) -> str:
        return self.name()

    # stats for join-side decisions (rows upper bound; None = unknown)
    def approx_num_rows(self) -> Optional[float]:
        ests = [c.approx_num_rows() for c in self.children]
        if any(e is None for e in ests):
            return None
        return max(ests) if ests else None

    def explain_lines(self, indent: int = 0) -> List[str]:
        lines = ["  " * indent + f"* {self.describe()}"]
        for c in self.children:
            lines.extend(c.explain_lines(indent + 1))
        return lines

    def semantic_id(self) -> str:
        child = ",".join(c.semantic_id() for c in self.children)
        return f"{self.describe()}[{child}]"


# global-row-count hints for SPMD planning: cache_key -> total rows across
# all ranks (populated by the distributed runner before optimization)
GLOBAL_ROW_HINTS: dict = {}


class Source(LogicalPlan):
    """In-memory source: holds partition refs (MicroPartition cache keys).

    `partitioning`: optional (token, key_names) declaring that rows with
    equal key values are colocated on one rank under distribution `token`
    (ref capability: partition specs on scans/tables; lets the distributed
    planner skip exchanges for co-sharded tables)."""

    def __init__(self, schema: Schema, cache_key: str, num_rows: int,
                 size_bytes: int = 0, partitioning=None, columns=None):
        super().__init__([])
        self._full_schema = schema
        self.cache_key = cache_key
        self.num_rows = num_rows
        self.size_bytes = size_bytes
        self.partitioning = partitioning
        # projection pushed INTO the source: cached partitions narrow to
        # these columns before any H2D morsel transfer (out-of-core scans
        # only move what the query reads)
        self.columns = columns

    def _compute_schema(self):
        if self.columns is not None:
            return Schema([f for f in self._full_schema
                           if f.name in set(self.columns)])
        return self._full_schema

    def with_children(self, children):
        assert not children
        return self

    def approx_num_rows(self):
        # SPMD invariant: under a multi-rank run every rank MUST derive the
        # same plan, so estimates use the GLOBAL row count (synced by the
        # distributed runner), never the rank-local shard size
        hint = GLOBAL_ROW_HINTS.get(self.cache_key)
        if hint is not None:
            return float(hint)
        return float(self.num_rows)

    def describe(self):
        return f"Source(rows={self.num_rows})"

    def semantic_id(self) -> str:
        cols = ",".join(self.columns) if self.columns is not None else "*"
        return f"Source({self.cache_key}|{cols})"


class ScanSource(LogicalPlan):
    """File scan source (parquet/csv/json); expands to scan tasks at
    physical planning (ref: daft-scan ScanTask, daft-logical-plan Source)."""

    def __init__(self, schema: Schema, paths: List[str], file_format: str,
                 storage_options: Optional[dict] = None,
                 pushdown_columns: Optional[List[str]] = None,
                 pushdown_filter: Optional[ExprNode] = None,
                 pushdown_limit: Optional[int] = None,
                 read_options: Optional[dict] = None):
        super().__init__([])
        self._full_schema = schema
        self.paths = paths
        self.file_format = file_format
        self.storage_options = storage_options or {}
        self.read_options = read_options or {}
        self.pushdown_columns = pushdown_columns
        self.pushdown_filter = pushdown_filter
        self.pushdown_limit = pushdown_limit

    def _compute_schema(self):
        if self.pushdown_columns is not None:
            return self._full_schema.select(self.pushdown_columns)
        return self._full_schema

    def with_children(self, children):
        assert not children
        return self

    def approx_num_rows(self):
        return None

    def describe(self):
        pd = []
        if self.pushdown_columns is not None:
            pd.append(f"cols={self.pushdown_columns}")
        if self.pushdown_filter is not None:
            pd.append(f"filter={self.pushdown_filter!r}")
        if self.pushdown_limit is not None:
            pd.append(f"limit={self.pushdown_limit}")
        extra = (", " + ", ".join(pd)) if pd else ""
        return (f"ScanSource({self.file_format}, files={len(self.paths)}"
                f"{extra})")


class Project(LogicalPlan):
    def __init__(self, child: LogicalPlan, exprs: List[ExprNode]):
        super().__init__([child])
        self.exprs = exprs

    def _compute_schema(self):
        cschema = self.children[0].schema
        return Schema([e.to_field(cschema) for e in self.exprs])

    def with_children(self, children):
        return Project(children[0], self.exprs)

    def describe(self):
        return f"Project({', '.join(map(repr, self.exprs))})"


class UDFProject(LogicalPlan):
    """Projection isolated to run an expensive UDF (ref: UDFProject node +
    SplitUDFs rule)."""

    def __init__(self, child: LogicalPlan, udf_expr: ExprNode,
                 passthrough: List[ExprNode]):
        super().__init__([child])
        self.udf_expr = udf_expr
        self.passthrough = passthrough

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = [e.to_field(cschema) for e in self.passthrough]
        fields.append(self.udf_expr.to_field(cschema))
        return Schema(fields)

    def with_children(self, children):
        return UDFProject(children[0], self.udf_expr, self.passthrough)

    def describe(self):
        return f"UDFProject({self.udf_expr!r})"


class Filter(LogicalPlan):
    def __init__(self, child: LogicalPlan, predicate: ExprNode):
        super().__init__([child])
        self.predicate = predicate

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Filter(children[0], self.predicate)

    def approx_num_rows(self):
        e = self.children[0].approx_num_rows()
        if e is None:
            return None
        from ..optimizer.join_reorder import selectivity
        return e * selectivity(self.predicate)

    def describe(self):
        return f"Filter({self.predicate!r})"


class Limit(LogicalPlan):
    def __init__(self, child: LogicalPlan, limit: int, offset: int = 0):
        super().__init__([child])
        self.limit = limit
        self.offset = offset

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Limit(children[0], self.limit, self.offset)

    def approx_num_rows(self):
        e = self.children[0].approx_num_rows()
        return float(self.limit) if e is None else min(e, float(self.limit))

    def describe(self):
        return f"Limit({self.limit}, offset={self.offset})"


class Explode(LogicalPlan):
    def __init__(self, child: LogicalPlan, exprs: List[ExprNode]):
        super().__init__([child])
        self.exprs = exprs

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = []
        explode_names = {e.to_field(cschema).name for e in self.exprs}
        for f in cschema:
            if f.name in explode_names:
                dt = f.dtype
                inner = dt.inner if dt.is_list() else dt
                fields.append(Field(f.name, inner))
            else:
                fields.append(f)
        return Schema(fields)

    def with_children(self, children):
        return Explode(children[0], self.exprs)

    def describe(self):
        return f"Explode({self.exprs!r})"


class Unpivot(LogicalPlan):
    def __init__(self, child: LogicalPlan, ids: List[ExprNode],
                 values: List[ExprNode], variable_name: str,
                 value_name: str):
        super().__init__([child])
        self.ids = ids
        self.values = values
        self.variable_name = variable_name
        self.value_name = value_name

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = [e.to_field(cschema) for e in self.ids]
        vfields = [e.to_field(cschema) for e in self.values]
        dt = vfields[0].dtype
        from ..schema import supertype
        for f in vfields[1:]:
            dt = supertype(dt, f.dtype)
        fields.append(Field(self.variable_name, DataType.string()))
        fields.append(Field(self.value_name, dt))
        return Schema(fields)

    def with_children(self, children):
        return Unpivot(children[0], self.ids, self.values,
                       self.variable_name, self.value_name)


class Sort(LogicalPlan):
    def __init__(self, child: LogicalPlan, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool]):
        super().__init__([child])
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Sort(children[0], self.by, self.descending, self.nulls_first)

    def describe(self):
        return f"Sort({self.by!r}, desc={self.descending})"


class TopN(LogicalPlan):
    def __init__(self, child: LogicalPlan, by: List[ExprNode],
                 descending: List[bool], nulls_first: List[bool],
                 limit: int, offset: int = 0):
        super().__init__([child])
        self.by = by
        self.descending = descending
        self.nulls_first = nulls_first
        self.limit = limit
        self.offset = offset

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return TopN(children[0], self.by, self.descending, self.nulls_first,
                    self.limit, self.offset)

    def describe(self):
        return f"TopN({self.by!r}, n={self.limit})"


class Repartition(LogicalPlan):
    """scheme in {hash, random, range, into}; exchanged over RCCL when
    distributed (ref: LogicalPlan::Repartition + daft-shuffles)."""

    def __init__(self, child: LogicalPlan, scheme: str,
                 num_partitions: Optional[int],
                 by: Optional[List[ExprNode]] = None):
        super().__init__([child])
        self.scheme = scheme
        self.num_partitions = num_partitions
        self.by = by or []

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Repartition(children[0], self.scheme, self.num_partitions,
                           self.by)

    def describe(self):
        return f"Repartition({self.scheme}, n={self.num_partitions})"


class Distinct(LogicalPlan):
    def __init__(self, child: LogicalPlan,
                 subset: Optional[List[ExprNode]] = None):
        super().__init__([child])
        self.subset = subset

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Distinct(children[0], self.subset)


class Aggregate(LogicalPlan):
    def __init__(self, child: LogicalPlan, groupby: List[ExprNode],
                 aggs: List[ExprNode]):
        super().__init__([child])
        self.groupby = groupby
        self.aggs = aggs

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = [e.to_field(cschema) for e in self.groupby]
        fields.extend(e.to_field(cschema) for e in self.aggs)
        return Schema(fields)

    def with_children(self, children):
        return Aggregate(children[0], self.groupby, self.aggs)

    def approx_num_rows(self):
        if not self.groupby:
            return 1.0
        return super().approx_num_rows()

    def describe(self):
        return f"Aggregate(by={self.groupby!r}, aggs={self.aggs!r})"


class Pivot(LogicalPlan):
    def __init__(self, child: LogicalPlan, groupby: List[ExprNode],
                 pivot_col: ExprNode, value_col: ExprNode, agg_kind: str,
                 names: List[str]):
        super().__init__([child])
        self.groupby = groupby
        self.pivot_col = pivot_col
        self.value_col = value_col
        self.agg_kind = agg_kind
        self.names = names

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = [e.to_field(cschema) for e in self.groupby]
        vdt = Agg(self.agg_kind, self.value_col).to_field(cschema).dtype
        for n in self.names:
            fields.append(Field(n, vdt))
        return Schema(fields)

    def with_children(self, children):
        return Pivot(children[0], self.groupby, self.pivot_col,
                     self.value_col, self.agg_kind, self.names)


class Concat(LogicalPlan):
    def __init__(self, a: LogicalPlan, b: LogicalPlan):
        super().__init__([a, b])

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Concat(children[0], children[1])

    def approx_num_rows(self):
        ests = [c.approx_num_rows() for c in self.children]
        if any(e is None for e in ests):
            return None
        return sum(ests)


class Join(LogicalPlan):
    def __init__(self, left: LogicalPlan, right: LogicalPlan,
                 left_on: List[ExprNode], right_on: List[ExprNode],
                 how: str, suffix: str = "_right",
                 prefix: Optional[str] = None):
        super().__init__([left, right])
        self.left_on = left_on
        self.right_on = right_on
        self.how = how
        self.suffix = suffix
        self.prefix = prefix

    def _compute_schema(self):
        ls, rs = self.children[0].schema, self.children[1].schema
        if self.how in ("semi", "anti"):
            self._right_cols = []
            return ls
        fields = ls.fields()
        # join keys with identical names merge (daft semantics: right key cols
        # with the same name as left key cols are dropped)
        left_key_names = {e.to_field(ls).name for e in self.left_on}
        right_key_names = {e.to_field(rs).name for e in self.right_on}
        taken = set(ls.names())
        self._right_cols: List[Tuple[str, str]] = []  # (src, out) names
        for f in rs:
            if f.name in right_key_names and f.name in left_key_names:
                continue
            out = f.name
            if out in taken:
                out = (self.prefix + out) if self.prefix else out + self.suffix
                i = 1
                while out in taken:
                    out = f"{f.name}{self.suffix}{i}"
                    i += 1
            taken.add(out)
            fields.append(Field(out, f.dtype))
            self._right_cols.append((f.name, out))
        return Schema(fields)

    def right_passthrough(self) -> List[Tuple[str, str]]:
        _ = self.schema
        return self._right_cols

    def with_children(self, children):
        return Join(children[0], children[1], self.left_on, self.right_on,
                    self.how, self.suffix, self.prefix)

    def approx_num_rows(self):
        ests = [c.approx_num_rows() for c in self.children]
        if any(e is None for e in ests):
            return None
        return max(ests)

    def describe(self):
        return (f"Join({self.how}, on={self.left_on!r}=={self.right_on!r})")


class AsofJoin(LogicalPlan):
    """Nearest-key join (ref: LogicalPlan::AsofJoin + join/asof_join.rs)."""

    def __init__(self, left: LogicalPlan, right: LogicalPlan, left_on: str,
                 right_on: str, left_by: List[str], right_by: List[str],
                 strategy: str = "backward", suffix: str = "_right"):
        super().__init__([left, right])
        self.left_on = left_on
        self.right_on = right_on
        self.left_by = left_by
        self.right_by = right_by
        self.strategy = strategy
        self.suffix = suffix

    def _compute_schema(self):
        ls, rs = self.children[0].schema, self.children[1].schema
        fields = ls.fields()
        taken = set(ls.names())
        skip = set(self.right_by) | {self.right_on}
        self._right_cols = []
        for f in rs:
            if f.name in skip and (f.name in set(self.left_by) or
                                   f.name == self.left_on):
                continue
            out = f.name if f.name not in taken else f.name + self.suffix
            i = 1
            while out in taken:
                out = f"{f.name}{self.suffix}{i}"
                i += 1
            taken.add(out)
            fields.append(Field(out, f.dtype))
            self._right_cols.append((f.name, out))
        return Schema(fields)

    def right_passthrough(self):
        _ = self.schema
        return self._right_cols

    def with_children(self, children):
        return AsofJoin(children[0], children[1], self.left_on,
                        self.right_on, self.left_by, self.right_by,
                        self.strategy, self.suffix)

    def describe(self):
        return (f"AsofJoin({self.left_on}~{self.right_on}, "
                f"by={self.left_by}, {self.strategy})")


class Sample(LogicalPlan):
    def __init__(self, child: LogicalPlan, fraction: float,
                 with_replacement: bool = False, seed: Optional[int] = None):
        super().__init__([child])
        self.fraction = fraction
        self.with_replacement = with_replacement
        self.seed = seed

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return Sample(children[0], self.fraction, self.with_replacement,
                      self.seed)


class MonotonicallyIncreasingId(LogicalPlan):
    def __init__(self, child: LogicalPlan, column_name: str):
        super().__init__([child])
        self.column_name = column_name

    def _compute_schema(self):
        fields = [Field(self.column_name, DataType.uint64())]
        fields.extend(self.children[0].schema.fields())
        return Schema(fields)

    def with_children(self, children):
        return MonotonicallyIncreasingId(children[0], self.column_name)


class IntoBatches(LogicalPlan):
    def __init__(self, child: LogicalPlan, batch_size: int):
        super().__init__([child])
        self.batch_size = batch_size

    def _compute_schema(self):
        return self.children[0].schema

    def with_children(self, children):
        return IntoBatches(children[0], self.batch_size)


class Window(LogicalPlan):
    """Window functions over partition/order spec (ref: LogicalPlan::Window,
    daft-dsl expr/window.rs)."""

    def __init__(self, child: LogicalPlan, window_exprs: List[ExprNode],
                 partition_by: List[ExprNode], order_by: List[ExprNode],
                 descending: List[bool], names: List[str]):
        super().__init__([child])
        self.window_exprs = window_exprs
        self.partition_by = partition_by
        self.order_by = order_by
        self.descending = descending
        self.names = names

    def _compute_schema(self):
        cschema = self.children[0].schema
        fields = cschema.fields()
        for e, n in zip(self.window_exprs, self.names):
            from ..physical.window import window_out_field
            fields.append(window_out_field(e, n, cschema))
        return Schema(fields)

    def with_children(self, children):
        return Window(children[0], self.window_exprs, self.partition_by,
                      self.order_by, self.descending, self.names)


class Sink(LogicalPlan):
    """Write sink (parquet/csv/json); emits a manifest of written paths
    (ref: LogicalPlan::Sink -> PhysicalWrite + CommitWrite)."""

    def __init__(self, child: LogicalPlan, file_format: str, root_dir: str,
                 write_mode: str = "overwrite",
                 partition_cols: Optional[List[ExprNode]] = None,
                 options: Optional[dict] = None):
        super().__init__([child])
        self.file_format = file_format
        self.root_dir = root_dir
        self.write_mode = write_mode
        self.partition_cols = partition_cols or []
        self.options = options or {}

    def _compute_schema(self):
        return Schema([Field("path", DataType.string())])

    def with_children(self, children):
        return Sink(children[0], self.file_format, self.root_dir,
                    self.write_mode, self.partition_cols, self.options)

    def describe(self):
        return f"Sink({self.file_format} -> {self.root_dir})"
