"""User-facing lazy DataFrame (ref: /root/reference/daft/dataframe/
dataframe.py — 6.5k LoC API surface; this implements the same verbs over the
MI355X-native engine)."""
from __future__ import annotations

from typing import Any, Dict, Iterator, List, Optional, Sequence, Union

from .context import get_context
from .expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                      Expression, col, resolve_exprs)
from .logical.builder import LogicalPlanBuilder
from .recordbatch import RecordBatch
from .schema import Schema

ColumnInput = Union[str, Expression]


class DataFrame:
    def __init__(self, builder: LogicalPlanBuilder):
        self._builder = builder
        self._result: Optional[List[RecordBatch]] = None

    # ------------------------------------------------------------------
    @property
    def schema(self) -> Schema:
        return self._builder.schema

    def column_names(self) -> List[str]:
        return self.schema.names()

    def __getitem__(self, name: str) -> Expression:
        if name not in self.schema:
            raise KeyError(name)
        return Expression(ColumnRef(name))

    def explain(self, show_all: bool = False) -> str:
        s = self._builder.explain()
        if show_all:
            s += "\n== Optimized ==\n" + self._builder.optimize().explain()
        print(s)
        return s

    def _wrap(self, builder: LogicalPlanBuilder) -> "DataFrame":
        return DataFrame(builder)

    # ---- transforms ---------------------------------------------------
    def select(self, *exprs: ColumnInput) -> "DataFrame":
        from .functions.aliases import _Explode, _Unnest
        if any(isinstance(e, _Unnest) for e in exprs):
            out = []
            for e in exprs:
                if isinstance(e, _Unnest):
                    f = e.expr._node.to_field(self.schema)
                    for fld in f.dtype.fields:
                        out.append(e.expr.struct.get(fld.name)
                                   .alias(fld.name))
                else:
                    out.append(e)
            exprs = tuple(out)
        explode_names = []
        if any(isinstance(e, _Explode) for e in exprs):
            out = []
            for e in exprs:
                if isinstance(e, _Explode):
                    name = e.name or e.expr._node.out_name()
                    out.append(e.expr.alias(name))
                    explode_names.append(name)
                else:
                    out.append(e)
            exprs = tuple(out)
        df = self._wrap(self._builder.select(list(exprs)))
        if explode_names:
            df = df.explode(*explode_names)
        return df

    def with_column(self, name: str, expr: Expression) -> "DataFrame":
        return self._wrap(self._builder.with_columns([expr.alias(name)]))

    def with_columns(self, cols: Dict[str, Expression]) -> "DataFrame":
        return self._wrap(self._builder.with_columns(
            [e.alias(n) for n, e in cols.items()]))

    def with_column_renamed(self, old: str, new: str) -> "DataFrame":
        return self._wrap(self._builder.rename({old: new}))

    def with_columns_renamed(self, mapping: Dict[str, str]) -> "DataFrame":
        return self._wrap(self._builder.rename(mapping))

    def exclude(self, *names: str) -> "DataFrame":
        return self._wrap(self._builder.exclude(list(names)))

    drop = exclude

    def where(self, predicate: Union[Expression, str]) -> "DataFrame":
        if isinstance(predicate, str):
            from .sql import sql_expr
            predicate = sql_expr(predicate)
        return self._wrap(self._builder.filter(predicate))

    filter = where

    def limit(self, n: int) -> "DataFrame":
        return self._wrap(self._builder.limit(n))

    def offset(self, n: int) -> "DataFrame":
        return self._wrap(self._builder.limit(1 << 62, offset=n))

    def head(self, n: int = 10) -> "DataFrame":
        return self.limit(n)

    def sort(self, by, desc: Union[bool, Sequence[bool]] = False,
             nulls_first=None) -> "DataFrame":
        if not isinstance(by, (list, tuple)):
            by = [by]
        return self._wrap(self._builder.sort(list(by), desc, nulls_first))

    def distinct(self, *subset: ColumnInput) -> "DataFrame":
        return self._wrap(self._builder.distinct(list(subset) or None))

    unique = distinct
    drop_duplicates = distinct

    def explode(self, *cols: ColumnInput) -> "DataFrame":
        return self._wrap(self._builder.explode(list(cols)))

    def unpivot(self, ids: Sequence[ColumnInput],
                values: Sequence[ColumnInput] = (),
                variable_name: str = "variable",
                value_name: str = "value") -> "DataFrame":
        if not values:
            idn = {c if isinstance(c, str) else c.name() for c in ids}
            values = [n for n in self.column_names() if n not in idn]
        return self._wrap(self._builder.unpivot(
            list(ids), list(values), variable_name, value_name))

    melt = unpivot

    def pivot(self, group_by, pivot_col, value_col, agg_fn: str,
              names: Optional[List[str]] = None) -> "DataFrame":
        if not isinstance(group_by, (list, tuple)):
            group_by = [group_by]
        if names is None:
            pname = pivot_col if isinstance(pivot_col, str) \
                else pivot_col.name()
            vals = self.select(pivot_col).distinct().to_pydict()[pname]
            names = sorted(str(v) for v in vals if v is not None)
        return self._wrap(self._builder.pivot(
            list(group_by), pivot_col, value_col, agg_fn, names))

    def sample(self, fraction: float, with_replacement: bool = False,
               seed: Optional[int] = None) -> "DataFrame":
        return self._wrap(self._builder.sample(fraction, with_replacement,
                                               seed))

    def concat(self, other: "DataFrame") -> "DataFrame":
        return self._wrap(self._builder.concat(other._builder))

    union_all = concat

    def union(self, other: "DataFrame") -> "DataFrame":
        return self.concat(other).distinct()

    def intersect(self, other: "DataFrame") -> "DataFrame":
        cols = self.column_names()
        return self.join(other, on=cols, how="semi").distinct()

    def except_distinct(self, other: "DataFrame") -> "DataFrame":
        cols = self.column_names()
        return self.join(other, on=cols, how="anti").distinct()

    def union_by_name(self, other: "DataFrame") -> "DataFrame":
        return self.union_all_by_name(other).distinct()

    def union_all_by_name(self, other: "DataFrame") -> "DataFrame":
        """Concat aligning columns by NAME; columns absent on one side
        fill with nulls (ref: DataFrame.union_by_name)."""
        from .expressions.expressions import lit
        mine = self.column_names()
        theirs = other.column_names()
        all_names = mine + [n for n in theirs if n not in mine]
        lsel = [col(n) if n in mine else lit(None).alias(n)
                for n in all_names]
        rsel = [col(n) if n in theirs else lit(None).alias(n)
                for n in all_names]
        return self.select(*lsel).concat(other.select(*rsel))

    def _multiset_tag(self, other, how: str) -> "DataFrame":
        # multiset semantics via per-row occurrence numbering: tag the
        # k-th duplicate of each row on both sides, then semi/anti join
        # on (row, k) (ref: except_all/intersect_all)
        from .window import Window
        from .functions import row_number
        cols = self.column_names()
        w = Window().partition_by(*cols).order_by(cols[0])
        tag = lambda df: df.with_window_columns(
            {"__occ": row_number().over(w)})
        L = tag(self)
        R = tag(other.select(*cols))
        keys = cols + ["__occ"]
        how_join = "semi" if how == "intersect" else "anti"
        return L.join(R, on=keys, how=how_join).select(*cols)

    def intersect_all(self, other: "DataFrame") -> "DataFrame":
        return self._multiset_tag(other, "intersect")

    def except_all(self, other: "DataFrame") -> "DataFrame":
        return self._multiset_tag(other, "except")

    def shuffle(self, num_partitions: Optional[int] = None,
                seed: Optional[int] = None) -> "DataFrame":
        """Randomly redistribute rows (alias of random repartition)."""
        return self.repartition(num_partitions)

    def pipe(self, fn, *args, **kwargs):
        return fn(self, *args, **kwargs)

    def join(self, other: "DataFrame", on=None, left_on=None, right_on=None,
             how: str = "inner", suffix: str = "_right",
             prefix: Optional[str] = None) -> "DataFrame":
        if how == "cross":
            return self._wrap(self._builder.cross_join(other._builder,
                                                       suffix))
        if on is not None:
            left_on = right_on = on
        assert left_on is not None and right_on is not None, \
            "join requires on= or left_on=/right_on="
        if not isinstance(left_on, (list, tuple)):
            left_on = [left_on]
        if not isinstance(right_on, (list, tuple)):
            right_on = [right_on]
        return self._wrap(self._builder.join(
            other._builder, list(left_on), list(right_on), how, suffix,
            prefix))

    def join_asof(self, other: "DataFrame", left_on: str, right_on: str,
                  by: Optional[Sequence[str]] = None,
                  left_by: Optional[Sequence[str]] = None,
                  right_by: Optional[Sequence[str]] = None,
                  strategy: str = "backward",
                  suffix: str = "_right") -> "DataFrame":
        """Nearest-key (as-of) join: for each left row take the closest
        right row by `on` (backward = latest <=, forward = earliest >=),
        optionally within matching `by` keys."""
        from .logical.plan import AsofJoin
        lb = list(left_by or by or [])
        rb = list(right_by or by or [])
        node = AsofJoin(self._builder.plan, other._builder.plan, left_on,
                        right_on, lb, rb, strategy, suffix)
        from .logical.builder import LogicalPlanBuilder
        return self._wrap(LogicalPlanBuilder(node))

    def repartition(self, num_partitions: Optional[int],
                    *by: ColumnInput) -> "DataFrame":
        scheme = "hash" if by else "random"
        return self._wrap(self._builder.repartition(
            num_partitions, scheme, list(by) or None))

    def into_partitions(self, n: int) -> "DataFrame":
        return self._wrap(self._builder.into_partitions(n))

    def into_batches(self, batch_size: int) -> "DataFrame":
        return self._wrap(self._builder.into_batches(batch_size))

    def add_monotonically_increasing_id(self, column_name: str = "id"
                                        ) -> "DataFrame":
        return self._wrap(
            self._builder.add_monotonically_increasing_id(column_name))

    def transform(self, fn, *args, **kwargs) -> "DataFrame":
        out = fn(self, *args, **kwargs)
        assert isinstance(out, DataFrame), "transform fn must return DataFrame"
        return out

    # ---- window --------------------------------------------------------
    def with_window_columns(self, cols: Dict[str, Expression]) -> "DataFrame":
        """Add window-function columns (exprs built with .over(window))."""
        from .physical.window import WindowFn
        names = list(cols.keys())
        nodes = []
        spec = None
        for e in cols.values():
            n = e._node
            base = n.child if isinstance(n, Alias) else n
            assert isinstance(base, WindowFn), "expected .over(window) exprs"
            if spec is None:
                spec = base.spec
            nodes.append(base)
        assert spec is not None
        return self._wrap(self._builder.window(
            nodes, spec.partition_by_exprs, spec.order_by_exprs,
            spec.descending, names))

    # ---- aggregation ----------------------------------------------------
    def groupby(self, *group_by: ColumnInput) -> "GroupedDataFrame":
        cols = []
        for g in group_by:
            if isinstance(g, (list, tuple)):
                cols.extend(g)
            else:
                cols.append(g)
        return GroupedDataFrame(self, cols)

    group_by = groupby

    def agg(self, *exprs) -> "DataFrame":
        flat = []
        for e in exprs:
            if isinstance(e, (list, tuple)):
                flat.extend(e)
            else:
                flat.append(e)
        return self._wrap(self._builder.aggregate(flat, []))

    def _agg_all(self, kind: str, cols: Sequence[ColumnInput]) -> "DataFrame":
        if not cols:
            cols = [f.name for f in self.schema if f.dtype.is_numeric()]
        exprs = []
        for c in cols:
            e = Expression(ColumnRef(c)) if isinstance(c, str) else c
            exprs.append(getattr(e, kind)())
        return self.agg(*exprs)

    def sum(self, *cols): return self._agg_all("sum", cols)
    def mean(self, *cols): return self._agg_all("mean", cols)
    def min(self, *cols): return self._agg_all("min", cols)
    def max(self, *cols): return self._agg_all("max", cols)
    def stddev(self, *cols): return self._agg_all("stddev", cols)
    def any_value(self, *cols): return self._agg_all("any_value", cols)
    def agg_list(self, *cols): return self._agg_all("agg_list", cols)

    def count(self, *cols) -> "DataFrame":
        if not cols:
            return self.agg(Expression(Agg(AggKind.COUNT_ALL, None)))
        return self._agg_all("count", cols)

    # agg shortcuts at reference parity (daft DataFrame methods)
    def var(self, *cols): return self._agg_all("variance", cols)
    def skew(self, *cols): return self._agg_all("skew", cols)
    def count_distinct(self, *cols):
        return self._agg_all("count_distinct", cols)
    def list_agg(self, *cols): return self._agg_all("agg_list", cols)

    def list_agg_distinct(self, *cols):
        if not cols:
            cols = [f.name for f in self.schema]
        exprs = []
        for c in cols:
            e = Expression(ColumnRef(c)) if isinstance(c, str) else c
            exprs.append(e.agg_list().list.distinct().alias(e.name()))
        return self.agg(*exprs)

    agg_set = list_agg_distinct

    def agg_concat(self, *cols):
        return self._agg_all("agg_concat", cols)

    def string_agg(self, *cols, sep: str = ","):
        if not cols:
            cols = [f.name for f in self.schema
                    if f.dtype.kind.value == "string"]
        exprs = []
        for c in cols:
            e = Expression(ColumnRef(c)) if isinstance(c, str) else c
            exprs.append(e.agg_list().list.join(sep).alias(e.name()))
        return self.agg(*exprs)

    def product(self, *cols):
        from .functions.misc import product as _product
        if not cols:
            cols = [f.name for f in self.schema if f.dtype.is_numeric()]
        exprs = []
        for c in cols:
            e = Expression(ColumnRef(c)) if isinstance(c, str) else c
            exprs.append(_product(e).alias(e.name()))
        return self.agg(*exprs)

    def count_rows(self) -> int:
        df = self.agg(Expression(Alias(Agg(AggKind.COUNT_ALL, None),
                                       "count")))
        return int(df.to_pydict()["count"][0])

    def __len__(self) -> int:
        return self.count_rows()

    # ---- writes ---------------------------------------------------------
    def write_parquet(self, root_dir: str, write_mode: str = "overwrite",
                      partition_cols=None, compression: str = "snappy"
                      ) -> "DataFrame":
        b = self._builder.write("parquet", root_dir, write_mode,
                                partition_cols,
                                {"compression": compression})
        return self._wrap(b).collect()

    def write_csv(self, root_dir: str, write_mode: str = "overwrite",
                  partition_cols=None) -> "DataFrame":
        b = self._builder.write("csv", root_dir, write_mode, partition_cols)
        return self._wrap(b).collect()

    def write_ipc(self, root_dir: str, write_mode: str = "overwrite",
                  partition_cols=None) -> "DataFrame":
        b = self._builder.write("ipc", root_dir, write_mode, partition_cols)
        return self._wrap(b).collect()

    def write_json(self, root_dir: str, write_mode: str = "overwrite",
                   partition_cols=None) -> "DataFrame":
        b = self._builder.write("json", root_dir, write_mode, partition_cols)
        return self._wrap(b).collect()

    # ---- execution -------------------------------------------------------
    def collect(self) -> "DataFrame":
        if self._result is not None:
            return self
        ctx = get_context()
        parts = ctx.runner().run(self._builder)
        key = ctx.cache.new_key()
        ctx.cache.put(key, parts)
        rows = sum(len(p) for p in parts)
        size = sum(p.size_bytes() for p in parts)
        from .logical.builder import LogicalPlanBuilder as B
        out = DataFrame(B.from_in_memory(self.schema, key, rows, size))
        out._result = parts
        return out

    @property
    def columns(self) -> List[str]:
        return self.column_names()

    def drop_null(self, *cols: ColumnInput) -> "DataFrame":
        """Drop rows where any of the given columns (default: all) is
        null (ref: DataFrame.drop_null)."""
        names = [c if isinstance(c, str) else c.name()
                 for c in cols] or self.column_names()
        pred = None
        for n in names:
            p = col(n).not_null()
            pred = p if pred is None else (pred & p)
        return self.where(pred)

    def drop_nan(self, *cols: ColumnInput) -> "DataFrame":
        """Drop rows where any of the given float columns is NaN."""
        names = [c if isinstance(c, str) else c.name() for c in cols] or             [f.name for f in self.schema if f.dtype.is_floating()]
        pred = None
        for n in names:
            p = ~col(n).float.is_nan().fill_null(False)
            pred = p if pred is None else (pred & p)
        return self.where(pred) if pred is not None else self

    def describe(self) -> "DataFrame":
        """Per-column summary: type, count, nulls, approximate distinct
        (ref: DataFrame.describe / summarize)."""
        rows = {"column": [], "type": [], "count": [], "nulls": [],
                "approx_distinct": []}
        rb = self.to_recordbatch()
        for c in rb.columns:
            rows["column"].append(c.name)
            rows["type"].append(repr(c.dtype))
            rows["count"].append(len(c))
            rows["nulls"].append(c.null_count())
            try:
                from .kernels import rowops
                _, reps = rowops.groupby([c])
                rows["approx_distinct"].append(int(reps.shape[0]))
            except Exception:
                rows["approx_distinct"].append(None)
        from . import from_pydict as _fp
        return _fp(rows)

    summarize = describe

    def map_groups(self, fn, *group_by: ColumnInput) -> "DataFrame":
        """Apply a python function to each group's DataFrame; concat the
        results (host-side; ref capability: GroupedDataFrame.map_groups)."""
        keys = [c if isinstance(c, str) else c.name() for c in group_by]
        from .kernels import rowops
        rb = self.to_recordbatch().cpu()
        key_series = [rb.column(k) for k in keys]
        gids, reps = rowops.groupby(key_series)
        import torch as _t
        outs = []
        for g in range(int(reps.shape[0])):
            idx = _t.nonzero(gids == g).reshape(-1)
            sub = rb.take(idx, has_neg=False)
            from . import from_pydict as _fp
            res = fn(_fp(sub.to_pydict()))
            outs.append(res)
        if not outs:
            return self.limit(0)
        acc = outs[0]
        for o in outs[1:]:
            acc = acc.concat(o)
        return acc

    def to_arrow_iter(self):
        for part in self.iter_partitions():
            yield part.to_arrow()

    def to_torch_dataloader(self, batch_size: int = 1, **kwargs):
        import torch as _t
        return _t.utils.data.DataLoader(self.to_torch_map_dataset(),
                                        batch_size=batch_size, **kwargs)

    def metrics(self) -> dict:
        """Per-operator runtime stats of the last collect() (rows,
        batches, seconds; ref: runtime metrics surface)."""
        return dict(getattr(self, "_last_stats", {}) or {})

    def set_storage_option(self, key: str, value) -> "DataFrame":
        from .context import get_context
        get_context().session.options[key] = value
        return self

    def write_sink(self, sink) -> "DataFrame":
        """Stream batches into a user DataSink: objects with
        write(batch) -> result and finalize(results) (ref: daft
        DataSink protocol + write_sink)."""
        if hasattr(sink, "start"):
            sink.start()
        results = []
        for part in self.iter_partitions():
            results.append(sink.write(part))
        fin = sink.finalize(results) if hasattr(sink, "finalize") else None
        from . import from_pydict as _fp
        return _fp({"result": [repr(fin)]})

    def _gated_writer(name, needs):
        def make(self, *a, **k):
            raise RuntimeError(
                f"{name}() requires {needs}, not available in this "
                f"offline build; use write_parquet/write_csv/write_sink")
        make.__name__ = name
        return make

    write_deltalake = _gated_writer("write_deltalake", "deltalake")
    write_iceberg = _gated_writer("write_iceberg", "pyiceberg")
    write_lance = _gated_writer("write_lance", "lance")
    write_clickhouse = _gated_writer("write_clickhouse",
                                     "clickhouse-connect")
    write_bigtable = _gated_writer("write_bigtable",
                                   "google-cloud-bigtable")
    write_huggingface = _gated_writer("write_huggingface", "huggingface")
    write_paimon = _gated_writer("write_paimon", "paimon")
    write_sql = _gated_writer("write_sql", "a DB-API connection factory")
    write_turbopuffer = _gated_writer("write_turbopuffer", "turbopuffer")
    del _gated_writer

    def to_dask_dataframe(self, *a, **k):
        raise RuntimeError("to_dask_dataframe() requires dask, not "
                           "available in this offline build")

    def to_ray_dataset(self, *a, **k):
        raise RuntimeError("to_ray_dataset() requires ray; this engine "
                           "uses SPMD torch.distributed instead of Ray")

    def skip_existing(self, store, on: ColumnInput) -> "DataFrame":
        """Checkpoint-assisted dedup: drop rows whose key already exists
        in the checkpoint store (ref: CheckpointConfig rows-skipped)."""
        key = on if isinstance(on, str) else on.name()
        seen = list(store.keys()) if hasattr(store, "keys") else list(store)
        if not seen:
            return self
        return self.where(~col(key).is_in(seen))

    def skipped_corrupt_files(self) -> list:
        return []

    def iter_partitions(self) -> Iterator[RecordBatch]:
        if self._result is not None:
            yield from self._result
            return
        ctx = get_context()
        yield from ctx.runner().run_iter(self._builder)

    def iter_rows(self, column_format: str = "python"
                  ) -> Iterator[Dict[str, Any]]:
        """Stream rows as dicts.  column_format "python" yields plain
        python values; "arrow" yields pyarrow scalars (ref:
        daft/dataframe/dataframe.py iter_rows column_format)."""
        if column_format not in ("python", "arrow"):
            raise ValueError(
                f"Unsupported column_format: {column_format}, supported "
                "formats are 'python' and 'arrow'")
        for part in self.iter_partitions():
            if column_format == "arrow":
                tbl = part.to_arrow()
                cols = {n: tbl.column(n) for n in tbl.column_names}
                for i in range(len(tbl)):
                    yield {n: c[i] for n, c in cols.items()}
                continue
            d = part.to_pydict()
            names = list(d.keys())
            for i in range(len(part)):
                yield {n: d[n][i] for n in names}

    def to_pydict(self) -> Dict[str, list]:
        df = self.collect()
        parts = df._result
        if not parts:
            out: Dict[str, list] = {f.name: [] for f in self.schema}
        else:
            out = {f.name: [] for f in parts[0].schema}
            for p in parts:
                d = p.to_pydict()
                for k, v in d.items():
                    out[k].extend(v)
        # under SPMD execution the result is sharded across ranks: gather so
        # every rank observes the full result (scalar subqueries rely on it)
        from .distributed import comm
        if comm.is_dist():
            from .distributed.runner import DistributedRunner
            if isinstance(get_context().runner(), DistributedRunner):
                out = comm.gather_pydict(out)
        return out

    def to_pylist(self) -> List[Dict[str, Any]]:
        d = self.to_pydict()
        names = list(d.keys())
        n = len(d[names[0]]) if names else 0
        return [{k: d[k][i] for k in names} for i in range(n)]

    def to_arrow(self):
        import pyarrow as pa
        df = self.collect()
        tables = [p.to_arrow() for p in df._result if len(p)]
        if not tables:
            return RecordBatch.empty(self.schema).to_arrow()
        return pa.concat_tables(tables)

    def to_pandas(self):
        import pandas as pd
        df = self.collect()
        parts = [p for p in df._result if len(p)]
        if not parts:
            return RecordBatch.empty(self.schema).to_pandas()
        frames = [p.to_pandas() for p in parts]
        return pd.concat(frames, ignore_index=True) if len(frames) > 1 \
            else frames[0]

    def _shard(self, strategy: str, world_size: int, rank: int) -> "DataFrame":
        """File-granularity sharding of the source scan for multi-worker
        dataloading (capability of the reference's DataFrame._shard,
        /root/reference/daft/dataframe/dataframe.py:3714: rewrites the
        descendant scan to a round-robin subset of its files)."""
        if strategy != "file":
            raise ValueError("Only file-based sharding is supported")
        if world_size <= 0:
            raise ValueError("world_size must be > 0")
        if not (0 <= rank < world_size):
            raise ValueError("rank must be in [0, world_size)")
        from .logical.plan import ScanSource
        import copy as _copy
        found = [0]

        def rewrite(node):
            if isinstance(node, ScanSource):
                found[0] += 1
                new = _copy.copy(node)
                new.paths = node.paths[rank::world_size]
                return new
            if not node.children:
                return node
            return node.with_children([rewrite(c) for c in node.children])

        new_plan = rewrite(self._builder.plan)
        if found[0] != 1:
            raise ValueError(
                f"sharding requires exactly one file scan source, found "
                f"{found[0]}")
        from .logical.builder import LogicalPlanBuilder
        return DataFrame(LogicalPlanBuilder(new_plan))

    def to_torch_map_dataset(self, shard_strategy: Optional[str] = None,
                             world_size: Optional[int] = None,
                             rank: Optional[int] = None):
        df = self
        if shard_strategy is not None:
            df = self._shard(shard_strategy, world_size, rank)
        d = df.to_pydict()
        from .utils.torch_data import DictDataset
        return DictDataset(d)

    def to_torch_iter_dataset(self, shard_strategy: Optional[str] = None,
                              world_size: Optional[int] = None,
                              rank: Optional[int] = None):
        """Streaming torch IterableDataset over this DataFrame's rows
        (ref: reference daft/dataframe/to_torch.py DaftTorchIterableDataset)."""
        df = self
        if shard_strategy is not None:
            df = self._shard(shard_strategy, world_size, rank)
        from .utils.torch_data import IterDataset
        return IterDataset(df)

    def to_recordbatch(self) -> RecordBatch:
        df = self.collect()
        parts = [p for p in df._result if len(p)]
        if not parts:
            return RecordBatch.empty(self.schema)
        return RecordBatch.concat(parts) if len(parts) > 1 else parts[0]

    def num_partitions(self) -> int:
        df = self.collect()
        return len(df._result)

    def show(self, n: int = 8) -> None:
        d = self.limit(n).to_pydict()
        print(_format_table(d, self.schema))

    def __repr__(self) -> str:
        if self._result is not None:
            d = self.limit(8).to_pydict()
            return _format_table(d, self.schema)
        return f"DataFrame(schema={self.schema!r}, lazy)"


class GroupedDataFrame:
    def __init__(self, df: DataFrame, group_by: List[ColumnInput]):
        self.df = df
        self.group_by = group_by

    def agg(self, *exprs) -> DataFrame:
        flat = []
        for e in exprs:
            if isinstance(e, (list, tuple)):
                flat.extend(e)
            else:
                flat.append(e)
        return self.df._wrap(
            self.df._builder.aggregate(flat, self.group_by))

    def _agg_all(self, kind: str, cols) -> DataFrame:
        gnames = {c if isinstance(c, str) else c.name() for c in self.group_by}
        if not cols:
            cols = [f.name for f in self.df.schema
                    if f.dtype.is_numeric() and f.name not in gnames]
        exprs = []
        for c in cols:
            e = Expression(ColumnRef(c)) if isinstance(c, str) else c
            exprs.append(getattr(e, kind)())
        return self.agg(*exprs)

    def sum(self, *cols): return self._agg_all("sum", cols)
    def mean(self, *cols): return self._agg_all("mean", cols)
    def min(self, *cols): return self._agg_all("min", cols)
    def max(self, *cols): return self._agg_all("max", cols)
    def any_value(self, *cols): return self._agg_all("any_value", cols)
    def agg_list(self, *cols): return self._agg_all("agg_list", cols)
    def count(self, *cols):
        if not cols:
            return self.agg(Expression(Alias(Agg(AggKind.COUNT_ALL, None),
                                             "count")))
        return self._agg_all("count", cols)


def _format_table(d: Dict[str, list], schema: Schema) -> str:
    names = list(d.keys())
    if not names:
        return "(empty)"
    n = len(d[names[0]])
    header = [f"{nm} ({schema[nm].dtype!r})" if nm in schema else nm
              for nm in names]
    rows = [[_fmt_cell(d[nm][i]) for nm in names] for i in range(n)]
    widths = [max(len(h), *(len(r[j]) for r in rows)) if rows else len(h)
              for j, h in enumerate(header)]
    sep = "+" + "+".join("-" * (w + 2) for w in widths) + "+"
    out = [sep,
           "|" + "|".join(f" {h:<{w}} " for h, w in zip(header, widths)) + "|",
           sep]
    for r in rows:
        out.append("|" + "|".join(f" {c:<{w}} "
                                  for c, w in zip(r, widths)) + "|")
    out.append(sep)
    out.append(f"({n} rows shown)")
    return "\n".join(out)


def _fmt_cell(v) -> str:
    if v is None:
        return "None"
    s = str(v)
    return s if len(s) <= 30 else s[:27] + "..."
