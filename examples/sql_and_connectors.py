"""Tour of the SQL frontend extras and the python connector APIs.

Run: python examples/sql_and_connectors.py   (CPU or GPU; no network)
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import daft_amd as daft  # noqa: E402
from daft_amd import col  # noqa: E402


def sql_tour():
    sales = daft.from_pydict({
        "region": ["east", "east", "west", "west", "north"],
        "amount": [10.0, 20.0, 5.0, 25.0, 40.0],
    })

    # windows over GROUP BY results + percent-of-total
    print(daft.sql("""
        SELECT region, sum(amount) AS total,
               RANK() OVER (ORDER BY sum(amount) DESC) AS rnk,
               sum(amount) * 100.0 / SUM(sum(amount)) OVER () AS pct
        FROM sales GROUP BY region ORDER BY rnk
    """).to_pydict())

    # VALUES, QUALIFY, GROUP BY ALL
    print(daft.sql("""
        SELECT x, y FROM (VALUES (1, 'a'), (2, 'b'), (3, 'b')) v(x, y)
        QUALIFY ROW_NUMBER() OVER (PARTITION BY y ORDER BY x DESC) = 1
    """).to_pydict())
    print(daft.sql("SELECT region, count(*) AS n FROM sales "
                   "GROUP BY ALL ORDER BY region").to_pydict())

    # FROM-position table functions over files
    with tempfile.TemporaryDirectory() as d:
        sales.write_parquet(d + "/t")
        print(daft.sql(f"SELECT count(*) AS rows FROM "
                       f"read_parquet('{d}/t/**/*.parquet')").to_pydict())


def connector_tour():
    from daft_amd.io import (DataSink, DataSource, DataSourceTask,
                             WriteResult, read_source)
    from daft_amd.recordbatch import RecordBatch
    from daft_amd.schema import DataType, Field, Schema
    from daft_amd.series import Series

    sch = Schema([Field("n", DataType.int64())])

    class Task(DataSourceTask):
        def __init__(self, lo, hi):
            self.lo, self.hi = lo, hi

        @property
        def schema(self):
            return sch

        def read(self):
            yield RecordBatch([Series.from_pylist(
                "n", list(range(self.lo, self.hi)), DataType.int64())],
                num_rows=self.hi - self.lo)

    class Source(DataSource):
        name = property(lambda self: "demo-range")
        schema = property(lambda self: sch)

        def get_tasks(self, pushdowns=None):
            for lo in range(0, 30, 10):
                yield Task(lo, lo + 10)

    df = read_source(Source()).where(col("n") % 7 == 0)
    print(df.to_pydict())

    class PrintSink(DataSink):
        def write(self, batch):
            n = len(batch)
            return WriteResult(result=n, rows_written=n)

        def finalize(self, results):
            return {"batches": len(results),
                    "rows": sum(r.rows_written for r in results)}

    df.write_sink(PrintSink())


def wide_decimal_tour():
    import decimal
    decimal.getcontext().prec = 50
    D = decimal.Decimal
    df = daft.from_pydict({
        "k": ["a", "b", "a"],
        "amt": [D("12345678901234567890.123456789"),
                D("-1.000000001"),
                D("99999999999999999999.999999999")],
    })
    print([str(f.dtype) for f in df.schema])          # Decimal128(29,9)
    print(df.groupby("k").agg(col("amt").sum().alias("s"))
            .sort("k").to_pydict())                   # exact i128 sums


if __name__ == "__main__":
    sql_tour()
    connector_tour()
    wide_decimal_tour()
    print("ok")
