"""daft_amd CLI (ref: /root/reference/daft/cli.py + src/daft-cli/).

    python -m daft_amd.cli sql "select * from t" --parquet t=path.parquet
    python -m daft_amd.cli bench --sf 1
    python -m daft_amd.cli schema path.parquet
"""
from __future__ import annotations

import argparse
import sys


def main(argv=None):
    ap = argparse.ArgumentParser(prog="daft_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    sq = sub.add_parser("sql", help="run a SQL query")
    sq.add_argument("query")
    sq.add_argument("--parquet", action="append", default=[],
                    help="name=path table bindings")
    sq.add_argument("--csv", action="append", default=[])
    sq.add_argument("--limit", type=int, default=20)

    sb = sub.add_parser("bench", help="run the TPC-H benchmark")
    sb.add_argument("--sf", type=float, default=1.0)
    sb.add_argument("--queries", default="all")

    ss = sub.add_parser("schema", help="print a file's schema")
    ss.add_argument("path")

    args = ap.parse_args(argv)

    if args.cmd == "sql":
        import daft_amd as daft
        from daft_amd.sql import register_table, sql
        for spec in args.parquet:
            name, path = spec.split("=", 1)
            register_table(name, daft.read_parquet(path))
        for spec in args.csv:
            name, path = spec.split("=", 1)
            register_table(name, daft.read_csv(path))
        sql(args.query).limit(args.limit).show(args.limit)
        return 0

    if args.cmd == "bench":
        import subprocess
        cmd = [sys.executable, "bench.py", "--sf", str(args.sf),
               "--queries", args.queries]
        return subprocess.call(cmd)

    if args.cmd == "schema":
        import daft_amd as daft
        fmt = "parquet" if args.path.endswith(".parquet") else "csv"
        df = daft.read_parquet(args.path) if fmt == "parquet" \
            else daft.read_csv(args.path)
        print(df.schema)
        return 0


if __name__ == "__main__":
    sys.exit(main())
