"""SQL frontend tests: basic SELECT coverage + all supported TPC-H SQL
queries vs the DataFrame implementations (ref pattern: tests/sql/)."""
import math

import pytest

import daft_amd as daft
from daft_amd import col
from daft_amd.sql import sql

from benchmarks.tpch import datagen, queries, queries_sql

SF = 0.01


@pytest.fixture(scope="module")
def T():
    return datagen.dataframes(SF, device="cpu")


def test_sql_basic_select():
    df = daft.from_pydict({"a": [1, 2, 3], "b": ["x", "y", "z"]})
    out = sql("select a + 1 as a1, b from df where a >= 2 order by a1 desc")
    assert out.to_pydict() == {"a1": [4, 3], "b": ["z", "y"]}


def test_sql_aggregate():
    df = daft.from_pydict({"g": ["a", "a", "b"], "v": [1.0, 2.0, 3.0]})
    out = sql("select g, sum(v) as s, count(*) as n from df "
              "group by g order by g")
    assert out.to_pydict() == {"g": ["a", "b"], "s": [3.0, 3.0], "n": [2, 1]}


def test_sql_join():
    a = daft.from_pydict({"k": [1, 2, 3], "v": [10, 20, 30]})
    b = daft.from_pydict({"k": [2, 3, 4], "w": [200, 300, 400]})
    out = sql("select a.k, v, w from a join b on a.k = b.k order by a.k")
    assert out.to_pydict() == {"k": [2, 3], "v": [20, 30], "w": [200, 300]}


def test_sql_comma_join_extraction():
    a = daft.from_pydict({"ak": [1, 2], "v": [10, 20]})
    b = daft.from_pydict({"bk": [1, 2], "w": [100, 200]})
    out = sql("select v, w from a, b where ak = bk and w > 100")
    assert out.to_pydict() == {"v": [20], "w": [200]}


def test_sql_case_when():
    df = daft.from_pydict({"x": [1, 5, 10]})
    out = sql("select case when x < 3 then 'lo' when x < 7 then 'mid' "
              "else 'hi' end as c from df")
    assert out.to_pydict()["c"] == ["lo", "mid", "hi"]


def test_sql_having():
    df = daft.from_pydict({"g": ["a", "a", "b"], "v": [1, 2, 3]})
    out = sql("select g, sum(v) as s from df group by g having sum(v) > 2 "
              "order by g")
    assert out.to_pydict() == {"g": ["a", "b"], "s": [3, 3]}


def test_sql_in_subquery():
    a = daft.from_pydict({"k": [1, 2, 3, 4]})
    b = daft.from_pydict({"k": [2, 4]})
    out = sql("select k from a where k in (select k from b) order by k")
    assert out.to_pydict()["k"] == [2, 4]
    out = sql("select k from a where k not in (select k from b) order by k")
    assert out.to_pydict()["k"] == [1, 3]


def test_sql_exists_correlated():
    a = daft.from_pydict({"k": [1, 2, 3]})
    b = daft.from_pydict({"fk": [2, 3, 3]})
    out = sql("select k from a where exists "
              "(select * from b where b.fk = a.k) order by k")
    assert out.to_pydict()["k"] == [2, 3]
    out = sql("select k from a where not exists "
              "(select * from b where b.fk = a.k) order by k")
    assert out.to_pydict()["k"] == [1]


def test_sql_scalar_subquery():
    a = daft.from_pydict({"v": [1, 2, 3, 4]})
    out = sql("select v from a where v > (select avg(v) from a) order by v")
    assert out.to_pydict()["v"] == [3, 4]


def test_sql_cte():
    df = daft.from_pydict({"v": [1, 2, 3]})
    out = sql("with t as (select v * 2 as w from df) "
              "select w from t where w > 2 order by w")
    assert out.to_pydict()["w"] == [4, 6]


def test_sql_expr():
    from daft_amd.sql import sql_expr
    df = daft.from_pydict({"a": [1, 2, 3]})
    out = df.where(sql_expr("a >= 2")).to_pydict()
    assert out["a"] == [2, 3]


def _norm_rows(d):
    rows = list(zip(*d.values()))
    key = lambda r: tuple(repr(x) for x in r if not isinstance(x, float))
    return sorted(rows, key=key)


@pytest.mark.parametrize("qi", queries_sql.SUPPORTED)
def test_tpch_sql_matches_dataframe(qi, T):
    got = queries_sql.run_sql_query(qi, T, sf=SF).to_pydict()
    want = queries.run_query(qi, T, sf=SF).to_pydict()
    assert len(got) == len(want), (list(got), list(want))
    g_rows, w_rows = _norm_rows(got), _norm_rows(want)
    assert len(g_rows) == len(w_rows), f"rows {len(g_rows)} vs {len(w_rows)}"
    for gr, wr in zip(g_rows, w_rows):
        for gx, wx in zip(gr, wr):
            if isinstance(wx, float):
                assert gx is not None and (
                    math.isclose(gx, wx, rel_tol=1e-9, abs_tol=1e-6) or
                    (math.isnan(gx) and math.isnan(wx))), (gx, wx)
            else:
                assert gx == wx, (gx, wx)


def test_sql_window_functions():
    """OVER clause: rank fns, running/partition aggregates, lag,
    first_value, ROWS BETWEEN frames."""
    df = daft.from_pydict({"g": ["a", "a", "b", "b", "b"],
                           "v": [3, 1, 2, 9, 8]})
    o = daft.sql("""select g, v,
        row_number() over (partition by g order by v) as rn,
        sum(v) over (partition by g order by v) as run,
        sum(v) over (partition by g) as tot,
        lag(v, 1) over (partition by g order by v) as prev,
        first_value(v) over (partition by g order by v) as fv
      from df order by g, v""").to_pydict()
    assert o["rn"] == [1, 2, 1, 2, 3]
    assert o["run"] == [1.0, 4.0, 2.0, 10.0, 19.0]
    assert o["tot"] == [4.0, 4.0, 19.0, 19.0, 19.0]
    assert o["prev"] == [None, 1, None, 2, 8]
    assert o["fv"] == [1, 1, 2, 2, 2]
    o2 = daft.sql("select g, v, sum(v) over (partition by g order by v "
                  "rows between 1 preceding and current row) as m "
                  "from df order by g, v").to_pydict()
    assert o2["m"] == [1.0, 4.0, 2.0, 10.0, 17.0]


def test_sql_set_operations():
    a = daft.from_pydict({"x": [1, 2, 3, 3]})
    b = daft.from_pydict({"x": [2, 3, 4]})
    assert sorted(daft.sql("select x from a union select x from b")
                  .to_pydict()["x"]) == [1, 2, 3, 4]
    assert sorted(daft.sql("select x from a union all select x from b")
                  .to_pydict()["x"]) == [1, 2, 2, 3, 3, 3, 4]
    assert sorted(daft.sql("select x from a intersect select x from b")
                  .to_pydict()["x"]) == [2, 3]
    assert sorted(daft.sql("select x from a except select x from b")
                  .to_pydict()["x"]) == [1]
    out = daft.sql("select x from a union select x from b "
                   "order by x desc limit 2").to_pydict()["x"]
    assert out == [4, 3]


def test_sql_scalar_function_registry():
    """nullif/greatest/least/ifnull plus fallback into the 318-name
    daft.functions registry (sin, levenshtein_distance, ...)."""
    import math
    df = daft.from_pydict({"a": [1.0, 2.0, 0.0], "b": [1.0, 5.0, 7.0]})
    assert daft.sql("select nullif(a, b) as n from df") \
        .to_pydict()["n"] == [None, 2, 0]
    assert daft.sql("select greatest(a, b) as g from df") \
        .to_pydict()["g"] == [1, 5, 7]
    assert daft.sql("select least(a, b) as l from df") \
        .to_pydict()["l"] == [1, 2, 0]
    assert daft.sql("select ifnull(nullif(a, b), 99) as f from df") \
        .to_pydict()["f"] == [99, 2, 0]
    out = daft.sql("select sin(a) as s from df").to_pydict()["s"]
    assert abs(out[1] - math.sin(2)) < 1e-12
    assert daft.sql("select levenshtein_distance('kitten', 'sitting') "
                    "as d from df limit 1").to_pydict()["d"] == [3]


def test_sql_catalog_and_explain():
    from daft_amd.sql import SQLCatalog
    cat = SQLCatalog({"t": daft.from_pydict({"a": [1, 2]})})
    assert daft.sql("select sum(a) as s from t",
                    catalog=cat).to_pydict()["s"] == [3]
    plan = daft.sql("explain select a from t where a > 1",
                    catalog=cat).to_pydict()["plan"]
    assert any("Filter" in l for l in plan)


def test_sql_values_clause():
    out = sql("SELECT x, y FROM (VALUES (1, 'a'), (2, 'b')) AS v(x, y) "
              "WHERE x > 1").to_pydict()
    assert out == {"x": [2], "y": ["b"]}
    out2 = sql("SELECT * FROM (VALUES (1), (2)) v").to_pydict()
    assert out2 == {"column1": [1, 2]}


def test_sql_select_without_from():
    out = sql("SELECT 1+1 AS x, 'hi' AS s").to_pydict()
    assert out == {"x": [2], "s": ["hi"]}


def test_sql_group_by_all():
    df = daft.from_pydict({"g": ["a", "a", "b"], "v": [1, 2, 3]})
    out = sql("SELECT g, sum(v) AS s FROM df GROUP BY ALL ORDER BY g") \
        .to_pydict()
    assert out == {"g": ["a", "b"], "s": [3, 3]}


def test_sql_qualify():
    df = daft.from_pydict({"g": ["a", "a", "b"], "v": [1, 2, 3]})
    out = sql("SELECT g, v, row_number() OVER (PARTITION BY g ORDER BY v "
              "DESC) AS rn FROM df QUALIFY rn = 1 ORDER BY g").to_pydict()
    assert out == {"g": ["a", "b"], "v": [2, 3], "rn": [1, 1]}
    out2 = sql("SELECT g, v FROM df QUALIFY row_number() OVER "
               "(ORDER BY v DESC) = 1").to_pydict()
    assert out2 == {"g": ["b"], "v": [3]}


def test_sql_table_function_read_parquet(tmp_path):
    df = daft.from_pydict({"a": [1, 2, 3]})
    df.write_parquet(str(tmp_path / "t"))
    out = sql(f"SELECT sum(a) AS s FROM "
              f"read_parquet('{tmp_path}/t/**/*.parquet')").to_pydict()
    assert out == {"s": [6]}


def test_sql_subquery_column_alias_list():
    df = daft.from_pydict({"a": [1, 2, 3]})
    out = sql("SELECT x * 10 AS y FROM (SELECT a FROM df) AS q(x) "
              "WHERE x >= 2 ORDER BY y").to_pydict()
    assert out == {"y": [20, 30]}


def test_sql_interval_on_date_columns():
    t = daft.from_pydict({"d": ["2024-01-31", "2024-02-01"]})
    out = sql("SELECT CAST(d AS DATE) + INTERVAL '1 month' AS m, "
              "CAST(d AS DATE) - INTERVAL '1 year' AS y, "
              "CAST(d AS DATE) + INTERVAL '3 day' AS dd FROM t").to_pydict()
    import datetime as dt
    assert out["m"] == [dt.date(2024, 2, 29), dt.date(2024, 3, 1)]
    assert out["y"] == [dt.date(2023, 1, 31), dt.date(2023, 2, 1)]
    assert out["dd"] == [dt.date(2024, 2, 3), dt.date(2024, 2, 4)]


def test_sql_datediff_units():
    t = daft.from_pydict({"d": ["2024-01-31"]})
    out = sql("SELECT DATEDIFF('day', CAST(d AS DATE), "
              "CAST('2024-03-01' AS DATE)) AS dd, "
              "DATEDIFF('month', CAST(d AS DATE), "
              "CAST('2024-06-15' AS DATE)) AS dm, "
              "DATEDIFF('year', CAST(d AS DATE), "
              "CAST('2026-01-01' AS DATE)) AS dy FROM t").to_pydict()
    assert out == {"dd": [30], "dm": [5], "dy": [2]}


def test_sql_window_over_group_by():
    """Windows over aggregated results: rank by sum, percent-of-total
    with a window embedded in arithmetic, hidden window aggregates
    (ref: daft-sql window-over-aggregate)."""
    t = daft.from_pydict({"g": ["a", "a", "b", "b", "c"],
                          "v": [1, 2, 3, 4, 10]})
    out = sql("SELECT g, sum(v) AS s, RANK() OVER (ORDER BY sum(v) DESC)"
              " AS r FROM t GROUP BY g ORDER BY g").to_pydict()
    assert out == {"g": ["a", "b", "c"], "s": [3, 7, 10], "r": [3, 2, 1]}
    out2 = sql("SELECT g, ROW_NUMBER() OVER (ORDER BY max(v)) AS rn "
               "FROM t GROUP BY g ORDER BY g").to_pydict()
    assert out2 == {"g": ["a", "b", "c"], "rn": [1, 2, 3]}
    out3 = sql("SELECT g, sum(v) AS s, sum(v) * 100.0 / "
               "SUM(sum(v)) OVER () AS pct FROM t GROUP BY g "
               "ORDER BY g").to_pydict()
    assert out3["pct"] == [15.0, 35.0, 50.0]


def test_sql_window_embedded_in_expression():
    t = daft.from_pydict({"g": ["a", "a", "b"], "v": [1, 3, 4]})
    out = sql("SELECT g, v, v * 100.0 / SUM(v) OVER (PARTITION BY g) AS "
              "pct FROM t ORDER BY g, v").to_pydict()
    assert out["pct"] == [25.0, 75.0, 100.0]


def test_sql_statements():
    """Non-SELECT statements (ref: daft-sql statement.rs — ShowTables,
    Use, CreateTable, Describe)."""
    t = daft.from_pydict({"a": [1, 2]})
    assert sql("CREATE TABLE tmp_ct AS SELECT a * 2 AS b FROM t") \
        .to_pydict() == {"b": [2, 4]}
    assert sql("SELECT * FROM tmp_ct").to_pydict() == {"b": [2, 4]}
    assert "tmp_ct" in sql("SHOW TABLES").to_pydict()["table"]
    assert sql("SHOW TABLES LIKE 'tmp%'").to_pydict()["table"] == \
        ["tmp_ct"]
    d = sql("DESCRIBE tmp_ct").to_pydict()
    assert d == {"column_name": ["b"], "type": ["Int64"]}
    d2 = sql("DESCRIBE SELECT a, a * 1.5 AS f FROM t").to_pydict()
    assert d2["type"] == ["Int64", "Float64"]
    sql("DROP TABLE tmp_ct")
    with pytest.raises(Exception):
        sql("SELECT * FROM tmp_ct").collect()
    sql("DROP TABLE IF EXISTS tmp_ct")      # no raise
    assert sql("USE cat.ns").to_pydict() == {"ok": [True]}


def test_sql_scalar_subquery_in_select():
    """Scalar subqueries in the SELECT list (ref: planner.rs
    SQLExpr::Subquery): uncorrelated inline, correlated decorrelate to
    a grouped LEFT join (missing keys -> NULL)."""
    t = daft.from_pydict({"k": [1, 2, 3], "v": [10, 20, 30]})
    u = daft.from_pydict({"k": [1, 2], "w": [5, 7]})
    out = sql("SELECT k, (SELECT max(w) FROM u) AS mw FROM t "
              "ORDER BY k").to_pydict()
    assert out == {"k": [1, 2, 3], "mw": [7, 7, 7]}
    out2 = sql("SELECT k, (SELECT w FROM u WHERE u.k = t.k) AS w "
               "FROM t ORDER BY k").to_pydict()
    assert out2 == {"k": [1, 2, 3], "w": [5, 7, None]}
    out3 = sql("SELECT k, v + (SELECT sum(w) FROM u WHERE u.k = t.k) "
               "AS vw FROM t ORDER BY k").to_pydict()
    assert out3["vw"] == [15, 27, None]
    out4 = sql("SELECT *, (SELECT max(w) FROM u WHERE u.k = t.k) AS mw "
               "FROM t ORDER BY k").to_pydict()
    assert list(out4.keys()) == ["k", "v", "mw"]


def test_sql_exists_range_correlation():
    """EXISTS with range (non-equality) correlation decorrelates via a
    row-id semi join (duckdb-style general unnesting)."""
    t = daft.from_pydict({"k": [1, 2, 3], "v": [10, 20, 30]})
    u = daft.from_pydict({"k": [1, 2, 2], "w": [5, 25, 19]})
    out = sql("SELECT k FROM t WHERE EXISTS (SELECT 1 FROM u WHERE "
              "u.k = t.k AND u.w < t.v) ORDER BY k").to_pydict()
    assert out == {"k": [1, 2]}
    out2 = sql("SELECT k FROM t WHERE NOT EXISTS (SELECT 1 FROM u WHERE "
               "u.k = t.k AND u.w < t.v) ORDER BY k").to_pydict()
    assert out2 == {"k": [3]}
    # pure range correlation (no equality keys): constant-key join
    out3 = sql("SELECT k FROM t WHERE EXISTS (SELECT 1 FROM u WHERE "
               "u.w >= t.v) ORDER BY k").to_pydict()
    assert out3 == {"k": [1, 2]}
