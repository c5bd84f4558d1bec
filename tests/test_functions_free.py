"""Free-function API sweep (capability of daft/functions/__init__.py
exports: math, temporal, string, list, serde, json, aggregates)."""
import datetime as dt
import math

import daft_amd as daft
from daft_amd import col
import daft_amd.functions as F


def test_free_function_sweep():

    df = daft.from_pydict({"x": [0.5, 1.0, None], "y": [2.0, 3.0, 4.0],
                           "i": [5, 6, 7], "s": ["hello world", "FooBar_baz", None],
                           "d": [dt.date(2024, 1, 15), dt.date(2024, 2, 29), dt.date(2023, 12, 31)],
                           "l": [[3, 1, 2], [5], None],
                           "j": ['{"a": 1, "b": [1,2]}', '[1,2,3]', None],
                           "u": ["https://example.com/p?q=1#f", None, None]})

    o = df.select(F.sin(col("x")).alias("sin"), F.log(col("y"), 10).alias("lg"),
                  F.hypot(col("x"), col("y")).alias("h"),
                  F.bitwise_and(col("i"), 3).alias("ba"),
                  F.shift_left(col("i"), 1).alias("sl"),
                  F.try_divide(col("y"), col("x") - 0.5).alias("td"),
                  F.sign(col("x")).alias("sg")).to_pydict()
    assert abs(o["sin"][0] - math.sin(0.5)) < 1e-12
    assert abs(o["lg"][1] - math.log10(3)) < 1e-12
    assert o["ba"] == [1, 2, 3] and o["sl"] == [10, 12, 14]
    assert o["td"][0] is None and abs(o["td"][1] - 6.0) < 1e-9

    t = df.select(F.year(col("d")).alias("y"), F.last_day(col("d")).alias("ld"),
                  F.add_months(col("d"), 1).alias("am"),
                  F.next_day(col("d"), "monday").alias("nd"),
                  F.strftime(col("d"), "%Y/%m").alias("sf"),
                  F.to_unix_epoch(col("d")).alias("ux"),
                  F.date_diff(col("d"), col("d")).alias("dd")).to_pydict()
    assert t["y"] == [2024, 2024, 2023]
    assert t["ld"][0] == dt.date(2024, 1, 31)
    assert t["am"][1] == dt.date(2024, 3, 29)
    assert t["sf"][0] == "2024/01" and t["dd"] == [0, 0, 0]
    assert t["ux"][0] == int(dt.datetime(2024, 1, 15).timestamp()) - int(dt.datetime(1970,1,1).timestamp())

    s = df.select(F.to_snake_case(col("s")).alias("sn"),
                  F.levenshtein_distance(col("s"), "hello word").alias("lv"),
                  F.soundex(col("s")).alias("sx"),
                  F.split_part(col("s"), " ", 2).alias("sp"),
                  F.regexp_extract(col("s"), r"(\w+)$", 1).alias("re"),
                  F.concat_ws("-", col("s"), col("i")).alias("cw")).to_pydict()
    assert s["sn"][1] == "foo_bar_baz", s["sn"]
    assert s["lv"][0] == 1
    assert s["sp"][0] == "world"
    assert s["re"][0] == "world"
    assert s["cw"][2] == "7"

    l = df.select(F.list_sort(col("l")).alias("ls"),
                  F.list_flatten(daft.lit([[1, 2], [3]])).alias("lf"),
                  F.list_append(col("l"), 9).alias("la")).to_pydict()
    assert l["ls"][0] == [1, 2, 3]
    assert l["la"][1] == [5, 9]

    j = df.select(F.json_array_length(col("j")).alias("n"),
                  F.json_object_keys(col("j")).alias("k"),
                  F.parse_url(col("u"), "host").alias("h")).to_pydict()
    assert j["n"][1] == 3 and j["k"][0] == ["a", "b"]
    assert j["h"][0] == "example.com"

    w = df.select(F.when(col("i") > 6, "big").when(col("i") > 5, "mid")
                  .otherwise("small").alias("w")).to_pydict()["w"]
    assert w == ["small", "mid", "big"], w

    c = df.select(F.compress(col("s")).alias("z")).select(
        F.decompress(col("z")).alias("u2")).to_pydict()["u2"]
    assert c[0] == b"hello world"

    agg = df.agg(F.median(col("y")).alias("med"),
                 F.var(col("y")).alias("v"),
                 F.product(col("y")).alias("p"),
                 F.string_agg(col("s"), "|").alias("sa"),
                 F.pearson_correlation(col("y"), col("y")).alias("r")).to_pydict()
    assert abs(agg["med"][0] - 3.0) < 0.1
    assert abs(agg["p"][0] - 24.0) < 1e-9
    assert abs(agg["r"][0] - 1.0) < 1e-9
    assert agg["sa"][0].count("|") == 2

    ens = df.select(F.eq_null_safe(col("x"), col("x")).alias("e")).to_pydict()["e"]
    assert ens == [True, True, True]
    print("free function sweep OK")


def test_jq_filters():
    """jq-style JSON filters (ref: daft-functions-json jaq filters):
    fields, indices, iteration, pipes, // defaults."""
    import daft_amd as daft
    from daft_amd import col
    from daft_amd.functions import jq
    df = daft.from_pydict({"j": [
        '{"u": {"id": 7, "tags": ["a", "b"]}, "n": 1}',
        '{"u": {"id": 8, "tags": []}}',
        'not json',
    ]})
    out = df.select(
        jq(col("j"), ".u.id").alias("id"),
        jq(col("j"), ".u.tags[1]").alias("t1"),
        jq(col("j"), ".u | .tags[]").alias("tags"),
        jq(col("j"), ".n // .u.id").alias("d"),
    ).to_pydict()
    assert out["id"] == ["7", "8", None]
    assert out["t1"] == ["b", None, None]
    assert out["tags"] == ['["a", "b"]', None, None]
    assert out["d"] == ["1", "8", None]


def test_list_sort_flatten_reverse():
    import daft_amd as daft
    from daft_amd import col
    a = daft.from_pydict({"l": [[3, 1, None, 2], [5, 4], None]})
    out = a.select(col("l").list.sort().alias("s"),
                   col("l").list.sort(desc=True).alias("d"),
                   col("l").list.reverse().alias("r")).to_pydict()
    assert out["s"] == [[1, 2, 3, None], [4, 5], None]
    assert out["d"] == [[3, 2, 1, None], [5, 4], None]
    assert out["r"] == [[2, None, 1, 3], [4, 5], None]
    b = daft.from_pydict({"l": [[[1, 2], [3]], [[4]], None]})
    fo = b.select(col("l").list.flatten().alias("f")).to_pydict()
    assert fo["f"] == [[1, 2, 3], [4], None]


def test_bpe_tokenizer_json_roundtrip(tmp_path):
    """tokenizers-JSON BPE loading + encode/decode through the expression
    API (CPU oracle; the GPU kernel is compared in test_gpu.py)."""
    import json
    from daft_amd.functions.tokenize import _bytes_to_unicode
    b2u = _bytes_to_unicode()
    vocab = {b2u[b]: b for b in range(256)}
    h, e = b2u[ord("h")], b2u[ord("e")]
    vocab[h + e] = 256
    doc = {"model": {"type": "BPE", "vocab": vocab,
                     "merges": [f"{h} {e}"]}}
    p = tmp_path / "tok.json"
    p.write_text(json.dumps(doc))
    import daft_amd as daft
    from daft_amd import col
    df = daft.from_pydict({"t": ["he", "heh", None]})
    enc = df.select(col("t").str.tokenize_encode(f"bpe:{p}").alias("ids"))
    ids = enc.to_pydict()["ids"]
    assert ids == [[256], [256, vocab[h]], None]
    dec = enc.select(daft.functions.tokenize_decode(
        col("ids"), f"bpe:{p}").alias("t")).to_pydict()
    assert dec["t"] == ["he", "heh", None]


def test_make_timestamp_and_conv_builtins():
    """Regression: module-level max/abs free functions must not shadow
    the builtins used inside make_timestamp/conv."""
    import daft_amd as daft
    import daft_amd.functions as F
    from daft_amd import col
    import datetime
    df = daft.from_pydict({"y": [2024], "mo": [3], "d": [1], "h": [10],
                           "mi": [5], "s": [6], "x": [255, ]})
    t = df.select(F.make_timestamp(col("y"), col("mo"), col("d"),
                                   col("h"), col("mi"), col("s"))
                  .alias("t")).to_pydict()["t"]
    assert t == [datetime.datetime(2024, 3, 1, 10, 5, 6)]
    hx = df.select(F.conv(col("x"), 10, 16).alias("h")).to_pydict()["h"]
    assert hx == ["ff"]
    # the new free-function forms coexist with python builtins
    out = df.select(F.abs(col("x") * -1).alias("a")).to_pydict()
    assert out["a"] == [255]
    m = df.agg(F.max(col("x")).alias("m")).to_pydict()
    assert m["m"] == [255]
