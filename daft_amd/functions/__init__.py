"""Function library (ref: /root/reference/daft/functions/ and
src/daft-functions*)."""
from __future__ import annotations

from typing import Optional

from ..expressions.expressions import (Coalesce, Expression, _to_node,
                                       ScalarFn)
from ..schema import DataType


def coalesce(*exprs) -> Expression:
    return Expression(Coalesce([_to_node(e) for e in exprs]))


def row_number() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("row_number", None, None))


def rank() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("rank", None, None))


def dense_rank() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("dense_rank", None, None))


def w_first_value(expr) -> Expression:
    """first_value window function (ref: daft/functions/window.py:310)."""
    from ..physical.window import WindowFn
    from ..expressions.expressions import _to_node
    return Expression(WindowFn("first_value", _to_node(expr), None))


def w_last_value(expr) -> Expression:
    """last_value window function (SQL default frame: running last =
    current row; ref: daft/functions/window.py:371)."""
    from ..physical.window import WindowFn
    from ..expressions.expressions import _to_node
    return Expression(WindowFn("last_value", _to_node(expr), None))


def monotonically_increasing_id() -> Expression:
    raise NotImplementedError(
        "use DataFrame.add_monotonically_increasing_id()")


def columns_sum(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        out = out + _expr(e)
    return out


def columns_avg(*exprs) -> Expression:
    return columns_sum(*exprs) / float(len(exprs))


def columns_min(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        nxt = _expr(e)
        out = (out <= nxt).if_else(out, nxt)
    return out


def columns_max(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        nxt = _expr(e)
        out = (out >= nxt).if_else(out, nxt)
    return out


def _expr(e) -> Expression:
    from ..expressions.expressions import col
    return col(e) if isinstance(e, str) else e


def cosine_distance(a, b) -> Expression:
    return _expr(a).embedding.cosine_distance(_expr(b))


def uuid() -> Expression:
    import uuid as _uuid
    from ..series import Series

    def gen(s) -> Series:
        vals = [str(_uuid.uuid4()) for _ in range(len(s))]
        out = Series.from_pylist("uuid", vals, DataType.string())
        return out.to(s.device) if s.is_gpu() else out
    from ..expressions.expressions import ColumnRef
    raise NotImplementedError("uuid() requires a column context; "
                              "use df.add_monotonically_increasing_id")

from .spatial import great_circle_distance  # noqa: E402,F401


def file(expr) -> "Expression":
    """Wrap a path (string) or bytes column as File objects (ref
    capability: daft.functions.file / daft.File columns)."""
    from ..expressions.expressions import Expression, ScalarFn, _to_node
    from ..schema import DataType
    from ..series import Series
    from ..file import File

    def run(s):
        vals = s.cpu().to_pylist()
        objs = [None if v is None else File(v) for v in vals]
        return Series(s.name, DataType.python(), pyobjs=objs,
                      validity=None, length=len(vals))
    return Expression(ScalarFn("file", run, [_to_node(expr)],
                               DataType.python()))


def file_size(expr) -> "Expression":
    """Size in bytes of a File column (ref: daft/functions/file_.py:110)."""
    from ..expressions.expressions import Expression, ScalarFn, _to_node
    from ..schema import DataType
    from ..series import Series

    def run(s):
        out = [None if f is None else f.size() for f in s.pyobjs]
        return Series.from_pylist(s.name, out, DataType.int64())
    return Expression(ScalarFn("file_size", run, [_to_node(expr)],
                               DataType.int64()))

# free-function API surface (ref: daft/functions/__init__.py exports)
from .math import (  # noqa: E402,F401
    sin, cos, tan, cot, sec, csc, sinh, cosh, tanh, arcsin, arccos,
    arctan, arcsinh, arccosh, arctanh, arctan2, degrees, radians, exp,
    expm1, ln, log, log2, log10, log1p, sqrt, cbrt, hypot, sign, signum,
    negate, negative, pmod, power, factorial, e, pi, trunc, bitwise_and,
    bitwise_or, bitwise_xor, shift_left, shift_right, try_divide,
    random_int)
from .math import pow  # noqa: E402,F401,A004
from .temporal import (  # noqa: E402,F401
    year, month, day, dayofmonth, day_of_month, dayofyear, day_of_year,
    weekofyear, week_of_year, quarter, hour, minute, second, day_of_week,
    dayofweek, microsecond, millisecond, nanosecond, date_trunc, to_date,
    date_add, dateadd, date_sub, date_diff, datediff, datepart,
    make_date, last_day, next_day, add_months, months_between, strftime,
    date_format, from_unixtime, timestamp_seconds, timestamp_millis,
    timestamp_micros, to_unix_epoch, unix_date, date_from_unix_date,
    current_date, current_timestamp, current_timezone, total_hours,
    total_minutes, total_milliseconds, total_microseconds,
    total_nanoseconds, to_datetime)
from .strings_extra import (  # noqa: E402,F401
    to_snake_case, to_upper_snake_case, to_kebab_case,
    to_upper_kebab_case, to_camel_case, to_upper_camel_case,
    to_title_case, normalize, concat_ws, format, ascii_func, chr_func,
    space, translate, replace, split_part, substring_index,
    count_matches, regexp, regexp_count, regexp_extract,
    regexp_extract_all, regexp_replace, regexp_split,
    levenshtein_distance, damerau_levenshtein_distance,
    hamming_distance_str, jaro_similarity, jaro_winkler_similarity,
    jaccard_similarity, soundex)
from .misc import (  # noqa: E402,F401
    list_contains, list_distinct, list_join, list_sum, list_min,
    list_max, list_mean, list_count, list_chunk, list_slice, list_agg,
    to_list, list_agg_distinct, list_sort, list_append, list_flatten,
    list_bool_and, list_bool_or, list_map, list_filter, to_struct,
    struct, eq_null_safe, not_nan, try_cast, when, serialize,
    deserialize, try_deserialize, compress, decompress, try_compress,
    try_decompress, encode, decode, try_encode, try_decode,
    json_array_length, json_object_keys, json_tuple, parse_url,
    guess_mime_type, file_exists, file_path, median, percentile,
    approx_percentiles, var, product, string_agg, columns_mean,
    columns_sum, pearson_correlation, audio_file, audio_metadata,
    video_file, video_frames, video_keyframes, video_metadata,
    hdf5_file, hdf5_keys, hdf5_attrs, hdf5_metadata, run_process,
    llm_generate)
from .image import (  # noqa: E402,F401
    image_height, image_width, image_channel, image_mode,
    image_attribute, image_hash, convert_image, decode_image,
    encode_image, image_to_tensor)

from .aliases import (  # noqa: E402,F401
    lower, upper, capitalize, strip, lstrip, rstrip, reverse, length,
    length_bytes, contains, startswith, endswith, like, ilike, find,
    split, substr, left, right, lpad, rpad, repeat, concat,
    tokenize_encode, tokenize_decode, is_nan, is_inf, fill_nan, get,
    chunk, value_counts, explode, date, total_days, total_seconds, time,
    download, upload, resize, crop, image_file, image_file_metadata,
    decode_image_file, dot_product, euclidean_distance,
    cosine_similarity, hamming_distance, map_get, map_keys, map_values,
    to_utc_timestamp, from_utc_timestamp, convert_time_zone,
    convert_timezone, replace_time_zone, make_timestamp,
    make_timestamp_ltz, partition_days, partition_months,
    partition_years, partition_hours, partition_iceberg_bucket,
    partition_iceberg_truncate, seq, bin, conv, unnest, first_value,
    last_value, jq, extract_month_uuid7, extract_day_uuid7,
    extract_hour_uuid7, extract_minute_uuid7, resample)
from .aliases import slice  # noqa: E402,F401,A004
from .ai import (  # noqa: E402,F401
    embed_text, embed_image, classify_text, classify_image, prompt)
from .aliases import (  # noqa: F401  method-form free functions
    abs, any_value, approx_count_distinct, avg, between, bool_and,
    bool_or, cast, ceil, clip, count, count_distinct, fill_null, floor,
    hash, is_in, is_null, lag, lead, max, mean, min, minhash, not_null,
    over, round, simhash, skew, stddev, sum)
