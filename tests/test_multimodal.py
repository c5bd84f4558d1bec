"""Multimodal path tests (CPU tier): minhash, image decode/resize/tensor,
tokenize, url_download, AI functions (ref pattern: tests/series/image,
tests/functions in the reference)."""
import io
import os

import numpy as np
import pytest
import torch

import daft_amd as daft
from daft_amd import DataType, Series, col


def test_minhash_determinism_and_similarity():
    docs = [
        "the quick brown fox jumps over the lazy dog",
        "the quick brown fox jumps over the lazy cat",
        "completely different words entirely unrelated text here",
        None,
    ]
    df = daft.from_pydict({"t": docs})
    out = df.select(col("t").minhash(64, ngram_size=2).alias("mh")) \
        .to_pydict()["mh"]
    assert out[3] is None
    assert len(out[0]) == 64
    # re-run: deterministic
    out2 = df.select(col("t").minhash(64, ngram_size=2).alias("mh")) \
        .to_pydict()["mh"]
    assert out == out2
    sim01 = sum(a == b for a, b in zip(out[0], out[1])) / 64
    sim02 = sum(a == b for a, b in zip(out[0], out[2])) / 64
    assert sim01 > sim02, (sim01, sim02)
    assert sim01 > 0.4


def _png_bytes(h, w, color):
    from PIL import Image
    arr = np.full((h, w, 3), color, dtype=np.uint8)
    buf = io.BytesIO()
    Image.fromarray(arr, "RGB").save(buf, format="PNG")
    return buf.getvalue()


def test_image_decode_resize_tensor():
    imgs = [_png_bytes(10, 20, 100), _png_bytes(32, 16, 200), None]
    df = daft.from_pydict({"b": imgs})
    out = (df.with_column("img", col("b").image.decode())
           .with_column("small", col("img").image.resize(8, 8))
           .with_column("t", col("small").image.to_tensor())
           .select("small", "t"))
    d = out.to_pydict()
    assert d["small"][2] is None
    assert len(d["small"][0]) == 8 * 8 * 3
    assert all(v == 100 for v in d["small"][0])
    t = d["t"][1]
    assert t.shape == (3, 8, 8)
    assert abs(float(t[0, 0, 0]) - 200 / 255) < 1e-5


def test_image_encode_roundtrip():
    imgs = [_png_bytes(4, 4, 42)]
    df = daft.from_pydict({"b": imgs})
    out = df.select(col("b").image.decode().image.encode("PNG").alias("e")) \
        .to_pydict()["e"]
    from PIL import Image
    arr = np.asarray(Image.open(io.BytesIO(out[0])))
    assert arr.shape == (4, 4, 3) and (arr == 42).all()


def test_image_crop():
    imgs = [_png_bytes(10, 10, 7)]
    df = daft.from_pydict({"b": imgs})
    out = df.select(col("b").image.decode().image.resize(10, 10)
                    .image.crop(2, 2, 4, 4).alias("c")).to_pydict()["c"]
    assert len(out[0]) == 4 * 4 * 3


def test_tokenize():
    df = daft.from_pydict({"t": ["hello world", "hello", None]})
    out = df.select(col("t").str.tokenize_encode("simple").alias("tok")) \
        .to_pydict()["tok"]
    assert out[2] is None
    assert out[0][0] == out[1][0]  # same word -> same token
    out_b = df.select(col("t").str.tokenize_encode("bytes").alias("tok")) \
        .to_pydict()["tok"]
    assert out_b[0][:5] == [104, 101, 108, 108, 111]


def test_url_download(tmp_path):
    p1 = tmp_path / "a.bin"
    p1.write_bytes(b"hello")
    df = daft.from_pydict({"u": [str(p1), None]})
    out = df.select(col("u").url.download().alias("d")).to_pydict()["d"]
    assert out == [b"hello", None]


def test_url_upload(tmp_path):
    df = daft.from_pydict({"d": [b"abc", b"def"], "name": ["x", "y"]})
    out = df.select(col("d").url.upload(str(tmp_path), col("name"))
                    .alias("p")).to_pydict()["p"]
    assert open(out[0], "rb").read() == b"abc"


def test_embed_text_hash_provider():
    from daft_amd.functions.ai import embed_text
    df = daft.from_pydict({"t": ["alpha beta", "alpha beta", "gamma"]})
    out = df.select(embed_text(col("t"), dimensions=64).alias("e")) \
        .to_pydict()["e"]
    assert out[0] == out[1]
    assert len(out[0]) == 64
    assert out[0] != out[2]


def test_classify_text():
    from daft_amd.functions.ai import classify_text
    df = daft.from_pydict({"t": ["alpha alpha alpha", "beta beta"]})
    out = df.select(classify_text(col("t"), ["alpha", "beta"]).alias("c")) \
        .to_pydict()["c"]
    assert out == ["alpha", "beta"]


def test_embed_image_torch_provider():
    from daft_amd.functions.ai import embed_image
    imgs = [_png_bytes(16, 16, 10), _png_bytes(16, 16, 250)]
    df = daft.from_pydict({"b": imgs})
    out = (df.with_column("t", col("b").image.decode()
                          .image.resize(32, 32).image.to_tensor())
           .select(embed_image(col("t"), provider="torch", dimensions=32,
                               image_size=32).alias("e"))
           .to_pydict()["e"])
    assert len(out[0]) == 32
    assert out[0] != out[1]


def test_cosine_distance():
    df = daft.from_pydict({
        "a": [[1.0, 0.0], [0.0, 1.0]],
        "b": [[1.0, 0.0], [1.0, 0.0]],
    })
    a = col("a").cast(DataType.embedding(DataType.float32(), 2))
    b = col("b").cast(DataType.embedding(DataType.float32(), 2))
    out = df.select(a.embedding.cosine_distance(b).alias("d")) \
        .to_pydict()["d"]
    assert out[0] == pytest.approx(0.0, abs=1e-6)
    assert out[1] == pytest.approx(1.0, abs=1e-6)


def test_list_extras():
    df = daft.from_pydict({"l": [[1, 2, 2, 3], [5], None]})
    out = df.select(
        col("l").list.distinct().alias("d"),
        col("l").list.contains(2).alias("c"),
        col("l").list.chunk(2).alias("ch"),
        col("l").list.slice(1, 3).alias("sl"),
    ).to_pydict()
    assert out["d"] == [[1, 2, 3], [5], None]
    assert out["c"] == [True, False, None]
    assert out["ch"] == [[[1, 2], [2, 3]], [[5]], None]
    assert out["sl"] == [[2, 2], [], None]
    vc = df.select(col("l").list.value_counts().alias("v")).to_pydict()["v"]
    assert {d["value"]: d["count"] for d in vc[0]} == {1: 1, 2: 2, 3: 1}


def test_prompt_echo():
    from daft_amd.functions.ai import prompt
    df = daft.from_pydict({"t": ["world", None]})
    out = df.select(prompt(col("t"), template="hello {input}").alias("p")) \
        .to_pydict()["p"]
    assert out == ["hello world", None]


def test_simhash():
    docs = ["the quick brown fox jumps over the lazy dog",
            "the quick brown fox jumps over the lazy cat",
            "entirely different words and other content here", None, "ab"]
    df = daft.from_pydict({"t": docs})
    out = df.select(col("t").simhash(4).alias("h")).to_pydict()["h"]
    assert out[3] is None
    assert out[4] == 0  # shorter than ngram
    assert out == df.select(col("t").simhash(4).alias("h")) \
        .to_pydict()["h"]  # deterministic

    def ham(a, b):
        return bin(a ^ b).count("1")
    assert ham(out[0], out[1]) < ham(out[0], out[2])


def test_image_to_mode():
    arr = np.zeros((4, 4, 3), np.uint8)
    arr[..., 0] = 200
    arr[..., 1] = 100
    arr[..., 2] = 50
    import io as _io
    from PIL import Image
    b = _io.BytesIO()
    Image.fromarray(arr, "RGB").save(b, format="PNG")
    df = daft.from_pydict({"b": [b.getvalue()]})
    img = col("b").image.decode().image.resize(4, 4)
    l = df.select(img.image.to_mode("L").alias("l")).to_pydict()["l"][0]
    want = np.asarray(Image.fromarray(arr, "RGB").convert("L"))[0, 0]
    assert abs(int(np.array(l).reshape(4, 4)[0, 0]) - int(want)) <= 1
    a = df.select(img.image.to_mode("RGBA").alias("a")).to_pydict()["a"][0]
    assert len(a) == 4 * 4 * 4 and a[3] == 255
