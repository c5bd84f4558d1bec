import pytest
import torch


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an MI355X GPU (run via gpurun)")
    config.addinivalue_line(
        "markers", "distributed: multi-process torch.distributed test")


@pytest.fixture
def gpu_device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU available")
    return "cuda:0"


@pytest.fixture(params=["cpu"])
def any_device(request):
    return request.param


def assert_df_eq(actual: dict, expected: dict, sort_by=None, approx=False,
                 rel=1e-6):
    """Compare to_pydict results, optionally sorting rows by a key column."""
    assert set(actual.keys()) == set(expected.keys()), \
        f"columns differ: {sorted(actual)} vs {sorted(expected)}"
    names = list(expected.keys())
    if sort_by is not None:
        def keyed(d):
            cols = [d[k] for k in (sort_by if isinstance(sort_by, list)
                                   else [sort_by])]
            order = sorted(range(len(cols[0])),
                           key=lambda i: tuple(
                               (v[i] is None, v[i]) for v in cols))
            return {k: [d[k][i] for i in order] for k in names}
        actual, expected = keyed(actual), keyed(expected)
    for k in names:
        a, e = actual[k], expected[k]
        assert len(a) == len(e), f"{k}: {len(a)} rows vs {len(e)}"
        for i, (x, y) in enumerate(zip(a, e)):
            if y is None:
                assert x is None, f"{k}[{i}]: {x} != None"
            elif approx and isinstance(y, float):
                assert x == pytest.approx(y, rel=rel), f"{k}[{i}]: {x} != {y}"
            else:
                assert x == y, f"{k}[{i}]: {x!r} != {y!r}"


@pytest.fixture
def df_eq():
    return assert_df_eq
