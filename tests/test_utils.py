import torch

from bagua_amd.utils import (
    StatisticalAverage,
    flatten,
    show_version,
    to_bagua_datatype,
    unflatten,
)


def test_statistical_average():
    sa = StatisticalAverage()
    sa.record(10.0, now=100.0)
    sa.record(20.0, now=110.0)
    sa.record(30.0, now=160.0)
    assert sa.get(window_s=55.0, now=160.0) == 25.0
    assert sa.get(window_s=5.0, now=160.0) == 30.0
    assert sa.get(window_s=1.0, now=300.0) == 0.0
    assert sa.total_recording_time() == 60.0


def test_flatten_unflatten():
    a, b = torch.randn(3, 4), torch.randn(5)
    flat = flatten([a, b])
    ra, rb = unflatten(flat, [a, b])
    assert torch.equal(ra, a) and torch.equal(rb, b)


def test_to_bagua_datatype():
    assert to_bagua_datatype(torch.bfloat16) == "bf16"


def test_show_version_runs():
    lines = show_version()
    assert any("bagua_amd" in ln for ln in lines)
