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


def test_backend_trace_dump(tmp_path, monkeypatch):
    """BAGUA_TRACE=1 records per-bucket comm spans dumpable as a
    chrome://tracing JSON (per-op tracing depth, reference: tracing crate
    spans around execute_ops)."""
    import json
    import os

    from tests.internal.multi_process import run_multi_process

    out = run_multi_process(2, _worker_trace, args=(str(tmp_path),))
    for rank in range(2):
        path = os.path.join(str(tmp_path), "trace_%d.json" % rank)
        with open(path) as f:
            doc = json.load(f)
        events = doc["traceEvents"]
        assert events, "no spans recorded"
        assert all(e["ph"] == "X" and e["dur"] >= 0 for e in events)
        assert any("CentralizedSyncOp" in e["cat"] for e in events)


def _worker_trace(rank, nprocs, tmpdir):
    import os

    os.environ["BAGUA_TRACE"] = "1"
    import torch
    import torch.nn.functional as F

    import bagua_amd
    from bagua_amd.parallel.algorithms.gradient_allreduce import (
        GradientAllReduceAlgorithm,
    )
    from tests.test_algorithms import Net, _make_data

    bagua_amd.init_process_group()
    torch.manual_seed(1)
    model = Net()
    optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    ddp = bagua_amd.DistributedDataParallel(
        model, optimizers=[optimizer],
        algorithm=GradientAllReduceAlgorithm())
    for step in range(3):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        F.mse_loss(ddp(data), target).backward()
        optimizer.step()
    ddp.inner.bagua_backend.dump_trace(
        os.path.join(tmpdir, "trace_%d.json" % rank))
    bagua_amd.deinit_process_group()
    return True


def test_flattened_call_and_contiguous():
    import torch

    from bagua_amd.utils import (
        apply_flattened_call_all,
        check_contiguous,
        flatten,
    )

    ts = [torch.randn(3), torch.randn(5), torch.ones(2, dtype=torch.int64)]
    ref = [t * 2 if t.is_floating_point() else t * 2 for t in ts]
    apply_flattened_call_all(ts, lambda flat: flat.mul_(2))
    for a, b in zip(ts, ref):
        assert torch.equal(a, b)

    flat = flatten([torch.randn(3), torch.randn(4)])
    views = [flat[:3], flat[3:]]
    assert check_contiguous(views)
    assert not check_contiguous([torch.randn(3), torch.randn(3)])


def test_average_by_removing_extreme_values():
    from bagua_amd.utils import average_by_removing_extreme_values

    samples = [0.0, 0.0, 100.0, 101.0, 99.0, 100.5, 1000.0, 100.2]
    mean, std, kept = average_by_removing_extreme_values(samples)
    assert 95.0 < mean < 105.0, mean
    assert 1000.0 not in kept
