"""Autotune service behavior with mocked clients
(reference pattern: tests/service/test_autotune_service.py:29-50)."""

import os

import pytest

from bagua_amd.defines import BaguaHyperparameter
from bagua_amd.service.autotune_service import (
    AutotuneClient,
    start_autotune_server,
)
from bagua_amd.service.autotune_task_manager import (
    split_bucket_by_bucket_size,
)
from bagua_amd.defines import TensorDeclaration, TensorDtype


def _decls():
    return [
        TensorDeclaration(name="t%d" % i, num_elements=1 << 18,
                          dtype=TensorDtype.F32).dict()
        for i in range(16)
    ]


@pytest.fixture()
def server(monkeypatch):
    monkeypatch.setenv("BAGUA_AUTOTUNE", "1")
    monkeypatch.setenv("BAGUA_AUTOTUNE_WARMUP_TIME_S", "0")
    monkeypatch.setenv("BAGUA_AUTOTUNE_SAMPLING_CONFIDENCE_TIME_S", "0")
    monkeypatch.setenv("BAGUA_AUTOTUNE_MAX_SAMPLES", "4")
    srv = start_autotune_server(0, world_size=2)
    yield srv
    srv.shutdown()


def test_autotune_loop(server):
    client = AutotuneClient("127.0.0.1", server.port)
    assert client.health_check()

    rsp = client.register_tensors("m", _decls())
    hp0 = BaguaHyperparameter(**rsp["recommended_hyperparameters"])
    assert hp0.buckets, "initial size-based bucketing missing"

    seen_sizes = set()
    completed = False
    for it in range(1, 20):
        for rank in range(2):
            client.report_metrics("m", rank, it, hp0.dict(),
                                  speed=1e9 * (1 + it % 3))
        rsp = client.ask_hyperparameters("m", rank=0, train_iter=it)
        hp = BaguaHyperparameter(**rsp["recommended_hyperparameters"])
        seen_sizes.add(hp.bucket_size)
        if rsp["is_autotune_completed"]:
            completed = True
            break
    assert completed, "autotune never completed"
    assert len(seen_sizes) > 1, "optimizer never explored"


def test_execution_order_resorts_buckets(server):
    client = AutotuneClient("127.0.0.1", server.port)
    client.register_tensors("m2", _decls())
    # report reversed execution order
    spans = [{"trace_id": 0, "action": "tensor_ready",
              "tensor_name": "t%d" % i, "start_time": 100 - i,
              "end_time": 100 - i, "model_name": "m2"}
             for i in range(16)]
    client.report_tensor_execution_order(spans)
    mgr = server.state.manager("m2")
    ordered = mgr.ordered_tensor_list()
    assert [t.name for t in ordered][:3] == ["t15", "t14", "t13"]


def test_split_bucket_by_bucket_size():
    decls = [TensorDeclaration(name="a", num_elements=100,
                               dtype=TensorDtype.F32),
             TensorDeclaration(name="b", num_elements=100,
                               dtype=TensorDtype.F32),
             TensorDeclaration(name="c", num_elements=100,
                               dtype=TensorDtype.F16),
             TensorDeclaration(name="d", num_elements=1000,
                               dtype=TensorDtype.F32)]
    buckets = split_bucket_by_bucket_size(decls, 800)
    # a+b fit one f32 bucket; c splits on dtype; d overflows
    assert [[t.name for t in b] for b in buckets] == [
        ["a", "b"], ["c"], ["d"]]
