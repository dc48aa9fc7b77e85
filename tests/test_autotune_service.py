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


def test_stale_iteration_samples_do_not_score(server):
    """All-ranks-synced-per-hp gate: a speed sample reported for an
    iteration at or before the current hp's install point must not score
    it (reference gated per-iteration with all ranks synced,
    autotune_service.py:78-272)."""
    client = AutotuneClient("127.0.0.1", server.port)
    client.register_tensors("m2", _decls())
    st = server.state

    # both ranks report at iter 5, then a proposal happens at iter 5
    for rank in range(2):
        client.report_metrics("m2", rank, 5, {}, 100.0)
    client.ask_hyperparameters("m2", 0, 5)
    assert st.sample_count["m2"] == 1
    installed = st.hp_installed_iter["m2"]
    assert installed == 5

    # rank 0 reports a FRESH sample, rank 1 a stale one (iter <= installed)
    client.report_metrics("m2", 0, 7, {}, 100.0)
    client.report_metrics("m2", 1, installed, {}, 100.0)
    client.ask_hyperparameters("m2", 0, 7)
    assert st.sample_count["m2"] == 1, "stale sample scored the hp"

    # once rank 1 is fresh too the proposal goes through
    client.report_metrics("m2", 1, 7, {}, 100.0)
    client.ask_hyperparameters("m2", 0, 7)
    assert st.sample_count["m2"] == 2


def test_single_node_does_not_search_hierarchical(server):
    """One node => the hierarchical dimension is pure noise; it must not
    be in the search space and proposals must carry None so the engine
    leaves the user's algorithm flag alone."""
    client = AutotuneClient("127.0.0.1", server.port)
    client.register_tensors("m3", _decls())
    st = server.state
    mgr = st.manager("m3")
    assert not mgr.search_hierarchical
    for it in range(1, 8):
        for rank in range(2):
            client.report_metrics("m3", rank, it, {}, 100.0)
        rsp = client.ask_hyperparameters("m3", 0, it)
        hp = rsp["recommended_hyperparameters"]
        assert hp.get("is_hierarchical_reduce") is None


def test_multi_node_searches_hierarchical():
    """nnodes>1 puts is_hierarchical_reduce back into the search space."""
    from bagua_amd.service.autotune_task_manager import AutotuneTaskManager

    mgr = AutotuneTaskManager("m", search_hierarchical=True)
    assert mgr.search_hierarchical
    hp = BaguaHyperparameter(bucket_size=1 << 25)
    seen = set()
    for i in range(12):
        hp = mgr.tell_and_ask(hp, 100.0 + i)
        assert hp.is_hierarchical_reduce in (True, False)
        seen.add(hp.is_hierarchical_reduce)
    assert len(seen) == 2, "hierarchical dimension never explored"
