"""BaguaStrategy smoke tests, 2-proc gloo.

pytorch_lightning is absent from this image, so the tests drive the
strategy through the Lightning Strategy protocol directly
(setup_environment -> setup_module -> train -> reduce/broadcast/barrier
-> teardown), the same call order a Trainer uses
(reference: tests/pytorch_lightning/test_bagua_strategy.py:30-40).
"""

import pytest
import torch
import torch.nn.functional as F

from tests.internal.multi_process import run_multi_process


def _worker_strategy(rank, nprocs, algo, kwargs):
    import bagua_amd  # noqa: F401
    from bagua_amd.lightning import BaguaStrategy, BaguaStrategyCore
    from tests.test_algorithms import Net, _make_data

    # without lightning installed the bound name IS the core
    strategy = BaguaStrategy(algorithm=algo, **kwargs)
    assert isinstance(strategy, BaguaStrategyCore)
    assert strategy.strategy_name == "bagua"

    strategy.setup_environment()
    torch.manual_seed(13 + rank)
    model = Net()
    if algo == "qadam":
        from bagua_amd.parallel.algorithms.q_adam import QAdamOptimizer

        # warmup must be long enough for Adam's second moment to be
        # meaningful before it freezes (same regime as the reference's
        # strategy test, warmup_steps=20)
        optimizer = QAdamOptimizer(model.parameters(), lr=1e-3,
                                   warmup_steps=6)
    else:
        optimizer = torch.optim.SGD(model.parameters(), lr=0.05)
    wrapped = strategy.setup_module(model, [optimizer])

    for step in range(12):
        data, target = _make_data(rank, step)
        optimizer.zero_grad()
        loss = F.mse_loss(wrapped(data), target)
        loss.backward()
        optimizer.step()

    # strategy collectives
    t = torch.ones(4) * (rank + 1)
    strategy.reduce(t, reduce_op="mean")
    assert torch.allclose(t, torch.full((4,), (1 + nprocs) / 2))
    obj = strategy.broadcast({"rank": rank}, src=0)
    assert obj == {"rank": 0}
    strategy.barrier()
    strategy.teardown()

    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    bagua_amd.deinit_process_group()
    return flat


@pytest.mark.parametrize("algo,kwargs", [
    ("gradient_allreduce", {}),
    ("bytegrad", {}),
    ("qadam", {}),
    ("decentralized", {"peer_selection_mode": "all"}),
    ("low_precision_decentralized", {}),
])
def test_strategy_consensus(algo, kwargs):
    out = run_multi_process(2, _worker_strategy, args=(algo, dict(kwargs)))
    if algo in ("gradient_allreduce", "bytegrad", "qadam"):
        assert torch.equal(out[0], out[1]), "strategy ranks diverged"
    else:  # decentralized family: one local step past the last sync
        assert torch.allclose(out[0], out[1], atol=0.5)


def test_strategy_rejects_missing_qadam_optimizer():
    from bagua_amd.lightning import BaguaStrategyCore

    s = BaguaStrategyCore(algorithm="qadam")
    with pytest.raises(ValueError, match="QAdamOptimizer"):
        s._make_algorithm([torch.optim.SGD([torch.nn.Parameter(
            torch.zeros(1))], lr=0.1)])
