"""Model zoo fidelity: parameter counts must match the torchvision /
HF architectures the reference benchmarked, so throughput numbers are
comparable."""

import pytest
import torch

from bagua_amd.models import (
    BertConfig,
    BertForPretrainingShape,
    bert_large,
    create_model,
)


@pytest.mark.parametrize("name,params", [
    ("vgg16", 138357544),      # torchvision vgg16
    ("vgg11", 132863336),
    ("resnet18", 11689512),
    ("resnet50", 25557032),    # torchvision resnet50
    ("resnet101", 44549160),
])
def test_param_counts(name, params):
    m = create_model(name)
    assert sum(p.numel() for p in m.parameters()) == params


def test_bert_large_size():
    m = bert_large()
    n = sum(p.numel() for p in m.parameters())
    # BERT-Large encoder ~335M (plus qa head)
    assert 300e6 < n < 360e6, n


def test_bert_tiny_forward_backward():
    from bagua_amd.models.bert import bert_tiny

    m = bert_tiny()
    ids = torch.randint(0, 1000, (2, 16))
    s, e = m(ids)
    assert s.shape == (2, 16)
    (s.sum() + e.sum()).backward()


def test_forward_shapes():
    m = create_model("vgg16")
    out = m(torch.randn(2, 3, 224, 224))
    assert out.shape == (2, 1000)
    m = create_model("mnist")
    out = m(torch.randn(2, 1, 28, 28))
    assert out.shape == (2, 10)
