"""Benchmark model zoo (synthetic-data training shapes).

The reference benchmarked torchvision models
(examples/benchmark/synthetic_benchmark.py); torchvision is not in this
image, so the architectures are implemented here directly.
"""

from .vgg import VGG, vgg16, vgg11, vgg13, vgg19  # noqa: F401
from .resnet import (  # noqa: F401
    ResNet,
    resnet18,
    resnet50,
    resnet101,
    resnet152,
)
from .mnist import MnistNet  # noqa: F401
from .bert import BertConfig, BertForPretrainingShape, bert_large  # noqa: F401


def create_model(name: str, num_classes: int = 1000):
    name = name.lower()
    factory = {
        "vgg11": vgg11,
        "vgg13": vgg13,
        "vgg16": vgg16,
        "vgg19": vgg19,
        "resnet18": resnet18,
        "resnet50": resnet50,
        "resnet101": resnet101,
        "resnet152": resnet152,
    }
    if name in factory:
        return factory[name](num_classes=num_classes)
    if name == "mnist":
        return MnistNet()
    if name in ("bert-large", "bert_large"):
        return bert_large()
    raise ValueError("unknown model %r" % name)
