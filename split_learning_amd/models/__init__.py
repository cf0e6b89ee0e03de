"""Model registry: (model_name, data_name) -> partitioned model class.

Mirrors the reference's model selection (src/Server.py:234-239,
src/RpcClient.py:79-84, other/Vanilla_SL/src/model/__init__.py).
"""

from __future__ import annotations

from .bert import BERT_AGNEWS, BERT_EMOTION
from .kwt import KWT_SPEECHCOMMANDS
from .mobilenet import MobileNetv1_CIFAR10, MobileNetv1_MNIST
from .partitioned import PartitionedModel, SequentialUnits
from .vgg import VGG16_CIFAR10, VGG16_MNIST
from .vit import ViT_CIFAR10, ViT_MNIST

_REGISTRY = {
    ("VGG16", "CIFAR10"): VGG16_CIFAR10,
    ("VGG16", "MNIST"): VGG16_MNIST,
    ("BERT", "AGNEWS"): BERT_AGNEWS,
    ("BERT", "EMOTION"): BERT_EMOTION,
    ("KWT", "SPEECHCOMMANDS"): KWT_SPEECHCOMMANDS,
    ("MobileNetv1", "CIFAR10"): MobileNetv1_CIFAR10,
    ("MobileNetv1", "MNIST"): MobileNetv1_MNIST,
    ("ViT", "CIFAR10"): ViT_CIFAR10,
    ("ViT", "MNIST"): ViT_MNIST,
}


def get_model_class(model_name: str, data_name: str):
    try:
        return _REGISTRY[(model_name, data_name)]
    except KeyError:
        raise ValueError(f"No model for ({model_name!r}, {data_name!r}); "
                         f"known: {sorted(_REGISTRY)}")


def build_partition(model_name: str, data_name: str, layers, **kwargs):
    """Instantiate the partition for a stage's [start, end] range.

    Matches reference semantics (src/RpcClient.py:86-92): end == -1 means
    "through the last unit"; [0, 0] means the full model.
    """
    klass = get_model_class(model_name, data_name)
    start, end = layers
    if end == 0:
        return klass(**kwargs)
    return klass(start_layer=start, end_layer=end, **kwargs)


__all__ = [
    "PartitionedModel", "SequentialUnits", "get_model_class", "build_partition",
    "VGG16_CIFAR10", "VGG16_MNIST", "BERT_AGNEWS", "BERT_EMOTION",
    "KWT_SPEECHCOMMANDS", "MobileNetv1_CIFAR10", "MobileNetv1_MNIST",
    "ViT_CIFAR10", "ViT_MNIST",
]
