from typing import Any, Dict

from .base import ClientBatchedModel, Params
from .mlp import MLP
from .lenet import LeNet5
from .resnet import ResNet18

_REGISTRY = {
    "mlp": MLP,
    "lenet": LeNet5,
    "resnet18": ResNet18,
}


def build_model(name: str, **kwargs: Any) -> ClientBatchedModel:
    if name == "bert":
        from .bert import BertTiny  # deferred: heavier module
        return BertTiny(**kwargs)
    if name == "bert-base":
        from .bert import BertBase
        return BertBase(**kwargs)
    if name not in _REGISTRY:
        raise KeyError(f"unknown model {name!r}; known: {sorted(_REGISTRY)} "
                       f"+ ['bert', 'bert-base']")
    return _REGISTRY[name](**kwargs)


__all__ = ["ClientBatchedModel", "Params", "MLP", "LeNet5", "ResNet18",
           "build_model"]
