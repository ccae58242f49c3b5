from typing import Any

from .base import ClientBatchedModel, Params
from .mlp import MLP
from .lenet import LeNet5
from .resnet import ResNet18
from .lstm import CharLSTM

_REGISTRY = {
    "mlp": MLP,
    "lenet": LeNet5,
    "resnet18": ResNet18,
    "lstm": CharLSTM,
}


def build_model(name: str, **kwargs: Any) -> ClientBatchedModel:
    if name == "bert":
        from .bert import BertTiny  # deferred: heavier module
        return BertTiny(**kwargs)
    if name == "bert-base":
        from .bert import BertBase
        return BertBase(**kwargs)
    if ":" in name:
        # user model plug-in "pkg.module:ClassName" (the analogue of the
        # reference's user-supplied operator code)
        import importlib
        mod_name, cls_name = name.split(":", 1)
        cls = getattr(importlib.import_module(mod_name), cls_name)
        return cls(**kwargs)
    if name not in _REGISTRY:
        raise KeyError(f"unknown model {name!r}; known: {sorted(_REGISTRY)} "
                       f"+ ['bert', 'bert-base'] or 'pkg.module:ClassName'")
    return _REGISTRY[name](**kwargs)


__all__ = ["ClientBatchedModel", "Params", "MLP", "LeNet5", "ResNet18",
           "CharLSTM", "build_model"]
