from maggy_amd.models.mlp import MLP  # noqa: F401
from maggy_amd.models.resnet import (  # noqa: F401
    ResNet,
    resnet18_thin,
    resnet50,
    resnet101,
    resnet152,
)
from maggy_amd.models.llama import LlamaConfig, LlamaModel  # noqa: F401
from maggy_amd.models.transformer import SmallTransformer  # noqa: F401

__all__ = [
    "MLP", "ResNet", "resnet50", "resnet101", "resnet152", "resnet18_thin",
    "LlamaConfig", "LlamaModel", "SmallTransformer",
]
