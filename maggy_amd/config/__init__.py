from maggy_amd.config.lagom import LagomConfig  # noqa: F401
from maggy_amd.config.base import BaseConfig  # noqa: F401
from maggy_amd.config.hpo import HyperparameterOptConfig  # noqa: F401
from maggy_amd.config.ablation import AblationConfig  # noqa: F401
from maggy_amd.config.torch_dist import TorchDistributedConfig  # noqa: F401

__all__ = [
    "LagomConfig",
    "BaseConfig",
    "HyperparameterOptConfig",
    "AblationConfig",
    "TorchDistributedConfig",
]
