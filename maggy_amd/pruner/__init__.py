from maggy_amd.pruner.abstract import AbstractPruner  # noqa: F401
from maggy_amd.pruner.hyperband import Hyperband  # noqa: F401

__all__ = ["AbstractPruner", "Hyperband"]
