from maggy_amd.optimizer.bayes.base import BaseAsyncBO  # noqa: F401
from maggy_amd.optimizer.bayes.gp import GP  # noqa: F401
from maggy_amd.optimizer.bayes.tpe import TPE  # noqa: F401

__all__ = ["BaseAsyncBO", "GP", "TPE"]
