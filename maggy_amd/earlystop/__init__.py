from maggy_amd.earlystop.abstract import AbstractEarlyStop  # noqa: F401
from maggy_amd.earlystop.medianrule import MedianStoppingRule  # noqa: F401
from maggy_amd.earlystop.nostop import NoStoppingRule  # noqa: F401

__all__ = ["AbstractEarlyStop", "MedianStoppingRule", "NoStoppingRule"]
