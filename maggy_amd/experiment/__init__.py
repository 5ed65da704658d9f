from maggy_amd.experiment.experiment import (  # noqa: F401
    LagomHandle,
    lagom,
    lagom_async,
)

__all__ = ["lagom", "lagom_async", "LagomHandle"]
