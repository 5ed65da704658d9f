from maggy_amd.experiment.experiment import lagom  # noqa: F401

__all__ = ["lagom"]
