"""Single-run config.

Parity: /root/reference/maggy/config/base_config.py:23-38. The reference's
"python kernel" mode runs the training function inline in the driver process;
ours does the same (CPU or one GPU), making it the laptop-mode smoke path.
"""
from maggy_amd.config.lagom import LagomConfig


class BaseConfig(LagomConfig):
    def __init__(
        self,
        name="maggyExperiment",
        description="",
        hb_interval=1,
        model=None,
        dataset=None,
        hparams=None,
    ):
        super().__init__(name=name, description=description, hb_interval=hb_interval)
        self.model = model
        self.dataset = dataset
        self.hparams = hparams if hparams is not None else {}
