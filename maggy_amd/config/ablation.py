"""Ablation-study config.

Parity: /root/reference/maggy/config/ablation.py:29-67 (ablation_study,
ablator="loco", direction). Early stop is forced off for ablation runs, as in
the reference driver (/root/reference/maggy/core/experiment_driver/
ablation_driver.py:54).
"""
from maggy_amd.config.lagom import LagomConfig


class AblationConfig(LagomConfig):
    def __init__(
        self,
        ablation_study,
        ablator="loco",
        direction="max",
        name="ablationStudy",
        description="",
        hb_interval=1,
        model=None,
        dataset=None,
        num_workers=None,
    ):
        super().__init__(name=name, description=description, hb_interval=hb_interval)
        self.ablator = ablator
        self.ablation_study = ablation_study
        self.direction = direction
        self.model = model
        self.dataset = dataset
        self.num_workers = num_workers
