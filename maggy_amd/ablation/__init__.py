from maggy_amd.ablation.ablationstudy import AblationStudy, drop_layers  # noqa: F401
from maggy_amd.ablation.loco import LOCO, AbstractAblator  # noqa: F401

__all__ = ["AblationStudy", "LOCO", "AbstractAblator", "drop_layers"]
