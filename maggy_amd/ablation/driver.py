"""Ablation experiment driver.

Parity: /root/reference/maggy/core/experiment_driver/ablation_driver.py:
32-208 — an OptimizationDriver whose controller is the LOCO ablator, early
stop forced off, direction accepted without validation against a
searchspace.  (The reference's missing-config_class bug at
ablation_driver.py:80 does not exist here: there is no RPC server.)
"""
from maggy_amd.ablation.ablationstudy import AblationStudy, drop_layers
from maggy_amd.ablation.loco import LOCO, AbstractAblator
from maggy_amd.core.driver import OptimizationDriver


class ModelGeneratorDispatch:
    """Picklable model-generator wrapper shipped to pool workers: dispatches
    custom-model trials and applies drop_layers when the base generator does
    not itself understand ``ablated_layer``."""

    def __init__(self, base, customs):
        self.base = base
        self.customs = customs

    def __call__(self, ablated_layer="None"):
        if ablated_layer and ablated_layer.startswith("custom:"):
            return self.customs[ablated_layer[len("custom:"):]]()
        try:
            return self.base(ablated_layer=ablated_layer)
        except TypeError:
            return drop_layers(self.base(), ablated_layer)


def _resolve_ablator(ablator, study):
    if isinstance(ablator, str):
        if ablator.lower() != "loco":
            raise ValueError(
                "Unknown ablator '{}'; only 'loco' or an AbstractAblator "
                "instance".format(ablator))
        return LOCO(study)
    if isinstance(ablator, AbstractAblator):
        return ablator
    raise ValueError("ablator must be 'loco' or an AbstractAblator")


class AblationDriver(OptimizationDriver):
    def __init__(self, config, **kwargs):
        if not isinstance(config.ablation_study, AblationStudy):
            raise ValueError(
                "config.ablation_study must be an AblationStudy")
        self._ablator_obj = _resolve_ablator(
            config.ablator, config.ablation_study)
        config.num_trials = self._ablator_obj.get_number_of_trials()
        config.optimizer = None        # placeholder; controller swapped below
        config.searchspace = None
        config.es_policy = "none"      # reference forces ES off for ablation
        super().__init__(config, **kwargs)
        self._ablator_obj.final_store = self._final_store
        self.controller = self._ablator_obj

    def run_experiment(self, train_fn, payload_extra=None):
        study = self.controller.ablation_study
        extra = dict(payload_extra or {})
        extra.setdefault("model_generator", self._wrapped_model_generator(study))
        extra.setdefault("dataset_generator", study.dataset_generator)
        return super().run_experiment(train_fn, payload_extra=extra)

    @staticmethod
    def _wrapped_model_generator(study):
        """Compose the user's base model generator with layer dropping and
        custom-model dispatch (the PyTorch analog of the reference's Keras
        model surgery, loco.py:99-136)."""
        if study.model_generator is None:
            return None
        return ModelGeneratorDispatch(
            study.model_generator, dict(study.model.custom_model_generators))

    def config_to_dict(self):
        return self.controller.ablation_study.to_dict()
