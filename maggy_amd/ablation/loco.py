"""LOCO — leave-one-component-out ablator.

Parity: /root/reference/maggy/ablation/ablator/loco.py:138-261 — n+1 trials
built eagerly into a buffer: the base trial, one per included feature, one
per included layer, one per layer group, one per custom model.  Trial
params carry ``ablated_feature`` / ``ablated_layer`` exactly like the
reference so the .hparams.json artifacts match.
"""
from abc import ABC, abstractmethod

from maggy_amd.trial import Trial


class AbstractAblator(ABC):
    experiment_type = "ablation"

    def __init__(self, ablation_study, final_store=None):
        self.ablation_study = ablation_study
        self.final_store = final_store if final_store is not None else []
        self.trial_buffer = []

    @abstractmethod
    def get_number_of_trials(self):
        ...

    @abstractmethod
    def initialize(self):
        ...

    @abstractmethod
    def get_trial(self, ablation_trial=None):
        ...

    @abstractmethod
    def finalize_experiment(self, trials):
        ...

    def get_dataset_generator(self):
        return self.ablation_study.dataset_generator

    def get_model_generator(self):
        return self.ablation_study.model_generator

    def name(self):
        return str(self.__class__.__name__)

    # driver-compat shims (the OptimizationDriver drives controllers through
    # the optimizer contract)
    def _initialize(self, exp_dir=None):
        self.initialize()

    def _finalize_experiment(self, trials):
        self.finalize_experiment(trials)

    def get_suggestion(self, trial=None):
        return self.get_trial(trial)


class LOCO(AbstractAblator):
    def get_number_of_trials(self):
        study = self.ablation_study
        return (
            1
            + len(study.features.included_features)
            + len(study.model.layers.included_layers)
            + len(study.model.layers.included_groups)
            + len(study.model.custom_model_generators)
        )

    def initialize(self):
        """Eagerly build every ablation trial (parity loco.py:138-194)."""
        study = self.ablation_study
        # base trial: nothing ablated
        self.trial_buffer.append(self._make_trial("None", "None"))
        for feature in study.features.list_all():
            self.trial_buffer.append(self._make_trial(feature, "None"))
        for layer in study.model.layers.list_all():
            self.trial_buffer.append(self._make_trial("None", layer))
        for group in study.model.layers.list_groups():
            self.trial_buffer.append(
                self._make_trial("None", "+".join(group)))
        for name, _gen in study.model.custom_model_generators:
            self.trial_buffer.append(
                self._make_trial("None", "custom:" + name))

    def _make_trial(self, ablated_feature, ablated_layer):
        params = {
            "ablated_feature": ablated_feature,
            "ablated_layer": ablated_layer,
        }
        return Trial(params, trial_type="ablation")

    def get_trial(self, ablation_trial=None):
        if self.trial_buffer:
            return self.trial_buffer.pop(0)
        return None

    def finalize_experiment(self, trials):
        return
