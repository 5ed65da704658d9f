"""User-facing ablation study definition.

Parity: /root/reference/maggy/ablation/ablationstudy.py:160-385 — same
``features.include/exclude`` and ``model.layers.include/include_groups``
API (groups become frozensets), base/custom model generators and optional
custom dataset generator.  The Keras-specific JSON layer surgery of the
reference is replaced by the PyTorch idiom: generators are callables
``model_generator(ablated_layer=...)`` / ``dataset_generator(
ablated_feature=...)`` and the helper ``drop_layers`` removes named
children from an nn.Module.
"""


class Features:
    def __init__(self):
        self.included_features = set()

    def include(self, *args):
        for arg in args:
            if isinstance(arg, (list, tuple, set)):
                for f in arg:
                    self._add(f)
            else:
                self._add(arg)

    def _add(self, feature):
        if not isinstance(feature, str):
            raise ValueError(
                "Feature names must be strings, got {}".format(type(feature)))
        self.included_features.add(feature)

    def exclude(self, *args):
        for arg in args:
            if isinstance(arg, (list, tuple, set)):
                for f in arg:
                    self.included_features.discard(f)
            else:
                self.included_features.discard(arg)

    def list_all(self):
        return sorted(self.included_features)

    def __iter__(self):
        return iter(self.list_all())


class Layers:
    def __init__(self):
        self.included_layers = set()
        self.included_groups = set()  # of frozensets

    def include(self, *args):
        for arg in args:
            if isinstance(arg, (list, tuple, set)):
                for name in arg:
                    self._add(name)
            else:
                self._add(arg)

    def _add(self, name):
        if not isinstance(name, str):
            raise ValueError(
                "Layer names must be strings, got {}".format(type(name)))
        self.included_layers.add(name)

    def exclude(self, *args):
        for arg in args:
            if isinstance(arg, (list, tuple, set)):
                for name in arg:
                    self.included_layers.discard(name)
            else:
                self.included_layers.discard(arg)

    def include_groups(self, *args, prefix=None):
        """Add a group of layers ablated together: an explicit list, or all
        layers sharing a name prefix (resolved at trial-generation time by
        passing the prefix marker)."""
        if prefix is not None:
            if not isinstance(prefix, str):
                raise ValueError("prefix must be a string")
            self.included_groups.add(frozenset([prefix + "*"]))
        for arg in args:
            if not isinstance(arg, (list, tuple, set)) or len(arg) < 2:
                raise ValueError(
                    "A layer group must be a list of >= 2 layer names")
            self.included_groups.add(frozenset(arg))

    def list_all(self):
        return sorted(self.included_layers)

    def list_groups(self):
        return sorted(sorted(g) for g in self.included_groups)


class Model:
    def __init__(self):
        self.layers = Layers()
        # list of (name, generator) custom whole-model variants
        self.custom_model_generators = []

    def add_custom_generator(self, name, generator):
        self.custom_model_generators.append((name, generator))


class AblationStudy:
    """Defines what to ablate and how to rebuild model/dataset per trial.

    :param training_dataset_name / label_name: carried for artifact parity
        with the reference (feature-store metadata); unused locally.
    :param model_generator: callable(ablated_layer="None") -> nn.Module
    :param dataset_generator: callable(ablated_feature="None") -> dataset
    """

    def __init__(self, training_dataset_name="synthetic",
                 training_dataset_version=1, label_name="label",
                 model_generator=None, dataset_generator=None):
        self.hops_training_dataset_name = training_dataset_name
        self.hops_training_dataset_version = training_dataset_version
        self.label_name = label_name
        self.features = Features()
        self.model = Model()
        self.model_generator = model_generator
        self.dataset_generator = dataset_generator

    def set_base_model_generator(self, generator):
        self.model_generator = generator

    def set_dataset_generator(self, generator):
        self.dataset_generator = generator

    def to_dict(self):
        return {
            "training_dataset_name": self.hops_training_dataset_name,
            "training_dataset_version": self.hops_training_dataset_version,
            "label_name": self.label_name,
            "included_features": self.features.list_all(),
            "included_layers": self.model.layers.list_all(),
            "included_layer_groups": self.model.layers.list_groups(),
            "custom_models": [n for n, _ in
                              self.model.custom_model_generators],
        }


def drop_layers(module, ablated_layer):
    """Replace named children (dot paths allowed) with nn.Identity.

    ``ablated_layer`` is "None", a single name, a "+"-joined group, or a
    "prefix*" marker matching every child whose name starts with prefix.
    Returns the module (mutated in place).
    """
    import torch.nn as nn

    if not ablated_layer or ablated_layer == "None":
        return module
    names = ablated_layer.split("+")
    targets = []
    all_names = [n for n, _ in module.named_modules() if n]
    for name in names:
        if name.endswith("*"):
            prefix = name[:-1].rstrip(".")
            # "blocks*" targets the direct children of the 'blocks'
            # container (blocks.0, blocks.1, ...); when the prefix names no
            # container, it targets every top-level match
            children = [n for n in all_names
                        if n.startswith(prefix + ".")
                        and "." not in n[len(prefix) + 1:]]
            if children:
                targets.extend(children)
            else:
                matches = [n for n in all_names if n.startswith(prefix)]
                roots = [n for n in matches
                         if not any(n.startswith(m + ".") for m in matches
                                    if m != n)]
                targets.extend(roots)
        else:
            targets.append(name)
    for name in targets:
        parent = module
        parts = name.split(".")
        for p in parts[:-1]:
            parent = getattr(parent, p)
        if not hasattr(parent, parts[-1]):
            raise ValueError(
                "Cannot ablate '{}': module has no child '{}'".format(
                    name, parts[-1]))
        setattr(parent, parts[-1], nn.Identity())
    return module
