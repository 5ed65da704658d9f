"""Hyperparameter search-space definition.

Parity: /root/reference/maggy/searchspace.py:23-479 — same four parameter
types (DOUBLE / INTEGER / DISCRETE / CATEGORICAL), same (type, region)
tuple API with dot-notation attribute access, random sampling, unit-hypercube
transform/inverse_transform with categorical index encoding, and dict<->list
conversion used by the Bayesian optimizers.
"""
import json
import random


class Searchspace:
    DOUBLE = "DOUBLE"
    INTEGER = "INTEGER"
    DISCRETE = "DISCRETE"
    CATEGORICAL = "CATEGORICAL"

    def __init__(self, **kwargs):
        self._hparam_types = {}
        self._names = []
        for name, value in kwargs.items():
            self.add(name, value)

    def add(self, name, value):
        """Add a hyperparameter ``name -> (type, feasible_region)``.

        DOUBLE/INTEGER regions are ``[lower, upper]`` bounds; DISCRETE and
        CATEGORICAL regions are explicit value lists.
        """
        if getattr(self, name, None) is not None:
            raise ValueError("Hyperparameter name is reserved: {}".format(name))
        if not isinstance(value, (tuple, list)):
            raise ValueError(
                "Hyperparameter has to be a tuple (type, feasible_region): "
                "{}, {}".format(name, value)
            )
        if len(value) != 2:
            raise ValueError(
                "Hyperparameter tuple has to be of length two and format "
                "(type, list): {}, {}".format(name, value)
            )

        param_type = str(value[0]).upper()
        param_values = value[1]
        if param_type not in (
            Searchspace.DOUBLE,
            Searchspace.INTEGER,
            Searchspace.DISCRETE,
            Searchspace.CATEGORICAL,
        ):
            raise ValueError(
                "Hyperparameter type is not of type DOUBLE, INTEGER, DISCRETE "
                "or CATEGORICAL: {}, {}".format(name, value)
            )
        if len(param_values) == 0:
            raise ValueError(
                "Hyperparameter feasible region list cannot be empty: "
                "{}, {}".format(name, param_values)
            )
        if param_type in (Searchspace.DOUBLE, Searchspace.INTEGER):
            if len(param_values) != 2:
                raise ValueError(
                    "For DOUBLE or INTEGER type parameters, the feasible "
                    "region must be [lower, upper]: {}, {}".format(
                        name, param_values
                    )
                )
            lower, upper = param_values
            if param_type == Searchspace.DOUBLE:
                for b in (lower, upper):
                    if not isinstance(b, (int, float)):
                        raise ValueError(
                            "Bounds of DOUBLE parameter {} must be numeric: "
                            "{}".format(name, param_values)
                        )
            else:
                for b in (lower, upper):
                    if not isinstance(b, int):
                        raise ValueError(
                            "Bounds of INTEGER parameter {} must be int: "
                            "{}".format(name, param_values)
                        )
            if lower >= upper:
                raise ValueError(
                    "Lower bound must be smaller than upper bound for {}: "
                    "{}".format(name, param_values)
                )

        self._hparam_types[name] = param_type
        self._names.append(name)
        setattr(self, name, param_values)

    def names(self):
        """Return ``{name: type}`` in insertion order."""
        return {n: self._hparam_types[n] for n in self._names}

    def get(self, name, default=None):
        if name in self._hparam_types:
            return getattr(self, name)
        return default

    def keys(self):
        return list(self._names)

    def values(self):
        return [getattr(self, n) for n in self._names]

    def items(self):
        """List of ``{"name", "type", "values"}`` dicts in insertion order."""
        return [
            {"name": n, "type": self._hparam_types[n], "values": getattr(self, n)}
            for n in self._names
        ]

    def get_random_parameter_values(self, num):
        """Draw ``num`` random configurations as param dicts."""
        if not isinstance(num, int):
            raise ValueError("num has to be an int: {}".format(num))
        out = []
        for _ in range(num):
            params = {}
            for name, ptype in self.names().items():
                region = self.get(name)
                if ptype == Searchspace.DOUBLE:
                    params[name] = random.uniform(region[0], region[1])
                elif ptype == Searchspace.INTEGER:
                    params[name] = random.randint(region[0], region[1])
                else:  # DISCRETE / CATEGORICAL
                    params[name] = random.choice(region)
            out.append(params)
        return out

    # ------------------------------------------------------------------
    # unit-hypercube transforms (used by GP / TPE surrogates)
    # ------------------------------------------------------------------

    @staticmethod
    def _normalize_scalar(bounds, value):
        lo, hi = float(bounds[0]), float(bounds[1])
        return (float(value) - lo) / (hi - lo)

    @staticmethod
    def _inverse_normalize_scalar(bounds, norm):
        lo, hi = float(bounds[0]), float(bounds[1])
        return float(norm) * (hi - lo) + lo

    @staticmethod
    def _normalize_integer(bounds, value):
        lo, hi = int(bounds[0]), int(bounds[1])
        if hi == lo:
            return 0.0
        return (int(round(value)) - lo) / float(hi - lo)

    @staticmethod
    def _inverse_normalize_integer(bounds, norm):
        lo, hi = int(bounds[0]), int(bounds[1])
        return int(round(float(norm) * (hi - lo) + lo))

    @staticmethod
    def _encode_categorical(values, value):
        return list(values).index(value)

    @staticmethod
    def _decode_categorical(values, index):
        return list(values)[int(round(index))]

    def transform(self, hparams, normalize_categorical=False):
        """Transform one configuration (list in searchspace order) to the
        unit hypercube: min-max for DOUBLE/INTEGER, index (optionally
        normalized) for CATEGORICAL/DISCRETE."""
        transformed = []
        for hparam, spec in zip(hparams, self.items()):
            if spec["type"] == Searchspace.DOUBLE:
                transformed.append(self._normalize_scalar(spec["values"], hparam))
            elif spec["type"] == Searchspace.INTEGER:
                transformed.append(self._normalize_integer(spec["values"], hparam))
            elif spec["type"] in (Searchspace.CATEGORICAL, Searchspace.DISCRETE):
                enc = self._encode_categorical(spec["values"], hparam)
                if normalize_categorical:
                    enc = self._normalize_integer([0, len(spec["values"]) - 1], enc)
                transformed.append(enc)
            else:
                raise NotImplementedError(
                    "Unknown type {}".format(spec["type"])
                )
        return transformed

    def inverse_transform(self, transformed_hparams, normalize_categorical=False):
        """Inverse of :meth:`transform`."""
        hparams = []
        for hparam, spec in zip(transformed_hparams, self.items()):
            if spec["type"] == Searchspace.DOUBLE:
                hparams.append(self._inverse_normalize_scalar(spec["values"], hparam))
            elif spec["type"] == Searchspace.INTEGER:
                hparams.append(self._inverse_normalize_integer(spec["values"], hparam))
            elif spec["type"] in (Searchspace.CATEGORICAL, Searchspace.DISCRETE):
                if normalize_categorical:
                    idx = self._inverse_normalize_integer(
                        [0, len(spec["values"]) - 1], hparam
                    )
                else:
                    idx = hparam
                hparams.append(self._decode_categorical(spec["values"], idx))
            else:
                raise NotImplementedError(
                    "Unknown type {}".format(spec["type"])
                )
        return hparams

    def list_to_dict(self, hparams_list):
        """Convert a configuration list (searchspace order) to a dict."""
        if len(hparams_list) != len(self._names):
            raise ValueError(
                "hparams_list has {} entries, searchspace has {}".format(
                    len(hparams_list), len(self._names)
                )
            )
        return dict(zip(self._names, hparams_list))

    def dict_to_list(self, hparams_dict):
        """Convert a configuration dict to a list in searchspace order."""
        return [hparams_dict[n] for n in self._names]

    def to_dict(self):
        return {
            n: {"type": self._hparam_types[n], "values": self.get(n)}
            for n in self._names
        }

    def json(self):
        return json.dumps(self.to_dict(), sort_keys=True)

    def __str__(self):
        return self.json()

    def __iter__(self):
        # iterate (name, type, values) specs
        for spec in self.items():
            yield spec

    def __contains__(self, name):
        return name in self._hparam_types

    def sample(self, rng=None):
        """Draw one random configuration using an optional ``numpy`` RNG for
        deterministic tests."""
        if rng is None:
            return self.get_random_parameter_values(1)[0]
        params = {}
        for name, ptype in self.names().items():
            region = self.get(name)
            if ptype == Searchspace.DOUBLE:
                params[name] = float(rng.uniform(region[0], region[1]))
            elif ptype == Searchspace.INTEGER:
                params[name] = int(rng.integers(region[0], region[1] + 1))
            else:
                params[name] = region[int(rng.integers(0, len(region)))]
        return params
