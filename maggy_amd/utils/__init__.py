from maggy_amd.utils.jsonutil import json_default_numpy  # noqa: F401
